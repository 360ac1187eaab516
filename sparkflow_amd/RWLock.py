"""Drop-in parity module for reference ``sparkflow.RWLock``."""
from sparkflow_amd.parallel.rwlock import RWLock  # noqa: F401
