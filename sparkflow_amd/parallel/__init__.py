from sparkflow_amd.parallel.rwlock import RWLock  # noqa: F401
from sparkflow_amd.parallel.hogwild import HogwildTrainer, HogwildSparkModel  # noqa: F401
