from sparkflow_amd.utils import serialization  # noqa: F401
