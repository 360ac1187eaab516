"""Drop-in parity module for reference ``sparkflow.tensorflow_async``:
``from sparkflow_amd.torch_async import SparkAsyncDL, SparkAsyncDLModel``.
Requires pyspark (the Spark-ML API host)."""
from sparkflow_amd.spark.async_dl import (  # noqa: F401
    SparkAsyncDL,
    SparkAsyncDLModel,
    build_optimizer,
    handle_data,
)
