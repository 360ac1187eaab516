"""Spark-ML Pipeline API layer (parity with reference
sparkflow/tensorflow_async.py, pipeline_util.py, tensorflow_model_loader.py).

When a real pyspark is installed, this layer rides it unchanged.  When
it is not (the MI355X single-node image has no JVM), the bundled
``sparkflow_amd.localspark`` runtime registers itself under the pyspark
names the first time this package is imported, so ``SparkAsyncDL.fit``/
``transform``/pipeline save-load run for real either way.
"""
from sparkflow_amd import localspark as _localspark

_localspark.install(quiet=True)

__all__ = [
    "SparkAsyncDL",
    "SparkAsyncDLModel",
    "build_optimizer",
    "PysparkPipelineWrapper",
    "PysparkReaderWriter",
    "PysparkObjId",
    "load_torch_model",
    "attach_torch_model_to_pipeline",
]


def __getattr__(name):
    if name in ("SparkAsyncDL", "SparkAsyncDLModel", "build_optimizer"):
        from sparkflow_amd.spark import async_dl

        return getattr(async_dl, name)
    if name in ("PysparkPipelineWrapper", "PysparkReaderWriter", "PysparkObjId"):
        from sparkflow_amd.spark import pipeline_util

        return getattr(pipeline_util, name)
    if name in ("load_torch_model", "attach_torch_model_to_pipeline"):
        from sparkflow_amd.spark import model_loader

        return getattr(model_loader, name)
    raise AttributeError(name)
