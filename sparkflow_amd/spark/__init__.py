"""Spark-ML Pipeline API layer (parity with reference
sparkflow/tensorflow_async.py, pipeline_util.py, tensorflow_model_loader.py).

pyspark is imported lazily: the compute engine and kernels work without it
(local/torchrun modes); installing pyspark enables this layer unchanged.
"""

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
