"""pyspark.ml.util subset: persistence machinery.

On-disk layout = Spark's own (DefaultParamsWriter /
PipelineSharedReadWrite, Spark 2.4 line):

    path/metadata/part-00000   one JSON object: {"class": <java class>,
                               "timestamp": ms, "sparkVersion": ...,
                               "uid": ..., "paramMap": {...}}
    path/metadata/_SUCCESS
    path/stages/<idx>_<uid>/   per-stage subdirs (pipelines)

Python stages ride inside a StopWordsRemover carrier exactly as the
reference smuggles them through the JVM writer (reference
pipeline_util.py:109-127): the stage's ``_to_java()`` produces the
carrier, whose stopwords (dill+zlib payload + GUID) land in the
metadata paramMap — so a directory written here is laid out like one
written by Spark 2.4 itself.
"""
from __future__ import annotations

import json
import os
import shutil
import time

from sparkflow_amd.localspark.ml_param import Identifiable  # noqa: F401
from sparkflow_amd.localspark.ml_param import Params

SPARK_VERSION = "2.4.3+localspark"

# java-class-name -> localspark class; populated by ml_feature/ml_pipeline
JAVA_CLASS_REGISTRY = {}


def register_java_class(name: str, cls) -> None:
    JAVA_CLASS_REGISTRY[name] = cls


def _json_safe(v):
    import numpy as np

    if isinstance(v, (str, int, float, bool)) or v is None:
        return v
    if isinstance(v, (list, tuple)):
        return [_json_safe(x) for x in v]
    if isinstance(v, np.generic):
        return v.item()
    raise TypeError("param value %r is not persistable" % (v,))


def write_metadata(path: str, java_class: str, uid: str, param_map: dict,
                   extra: dict = None) -> None:
    meta_dir = os.path.join(path, "metadata")
    os.makedirs(meta_dir, exist_ok=True)
    meta = {
        "class": java_class,
        "timestamp": int(time.time() * 1000),
        "sparkVersion": SPARK_VERSION,
        "uid": uid,
        "paramMap": {k: _json_safe(v) for k, v in param_map.items()
                     if v is not None},
    }
    if extra:
        meta.update(extra)
    with open(os.path.join(meta_dir, "part-00000"), "w") as f:
        f.write(json.dumps(meta) + "\n")
    open(os.path.join(meta_dir, "_SUCCESS"), "w").close()


def read_metadata(path: str) -> dict:
    with open(os.path.join(path, "metadata", "part-00000")) as f:
        return json.loads(f.readline())


def save_stage(obj, path: str) -> None:
    """Write one stage (or nested pipeline) in the Spark layout."""
    from sparkflow_amd.localspark.ml_pipeline import (
        Pipeline,
        PipelineModel,
        save_pipeline,
    )

    if isinstance(obj, (Pipeline, PipelineModel)):
        save_pipeline(obj, path)
        return
    if hasattr(obj, "_to_java") and not hasattr(obj, "_java_class"):
        obj = obj._to_java()  # python stage -> carrier
    java_class = getattr(obj, "_java_class", None)
    if java_class is None:
        raise TypeError("stage %r has no persistence mapping" % (obj,))
    write_metadata(path, java_class, obj.uid, obj.extractParamMap())


def load_stage(path: str):
    """Read one stage dir back into a localspark object (carriers stay
    carriers — PysparkPipelineWrapper.unwrap does the python decoding,
    reference flow pipeline_util.py:56-74)."""
    from sparkflow_amd.localspark.ml_pipeline import load_pipeline

    meta = read_metadata(path)
    cls_name = meta["class"]
    if cls_name in ("org.apache.spark.ml.Pipeline",
                    "org.apache.spark.ml.PipelineModel"):
        return load_pipeline(path, meta)
    cls = JAVA_CLASS_REGISTRY.get(cls_name)
    if cls is None:
        raise ValueError("cannot load stage class %s" % cls_name)
    obj = cls.__new__(cls)
    Params.__init__(obj)
    obj.uid = meta["uid"]
    obj._set(**{k: v for k, v in meta.get("paramMap", {}).items()
                if obj.hasParam(k)})
    return obj


def _prepare_path(path: str, overwrite: bool) -> None:
    if os.path.exists(path):
        if not overwrite:
            raise IOError("Path %s already exists (use .overwrite())" % path)
        shutil.rmtree(path)


class MLWriter:
    def __init__(self, instance=None):
        self._instance = instance
        self._overwrite = False

    def overwrite(self) -> "MLWriter":
        self._overwrite = True
        return self

    def save(self, path: str) -> None:
        _prepare_path(path, self._overwrite)
        save_stage(self._instance, path)


class JavaMLWriter(MLWriter):
    """Writer handed out by PysparkReaderWriter.write() — same name and
    call shape as pyspark's (reference pipeline_util.py:85-87)."""


class MLReader:
    def __init__(self, clazz=None):
        self._clazz = clazz

    def load(self, path: str):
        return load_stage(path)


class JavaMLReader(MLReader):
    pass


class MLWritable:
    def write(self) -> MLWriter:
        return MLWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)


class MLReadable:
    @classmethod
    def read(cls) -> MLReader:
        return MLReader(cls)

    @classmethod
    def load(cls, path: str):
        return cls.read().load(path)


class DefaultParamsWriter(MLWriter):
    pass


class DefaultParamsReader(MLReader):
    pass
