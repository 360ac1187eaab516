"""pyspark.ml.wrapper shim: JavaParams._new_java_obj.

The persistence mixin (sparkflow_amd/spark/pipeline_util.py ``_to_java``)
constructs its StopWordsRemover carrier through this pyspark entry point;
here it resolves directly to the localspark class."""
from __future__ import annotations

from sparkflow_amd.localspark.ml_param import Params
from sparkflow_amd.localspark.ml_util import JAVA_CLASS_REGISTRY


class JavaParams(Params):
    @staticmethod
    def _new_java_obj(java_class: str, *args):
        cls = JAVA_CLASS_REGISTRY.get(java_class)
        if cls is None:
            raise ValueError("unknown java class %s" % java_class)
        obj = cls()
        if args:  # first ctor arg is the uid (pyspark convention)
            obj.uid = str(args[0])
        return obj


class JavaWrapper(JavaParams):
    pass
