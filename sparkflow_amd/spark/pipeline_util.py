"""Pipeline persistence — byte-compatible with reference
sparkflow/pipeline_util.py.

On-disk format (MUST stay byte-compatible, BASELINE.json north star): a
Python stage is dill-dumped, zlib-compressed, rendered as a single string
of comma-separated byte integers ("72,105,..." with a trailing comma),
stored together with the GUID sentinel ``4c1740b00d3c4ff6806a1402321572cb``
as the stopwords of a Java ``StopWordsRemover``, which Spark's native
pipeline writer persists.  Loading detects the carrier by the sentinel and
reverses the encoding.

The pure byte codec (``encode_py_obj`` / ``decode_py_obj``) has no Spark
dependency and is unit-tested without a JVM.
"""
from __future__ import annotations

import zlib
from typing import List

PY_OBJ_GUID = "4c1740b00d3c4ff6806a1402321572cb"


# ---------------------------------------------------------------------------
# pure codec (no pyspark needed)
# ---------------------------------------------------------------------------


def encode_py_obj(obj) -> List[str]:
    """Object -> the two-element stopwords list [payload, GUID]."""
    import dill

    raw = zlib.compress(dill.dumps(obj))
    payload = "".join(str(b) + "," for b in bytearray(raw))
    return [payload, PY_OBJ_GUID]


def decode_py_obj(stop_words: List[str]):
    """Stopwords (sentinel already stripped or not) -> original object."""
    import dill

    words = list(stop_words)
    if words and words[-1] == PY_OBJ_GUID:
        words = words[:-1]
    ints = words[0].split(",")[0:-1]  # trailing comma produces final ''
    raw = zlib.decompress(bytes(int(i) for i in ints))
    return dill.loads(raw)


# kept under the reference's name for drop-in familiarity
def load_byte_array(stop_words):
    return decode_py_obj(stop_words)


# ---------------------------------------------------------------------------
# Spark-facing classes
# ---------------------------------------------------------------------------


class PysparkObjId:
    """Constants identifying our smuggled python stages inside Spark-native
    pipeline files (parity: reference pipeline_util.py:17-31)."""

    @staticmethod
    def _getPyObjId():
        return PY_OBJ_GUID

    @staticmethod
    def _getCarrierClass(javaName=False):
        if javaName:
            return "org.apache.spark.ml.feature.StopWordsRemover"
        from pyspark.ml.feature import StopWordsRemover

        return StopWordsRemover


class PysparkPipelineWrapper:
    """Recursively restore python stages inside a loaded Pipeline /
    PipelineModel (parity: reference pipeline_util.py:48-74)."""

    @staticmethod
    def unwrap(pipeline):
        from pyspark.ml import Pipeline, PipelineModel

        if not isinstance(pipeline, (Pipeline, PipelineModel)):
            raise TypeError(
                "Cannot recognize a pipeline of type %s." % type(pipeline))
        stages = (pipeline.getStages() if isinstance(pipeline, Pipeline)
                  else pipeline.stages)
        carrier = PysparkObjId._getCarrierClass()
        for i, stage in enumerate(stages):
            if isinstance(stage, (Pipeline, PipelineModel)):
                stages[i] = PysparkPipelineWrapper.unwrap(stage)
            elif (isinstance(stage, carrier)
                  and stage.getStopWords()[-1] == PY_OBJ_GUID):
                stages[i] = decode_py_obj(stage.getStopWords()[:-1])
        if isinstance(pipeline, Pipeline):
            pipeline.setStages(stages)
        else:
            pipeline.stages = stages
        return pipeline


class PysparkReaderWriter:
    """Mixin giving custom python Estimators/Models Spark-native
    ``.write().save(path)`` / ``.load(path)`` through the StopWordsRemover
    carrier (parity: reference pipeline_util.py:77-127)."""

    def write(self):
        from pyspark.ml.util import JavaMLWriter

        return JavaMLWriter(self)

    @classmethod
    def read(cls):
        from pyspark.ml.util import JavaMLReader

        return JavaMLReader(PysparkObjId._getCarrierClass())

    @classmethod
    def load(cls, path):
        return cls._from_java(cls.read().load(path))

    @classmethod
    def _from_java(cls, java_obj):
        return decode_py_obj(java_obj.getStopWords()[:-1])

    def _to_java(self):
        from pyspark.context import SparkContext
        from pyspark.ml.wrapper import JavaParams

        words = encode_py_obj(self)
        sc = SparkContext._active_spark_context
        java_array = sc._gateway.new_array(sc._gateway.jvm.java.lang.String,
                                           len(words))
        java_array[0:2] = words[0:2]
        java_obj = JavaParams._new_java_obj(
            PysparkObjId._getCarrierClass(javaName=True), self.uid)
        java_obj.setStopWords(java_array)
        return java_obj
