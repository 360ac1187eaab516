"""pyspark.ml.feature subset: the transformers SparkFlow programs use.

StopWordsRemover exists here chiefly as the PERSISTENCE CARRIER: the
reference smuggles dill'd python stages through a Java StopWordsRemover's
stopwords (reference pipeline_util.py:30-31, 109-127), so this class is
what ``PysparkObjId._getCarrierClass()`` resolves to and what pipeline
load must hand back for ``unwrap`` to detect.
"""
from __future__ import annotations

import numpy as np

from sparkflow_amd.localspark.ml_base import Transformer
from sparkflow_amd.localspark.ml_linalg import DenseVector, SparseVector
from sparkflow_amd.localspark.ml_param import (
    HasInputCol,
    HasInputCols,
    HasOutputCol,
    Param,
    Params,
    TypeConverters,
    keyword_only,
)
from sparkflow_amd.localspark.ml_util import register_java_class
from sparkflow_amd.localspark.sql import Row


class StopWordsRemover(Transformer, HasInputCol, HasOutputCol):
    _java_class = "org.apache.spark.ml.feature.StopWordsRemover"

    stopWords = Param(Params._dummy(), "stopWords", "words to filter out",
                      typeConverter=TypeConverters.toListString)
    caseSensitive = Param(Params._dummy(), "caseSensitive",
                          "case sensitive matching",
                          typeConverter=TypeConverters.toBoolean)

    @keyword_only
    def __init__(self, inputCol=None, outputCol=None, stopWords=None,
                 caseSensitive=False):
        super().__init__()
        self._setDefault(caseSensitive=False, stopWords=[])
        kwargs = self._input_kwargs
        self._set(**{k: v for k, v in kwargs.items() if v is not None})

    def setStopWords(self, value):
        return self._set(stopWords=list(value))

    def getStopWords(self):
        return self.getOrDefault(self.stopWords)

    def _transform(self, dataset):
        stop = set(self.getStopWords() or [])
        if not self.getOrDefault(self.caseSensitive):
            stop = {w.lower() for w in stop}

        inp, out = self.getInputCol(), self.getOutputCol()

        def f(row):
            words = row[inp]
            kept = [w for w in words
                    if (w if self.getOrDefault(self.caseSensitive)
                        else w.lower()) not in stop]
            return kept

        return dataset.withColumn(out, f)


register_java_class(StopWordsRemover._java_class, StopWordsRemover)


def _as_values(v):
    if isinstance(v, (DenseVector, SparseVector)):
        return np.asarray(v, dtype=np.float64)
    if isinstance(v, (list, tuple, np.ndarray)):
        return np.asarray(v, dtype=np.float64).reshape(-1)
    return np.asarray([float(v)])


class VectorAssembler(Transformer, HasInputCols, HasOutputCol):
    _java_class = "org.apache.spark.ml.feature.VectorAssembler"

    @keyword_only
    def __init__(self, inputCols=None, outputCol=None):
        super().__init__()
        kwargs = self._input_kwargs
        self._set(**{k: v for k, v in kwargs.items() if v is not None})

    def _transform(self, dataset):
        cols = self.getInputCols()
        out = self.getOutputCol()

        def f(row):
            return DenseVector(np.concatenate(
                [_as_values(row[c]) for c in cols]))

        return dataset.withColumn(out, f)


register_java_class(VectorAssembler._java_class, VectorAssembler)


class OneHotEncoder(Transformer, HasInputCol, HasOutputCol):
    """Spark 2.x legacy OneHotEncoder: numeric index column -> one-hot
    SparseVector; dropLast=True (default) drops the final category."""

    _java_class = "org.apache.spark.ml.feature.OneHotEncoder"

    dropLast = Param(Params._dummy(), "dropLast", "drop the last category",
                     typeConverter=TypeConverters.toBoolean)

    @keyword_only
    def __init__(self, inputCol=None, outputCol=None, dropLast=True):
        super().__init__()
        self._setDefault(dropLast=True)
        kwargs = self._input_kwargs
        self._set(**{k: v for k, v in kwargs.items() if v is not None})

    def _transform(self, dataset):
        inp, out = self.getInputCol(), self.getOutputCol()
        drop_last = self.getOrDefault(self.dropLast)
        n_cat = 1 + max((int(r[inp]) for r in dataset.collect()), default=0)
        size = n_cat - 1 if drop_last else n_cat

        def f(row):
            idx = int(row[inp])
            if idx < size:
                return SparseVector(size, [idx], [1.0])
            return SparseVector(size, [], [])

        return dataset.withColumn(out, f)


register_java_class(OneHotEncoder._java_class, OneHotEncoder)


class Normalizer(Transformer, HasInputCol, HasOutputCol):
    _java_class = "org.apache.spark.ml.feature.Normalizer"

    p = Param(Params._dummy(), "p", "norm order",
              typeConverter=TypeConverters.toFloat)

    @keyword_only
    def __init__(self, p=2.0, inputCol=None, outputCol=None):
        super().__init__()
        self._setDefault(p=2.0)
        kwargs = self._input_kwargs
        self._set(**{k: v for k, v in kwargs.items() if v is not None})

    def _transform(self, dataset):
        inp, out = self.getInputCol(), self.getOutputCol()
        order = self.getOrDefault(self.p)

        def f(row):
            v = _as_values(row[inp])
            nrm = np.linalg.norm(v, ord=order)
            return DenseVector(v / nrm if nrm > 0 else v)

        return dataset.withColumn(out, f)


register_java_class(Normalizer._java_class, Normalizer)
