"""pyspark.ml.param subset: Param / Params / TypeConverters /
keyword_only, plus the shared-param mixins (pyspark.ml.param.shared)
SparkFlow's estimator uses (HasInputCol / HasLabelCol / HasPredictionCol).

Semantics follow pyspark: Param objects are declared on the CLASS with a
dummy parent and re-bound to each instance's uid in ``_copy_params``;
``_set`` runs the type converter; ``getOrDefault`` consults the explicit
paramMap then the defaults.
"""
from __future__ import annotations

import functools
import uuid
from typing import Any, Callable, Optional


def keyword_only(func: Callable) -> Callable:
    """Store the caller's kwargs in ``self._input_kwargs`` (pyspark
    contract used by every Params ctor/setParams)."""

    @functools.wraps(func)
    def wrapper(self, *args, **kwargs):
        if args:
            raise TypeError("Method %s forces keyword arguments."
                            % func.__name__)
        self._input_kwargs = kwargs
        return func(self, **kwargs)

    return wrapper


class Param:
    def __init__(self, parent, name: str, doc: str = "",
                 typeConverter: Optional[Callable] = None):
        self.parent = getattr(parent, "uid", parent)
        self.name = name
        self.doc = doc
        self.typeConverter = typeConverter or (lambda v: v)

    def _copy_new_parent(self, parent) -> "Param":
        return Param(parent, self.name, self.doc, self.typeConverter)

    def __repr__(self):
        return "Param(parent=%r, name=%r)" % (self.parent, self.name)

    def __hash__(self):
        return hash((self.parent, self.name))

    def __eq__(self, other):
        return (isinstance(other, Param) and self.parent == other.parent
                and self.name == other.name)


class TypeConverters:
    @staticmethod
    def _null_ok(f):
        def conv(v):
            return None if v is None else f(v)

        return conv

    toString = _null_ok.__func__(str)
    toInt = _null_ok.__func__(int)
    toFloat = _null_ok.__func__(float)
    toBoolean = _null_ok.__func__(bool)

    @staticmethod
    def toList(v):
        return None if v is None else list(v)

    @staticmethod
    def toListString(v):
        return None if v is None else [str(x) for x in v]

    @staticmethod
    def toListFloat(v):
        return None if v is None else [float(x) for x in v]

    @staticmethod
    def identity(v):
        return v


class Identifiable:
    """pyspark.ml.util.Identifiable (re-exported there)."""

    def __init__(self):
        self.uid = self._randomUID()

    @classmethod
    def _randomUID(cls) -> str:
        return "%s_%s" % (cls.__name__, uuid.uuid4().hex[12:])

    def __repr__(self):
        return self.uid


class Params(Identifiable):
    @staticmethod
    def _dummy() -> "Params":
        p = Params.__new__(Params)
        p.uid = "undefined"
        return p

    def __init__(self):
        super().__init__()
        self._paramMap = {}
        self._defaultParamMap = {}
        self._params = None
        self._copy_params()

    def _copy_params(self):
        """Re-bind class-level Param declarations to this instance."""
        for name in dir(type(self)):
            v = getattr(type(self), name, None)
            if isinstance(v, Param):
                setattr(self, name, v._copy_new_parent(self))

    @property
    def params(self):
        if self._params is None:
            self._params = sorted(
                (getattr(self, n) for n in dir(self)
                 if n != "params"
                 and isinstance(getattr(self, n, None), Param)),
                key=lambda p: p.name)
        return self._params

    def getParam(self, name: str) -> Param:
        p = getattr(self, name, None)
        if not isinstance(p, Param):
            raise ValueError("no param named %s" % name)
        return p

    def _resolveParam(self, param) -> Param:
        return param if isinstance(param, Param) else self.getParam(param)

    def _set(self, **kwargs) -> "Params":
        for name, value in kwargs.items():
            p = self.getParam(name)
            self._paramMap[p.name] = p.typeConverter(value)
        return self

    set = _set

    def _setDefault(self, **kwargs) -> "Params":
        for name, value in kwargs.items():
            p = self.getParam(name)
            self._defaultParamMap[p.name] = value
        return self

    def getOrDefault(self, param) -> Any:
        p = self._resolveParam(param)
        if p.name in self._paramMap:
            return self._paramMap[p.name]
        return self._defaultParamMap.get(p.name)

    def isSet(self, param) -> bool:
        return self._resolveParam(param).name in self._paramMap

    def hasDefault(self, param) -> bool:
        return self._resolveParam(param).name in self._defaultParamMap

    def isDefined(self, param) -> bool:
        return self.isSet(param) or self.hasDefault(param)

    def hasParam(self, name: str) -> bool:
        return isinstance(getattr(self, name, None), Param)

    def extractParamMap(self, extra=None) -> dict:
        out = dict(self._defaultParamMap)
        out.update(self._paramMap)
        if extra:
            out.update(extra)
        return out

    def copy(self, extra=None) -> "Params":
        import copy as _copy

        that = _copy.copy(self)
        that._paramMap = dict(self._paramMap)
        that._defaultParamMap = dict(self._defaultParamMap)
        if extra:
            for k, v in extra.items():
                p = that._resolveParam(k)
                that._paramMap[p.name] = v
        return that

    # pickling: Param objects and maps are plain data — default works
    def __getstate__(self):
        d = dict(self.__dict__)
        d["_params"] = None
        return d

    def __setstate__(self, state):
        self.__dict__.update(state)


# ---------------------------------------------------------------------------
# pyspark.ml.param.shared mixins
# ---------------------------------------------------------------------------


class HasInputCol(Params):
    inputCol = Param(Params._dummy(), "inputCol", "input column name.",
                     typeConverter=TypeConverters.toString)

    def getInputCol(self):
        return self.getOrDefault(self.inputCol)

    def setInputCol(self, value):
        return self._set(inputCol=value)


class HasInputCols(Params):
    inputCols = Param(Params._dummy(), "inputCols", "input column names.",
                      typeConverter=TypeConverters.toListString)

    def getInputCols(self):
        return self.getOrDefault(self.inputCols)

    def setInputCols(self, value):
        return self._set(inputCols=value)


class HasOutputCol(Params):
    outputCol = Param(Params._dummy(), "outputCol", "output column name.",
                      typeConverter=TypeConverters.toString)

    def getOutputCol(self):
        return self.getOrDefault(self.outputCol)

    def setOutputCol(self, value):
        return self._set(outputCol=value)


class HasLabelCol(Params):
    labelCol = Param(Params._dummy(), "labelCol", "label column name.",
                     typeConverter=TypeConverters.toString)

    def getLabelCol(self):
        return self.getOrDefault(self.labelCol)

    def setLabelCol(self, value):
        return self._set(labelCol=value)


class HasPredictionCol(Params):
    predictionCol = Param(Params._dummy(), "predictionCol",
                          "prediction column name.",
                          typeConverter=TypeConverters.toString)

    def getPredictionCol(self):
        return self.getOrDefault(self.predictionCol)

    def setPredictionCol(self, value):
        return self._set(predictionCol=value)
