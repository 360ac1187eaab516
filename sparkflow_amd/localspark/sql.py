"""pyspark.sql subset: Row, DataFrame, SparkSession (builder pattern).

DataFrames are driver-resident row lists partitioned Spark-style; every
action on their RDD view runs partitions CONCURRENTLY (threads for
``local[N]``, spawned processes for ``localproc[N]``) — the concurrency
semantics the async-PS engine is tested against.
"""
from __future__ import annotations

import random as _random
from typing import List, Optional

from sparkflow_amd.localspark import context as _context
from sparkflow_amd.localspark import rdd as _rdd


class Row(tuple):
    """pyspark Row: a tuple with named fields.  ``Row(**kwargs)`` sorts
    field names alphabetically — Spark 2.x behavior (the reference pins
    Spark 2.4.3), which fixes the column order of ``rdd.toDF()``."""

    def __new__(cls, *args, **kwargs):
        if args and kwargs:
            raise ValueError("Cannot use both args and kwargs")
        if kwargs:
            names = sorted(kwargs.keys())
            row = super().__new__(cls, tuple(kwargs[n] for n in names))
            row.__fields__ = names
            return row
        row = super().__new__(cls, args)
        row.__fields__ = None
        return row

    @classmethod
    def _make(cls, names: List[str], values) -> "Row":
        row = tuple.__new__(cls, tuple(values))
        row.__fields__ = list(names)
        return row

    def __getitem__(self, item):
        if isinstance(item, str):
            if not self.__fields__:
                raise ValueError("Row has no named fields")
            try:
                return tuple.__getitem__(self, self.__fields__.index(item))
            except ValueError:
                raise KeyError(item)
        return tuple.__getitem__(self, item)

    def __getattr__(self, name):
        if name.startswith("__"):
            raise AttributeError(name)
        fields = tuple.__getattribute__(self, "__fields__")
        if fields and name in fields:
            return tuple.__getitem__(self, fields.index(name))
        raise AttributeError(name)

    def asDict(self, recursive: bool = False) -> dict:
        if not self.__fields__:
            raise ValueError("Row has no named fields")
        return dict(zip(self.__fields__, self))

    def __reduce__(self):
        if self.__fields__:
            return (_row_from_parts, (self.__fields__, tuple(self)))
        return (Row, tuple(self))

    def __repr__(self):
        if self.__fields__:
            return "Row(%s)" % ", ".join(
                "%s=%r" % (k, v) for k, v in zip(self.__fields__, self))
        return "<Row(%s)>" % ", ".join(repr(v) for v in self)


def _row_from_parts(names, values):
    return Row._make(names, values)


class DataFrame:
    def __init__(self, rows: List[Row], columns: List[str], session,
                 num_partitions: Optional[int] = None):
        self._rows = list(rows)
        self._columns = list(columns)
        self._session = session
        self._nparts = num_partitions or max(
            1, session.sparkContext.defaultParallelism)

    # -- metadata ------------------------------------------------------------

    @property
    def columns(self) -> List[str]:
        return list(self._columns)

    @property
    def rdd(self) -> "_rdd.RDD":
        return _rdd.RDD.from_rows(self._rows, self._nparts, self._session)

    # -- transformations ------------------------------------------------------

    def select(self, *cols):
        if len(cols) == 1 and isinstance(cols[0], (list, tuple)):
            cols = tuple(cols[0])
        names = [c for c in cols]
        rows = [Row._make(names, [r[c] for c in names]) for r in self._rows]
        return DataFrame(rows, names, self._session, self._nparts)

    def withColumn(self, name, values_fn):
        """Minimal: values_fn is a plain callable row->value."""
        cols = self._columns + ([name] if name not in self._columns else [])
        rows = []
        for r in self._rows:
            d = r.asDict()
            d[name] = values_fn(r)
            rows.append(Row._make(cols, [d[c] for c in cols]))
        return DataFrame(rows, cols, self._session, self._nparts)

    def orderBy(self, *exprs):
        from sparkflow_amd.localspark.sql_functions import RandColumn

        rows = list(self._rows)
        if len(exprs) == 1 and isinstance(exprs[0], RandColumn):
            rng = _random.Random(exprs[0].seed)
            rng.shuffle(rows)
        elif exprs:
            keys = [e for e in exprs]
            rows.sort(key=lambda r: tuple(r[k] for k in keys))
        return DataFrame(rows, self._columns, self._session, self._nparts)

    def repartition(self, n: int) -> "DataFrame":
        return DataFrame(self._rows, self._columns, self._session, n)

    def coalesce(self, n: int) -> "DataFrame":
        return DataFrame(self._rows, self._columns, self._session,
                         min(n, self._nparts))

    def limit(self, n: int) -> "DataFrame":
        return DataFrame(self._rows[:n], self._columns, self._session,
                         self._nparts)

    def cache(self):
        return self

    persist = cache

    # -- actions --------------------------------------------------------------

    def collect(self) -> List[Row]:
        return list(self._rows)

    def count(self) -> int:
        return len(self._rows)

    def take(self, n: int) -> List[Row]:
        return self._rows[:n]

    def first(self) -> Optional[Row]:
        return self._rows[0] if self._rows else None

    def head(self, n: Optional[int] = None):
        return self.first() if n is None else self.take(n)

    def show(self, n: int = 20, truncate: bool = True):
        print(" | ".join(self._columns))
        for r in self._rows[:n]:
            print(" | ".join(str(v) for v in r))

    def toPandas(self):
        import pandas as pd

        return pd.DataFrame(self._rows, columns=self._columns)


class SparkSession:
    _active: Optional["SparkSession"] = None

    class Builder:
        def __init__(self):
            self._conf = {}
            self._master = "local[*]"
            self._app = "localspark"

        def master(self, m: str):
            self._master = m
            return self

        def appName(self, name: str):
            self._app = name
            return self

        def config(self, key=None, value=None, conf=None):
            if key is not None:
                self._conf[key] = value
            return self

        def getOrCreate(self) -> "SparkSession":
            if SparkSession._active is not None:
                return SparkSession._active
            sc = _context.SparkContext(master=self._master,
                                       appName=self._app,
                                       conf=_context.SparkConf(self._conf))
            session = SparkSession(sc)
            SparkSession._active = session
            return session

    builder = None  # replaced below with a property-like hook

    def __init__(self, sparkContext: "_context.SparkContext"):
        self.sparkContext = sparkContext
        self._sc = sparkContext

    @property
    def read(self):
        return DataFrameReader(self)

    def createDataFrame(self, data, schema=None) -> DataFrame:
        rows = list(data)
        if schema is not None and not isinstance(schema, (list, tuple)):
            raise TypeError("localspark supports list-of-names schemas only")
        if schema is None:
            if rows and isinstance(rows[0], Row) and rows[0].__fields__:
                schema = rows[0].__fields__
            elif rows and isinstance(rows[0], dict):
                schema = sorted(rows[0].keys())
                rows = [tuple(r[k] for k in schema) for r in rows]
            else:
                raise ValueError("schema (column names) required")
        named = []
        for r in rows:
            vals = list(r) if isinstance(r, (tuple, list, Row)) else [r]
            named.append(Row._make(list(schema), vals))
        return DataFrame(named, list(schema), self)

    def stop(self):
        SparkSession._active = None
        self.sparkContext.stop()

    @property
    def conf(self):
        return self.sparkContext.getConf()


class DataFrameReader:
    """Minimal spark.read: headerless CSV with optional type inference
    (the reference examples' `spark.read.option("inferSchema", "true")
    .csv(path)` — columns named _c0.._cN like Spark)."""

    def __init__(self, session):
        self._session = session
        self._options = {}

    def option(self, key, value):
        self._options[str(key)] = value
        return self

    def csv(self, path) -> DataFrame:
        import csv as _csv

        infer = str(self._options.get("inferSchema", "false")).lower() \
            == "true"
        rows = []
        with open(path, newline="") as f:
            for rec in _csv.reader(f):
                if infer:
                    vals = []
                    for v in rec:
                        try:
                            vals.append(int(v))
                        except ValueError:
                            try:
                                vals.append(float(v))
                            except ValueError:
                                vals.append(v)
                    rows.append(tuple(vals))
                else:
                    rows.append(tuple(rec))
        ncol = len(rows[0]) if rows else 0
        cols = ["_c%d" % i for i in range(ncol)]
        return self._session.createDataFrame(rows, cols)


class _BuilderAccessor:
    """``SparkSession.builder`` must hand out a FRESH builder per access
    (pyspark's is a classproperty-like singleton; fresh is safer)."""

    def __get__(self, obj, objtype=None):
        return SparkSession.Builder()


SparkSession.builder = _BuilderAccessor()
