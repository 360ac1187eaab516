"""SparkContext / SparkConf / TaskContext for localspark, plus the tiny
py4j-gateway shim that lets the reference-shaped persistence code
(sparkflow_amd/spark/pipeline_util.py ``_to_java``) run unchanged."""
from __future__ import annotations

import re
import socket
from typing import Optional


class SparkConf:
    def __init__(self, conf: Optional[dict] = None):
        self._conf = dict(conf or {})

    def get(self, key: str, default=None):
        if key == "spark.driver.host" and key not in self._conf:
            try:
                return socket.gethostbyname(socket.gethostname())
            except OSError:
                return "127.0.0.1"
        return self._conf.get(key, default)

    def set(self, key: str, value):
        self._conf[key] = value
        return self

    def getAll(self):
        return list(self._conf.items())


class TaskContext:
    """Partition identity inside a task (pyspark.TaskContext.get())."""

    _current: Optional["TaskContext"] = None

    def __init__(self, partition_id: int = 0):
        self._partition_id = partition_id

    def partitionId(self) -> int:
        return self._partition_id

    @classmethod
    def get(cls) -> Optional["TaskContext"]:
        return cls._current


class _JvmShim:
    """Attribute chain sink: ``gateway.jvm.java.lang.String`` etc."""

    def __getattr__(self, name):
        return self

    def __call__(self, *a, **k):
        return self


class _GatewayShim:
    """The one py4j surface the persistence path touches:
    ``sc._gateway.new_array(jvm.java.lang.String, n)`` — returns a plain
    python list of that length (slice-assignable, like a JavaArray)."""

    jvm = _JvmShim()

    def new_array(self, _jtype, n: int):
        return [None] * int(n)


class SparkContext:
    _active_spark_context: Optional["SparkContext"] = None

    def __init__(self, master: str = "local[*]", appName: str = "localspark",
                 conf: Optional[SparkConf] = None):
        self.master = master
        self.appName = appName
        self._conf = conf or SparkConf()
        self._gateway = _GatewayShim()
        self._jvm = self._gateway.jvm
        m = re.match(r"local(proc)?\[(\d+|\*)\]", master or "")
        self.process_mode = bool(m and m.group(1))
        if m and m.group(2) != "*":
            self.local_threads = int(m.group(2))
        else:
            import os

            self.local_threads = os.cpu_count() or 2
        SparkContext._active_spark_context = self

    @property
    def defaultParallelism(self) -> int:
        return self.local_threads

    def getConf(self) -> SparkConf:
        return self._conf

    def parallelize(self, data, numSlices: Optional[int] = None):
        from sparkflow_amd.localspark.rdd import RDD, _split

        class _Sess:  # minimal session facade for RDDs made from sc
            sparkContext = self

        rdd = RDD(_split(list(data), numSlices or self.defaultParallelism),
                  _Sess())
        return rdd

    def stop(self):
        SparkContext._active_spark_context = None

    def setLogLevel(self, level: str):
        pass
