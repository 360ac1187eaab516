"""pyspark RDD subset with Spark's execution semantics where they matter:

* data lives as a list of partitions;
* every ACTION runs one task per partition, CONCURRENTLY — threads under
  ``local[N]`` (N-slot pool, like Spark's local scheduler) or one
  spawned process per partition under ``localproc[N]`` (the executor
  shape: this is what lets SparkAsyncDL.fit place one rank per GPU and
  drive the RCCL star);
* closures shipped to process tasks are dill-serialized, like Spark's
  cloudpickle closure shipping.
"""
from __future__ import annotations

import random as _random
from concurrent.futures import ThreadPoolExecutor
from typing import Callable, List, Optional


def _proc_task_main(payload: bytes):
    """Entry point of a localproc executor task (spawned process)."""
    import dill

    fn, idx, part = dill.loads(payload)
    return dill.dumps(list(fn(idx, iter(part))))


class RDD:
    def __init__(self, partitions: List[list], session):
        self._parts = [list(p) for p in partitions]
        self._session = session

    # -- construction ---------------------------------------------------------

    @classmethod
    def from_rows(cls, rows: List, nparts: int, session) -> "RDD":
        return cls(_split(list(rows), nparts), session)

    # -- metadata -------------------------------------------------------------

    def getNumPartitions(self) -> int:
        return len(self._parts)

    # -- transformations (driver-side, eager like a materialized lineage) ----

    def map(self, f: Callable) -> "RDD":
        return RDD([[f(x) for x in p] for p in self._parts], self._session)

    def filter(self, f: Callable) -> "RDD":
        return RDD([[x for x in p if f(x)] for p in self._parts],
                   self._session)

    def flatMap(self, f: Callable) -> "RDD":
        return RDD([[y for x in p for y in f(x)] for p in self._parts],
                   self._session)

    def coalesce(self, n: int) -> "RDD":
        n = max(1, min(n, len(self._parts)))
        out = [[] for _ in range(n)]
        for i, p in enumerate(self._parts):
            out[i % n].extend(p)
        return RDD(out, self._session)

    def repartition(self, n: int) -> "RDD":
        rows = [x for p in self._parts for x in p]
        _random.shuffle(rows)  # Spark repartition = shuffle
        return RDD(_split(rows, max(1, n)), self._session)

    def cache(self):
        return self

    persist = cache

    # -- actions (concurrent per partition) -----------------------------------

    def collect(self) -> List:
        return [x for p in self._parts for x in p]

    def count(self) -> int:
        return sum(len(p) for p in self._parts)

    def first(self):
        for p in self._parts:
            if p:
                return p[0]
        raise ValueError("empty RDD")

    def take(self, n: int) -> List:
        out = []
        for p in self._parts:
            for x in p:
                if len(out) == n:
                    return out
                out.append(x)
        return out

    def foreachPartition(self, f: Callable) -> None:
        self._run_tasks(lambda idx, it: (f(it), [])[1])

    def mapPartitions(self, f: Callable) -> "RDD":
        results = self._run_tasks(lambda idx, it: list(f(it)))
        return RDD(results, self._session)

    def mapPartitionsWithIndex(self, f: Callable) -> "RDD":
        results = self._run_tasks(lambda idx, it: list(f(idx, it)))
        return RDD(results, self._session)

    def toDF(self, schema: Optional[List[str]] = None):
        from sparkflow_amd.localspark.sql import DataFrame, Row

        rows = self.collect()
        if schema is None:
            if rows and isinstance(rows[0], Row) and rows[0].__fields__:
                schema = rows[0].__fields__
            else:
                raise ValueError("toDF needs Rows with fields or a schema")
        named = [r if isinstance(r, Row) and r.__fields__
                 else Row._make(schema, list(r)) for r in rows]
        return DataFrame(named, list(schema), self._session,
                         num_partitions=len(self._parts))

    # -- task running ---------------------------------------------------------

    def _run_tasks(self, fn: Callable) -> List[list]:
        """Run fn(partition_index, iterator) per partition, concurrently.
        Task failures propagate to the driver (first error wins), like
        Spark failing the job."""
        sc = self._session.sparkContext
        if sc.process_mode:
            return self._run_tasks_procs(fn)
        slots = max(1, sc.local_threads)
        with ThreadPoolExecutor(max_workers=slots) as pool:
            futs = [pool.submit(fn, i, iter(p))
                    for i, p in enumerate(self._parts)]
            return [f.result() for f in futs]

    def _run_tasks_procs(self, fn: Callable) -> List[list]:
        import multiprocessing as mp

        import dill

        ctx = mp.get_context("spawn")
        slots = max(1, self._session.sparkContext.local_threads)
        results: List[Optional[list]] = [None] * len(self._parts)
        with ctx.Pool(processes=min(slots, max(1, len(self._parts)))) as pool:
            handles = [
                pool.apply_async(_proc_task_main,
                                 (dill.dumps((fn, i, p)),))
                for i, p in enumerate(self._parts)
            ]
            for i, h in enumerate(handles):
                results[i] = dill.loads(h.get())
        return results


def _split(rows: List, n: int) -> List[list]:
    """Spark-style contiguous range split into n partitions."""
    n = max(1, n)
    base, extra = divmod(len(rows), n)
    parts, start = [], 0
    for i in range(n):
        size = base + (1 if i < extra else 0)
        parts.append(rows[start:start + size])
        start += size
    return parts
