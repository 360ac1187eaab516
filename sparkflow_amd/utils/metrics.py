"""Lightweight training metrics (SURVEY.md §5 observability).

The reference exposes only a ``verbose`` flag and a ``loss_callback``
(HogwildSparkModel.py:94-100); both are preserved, and on top every worker
and the PS keep cheap counters/timers so per-iteration samples/sec and
update rates are available without a profiler (kernel-level profiling is
rocprofv3's job — see profiles/).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List


@dataclass
class Meter:
    """Windowed throughput meter."""

    count: int = 0
    total: float = 0.0
    _t0: float = field(default_factory=time.perf_counter)

    def add(self, n: int = 1) -> None:
        self.count += n

    def elapsed(self) -> float:
        return time.perf_counter() - self._t0

    def rate(self) -> float:
        e = self.elapsed()
        return self.count / e if e > 0 else 0.0

    def reset(self) -> None:
        self.count = 0
        self._t0 = time.perf_counter()


@dataclass
class WorkerStats:
    samples: Meter = field(default_factory=Meter)
    pushes: Meter = field(default_factory=Meter)
    iter_times: List[float] = field(default_factory=list)

    def record_iter(self, seconds: float, samples: int) -> None:
        self.iter_times.append(seconds)
        self.samples.add(samples)

    def summary(self) -> dict:
        n = len(self.iter_times)
        return {
            "iters": n,
            "samples": self.samples.count,
            "samples_per_sec": self.samples.rate(),
            "pushes": self.pushes.count,
            "mean_iter_ms": (1000.0 * sum(self.iter_times) / n) if n else 0.0,
        }
