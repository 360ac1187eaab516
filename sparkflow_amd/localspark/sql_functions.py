"""pyspark.sql.functions subset (what the reference examples touch)."""
from __future__ import annotations

from typing import Optional


class RandColumn:
    """Sentinel returned by rand(); DataFrame.orderBy detects it and
    shuffles (reference examples order training rows by rand())."""

    def __init__(self, seed: Optional[int] = None):
        self.seed = seed


def rand(seed: Optional[int] = None) -> RandColumn:
    return RandColumn(seed)


class Column:
    def __init__(self, name: str):
        self.name = name


def col(name: str) -> Column:
    return Column(name)
