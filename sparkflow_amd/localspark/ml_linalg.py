"""pyspark.ml.linalg subset: DenseVector / SparseVector / Vectors.

Both vector types are sequence-like (``len``/indexing/iteration) so
``np.asarray(vec)`` densifies them — the exact property the reference's
feature staging relies on (ml_util.py:86-101 feeds ``np.asarray(row[inp])``
to TF for dense AND sparse inputs, tests/dl_runner.py:164-185).
"""
from __future__ import annotations

import numpy as np


class DenseVector:
    def __init__(self, values):
        self._values = np.asarray(values, dtype=np.float64).reshape(-1)

    @property
    def values(self):
        return self._values

    def toArray(self) -> np.ndarray:
        return self._values.copy()

    def __array__(self, dtype=None, copy=None):
        a = self._values
        return a.astype(dtype) if dtype is not None else a.copy()

    def __len__(self):
        return len(self._values)

    def __getitem__(self, i):
        return self._values[i]

    def __iter__(self):
        return iter(self._values)

    def __eq__(self, other):
        if isinstance(other, (DenseVector, SparseVector)):
            return np.array_equal(np.asarray(self), np.asarray(other))
        return NotImplemented

    def __hash__(self):
        return hash(self._values.tobytes())

    def __repr__(self):
        return "DenseVector([%s])" % ", ".join(str(v) for v in self._values)

    # dill/pickle travels by value
    def __reduce__(self):
        return (DenseVector, (self._values.tolist(),))


class SparseVector:
    def __init__(self, size, *args):
        self.size = int(size)
        if len(args) == 1:  # dict or list of (idx, val) pairs
            pairs = sorted(args[0].items()) if isinstance(args[0], dict) \
                else sorted(args[0])
            self.indices = np.array([p[0] for p in pairs], dtype=np.int32)
            self.values = np.array([p[1] for p in pairs], dtype=np.float64)
        else:
            indices, values = args
            order = np.argsort(np.asarray(indices))
            self.indices = np.asarray(indices, dtype=np.int32)[order]
            self.values = np.asarray(values, dtype=np.float64)[order]
        if len(self.indices) and (self.indices[-1] >= self.size
                                  or self.indices[0] < 0):
            raise ValueError("index out of range")

    def toArray(self) -> np.ndarray:
        a = np.zeros(self.size, dtype=np.float64)
        a[self.indices] = self.values
        return a

    def __array__(self, dtype=None, copy=None):
        a = self.toArray()
        return a.astype(dtype) if dtype is not None else a

    def __len__(self):
        return self.size

    def __getitem__(self, i):
        if isinstance(i, slice):
            return self.toArray()[i]
        if i < 0:
            i += self.size
        if not (0 <= i < self.size):
            raise IndexError(i)
        j = np.searchsorted(self.indices, i)
        if j < len(self.indices) and self.indices[j] == i:
            return float(self.values[j])
        return 0.0

    def __iter__(self):
        return iter(self.toArray())

    def __eq__(self, other):
        if isinstance(other, (DenseVector, SparseVector)):
            return np.array_equal(np.asarray(self), np.asarray(other))
        return NotImplemented

    def __hash__(self):
        return hash(self.toArray().tobytes())

    def __repr__(self):
        return "SparseVector(%d, {%s})" % (
            self.size, ", ".join("%d: %s" % (i, v) for i, v in
                                 zip(self.indices, self.values)))

    def __reduce__(self):
        return (SparseVector,
                (self.size, self.indices.tolist(), self.values.tolist()))


class Vectors:
    @staticmethod
    def dense(*args):
        if len(args) == 1 and hasattr(args[0], "__len__"):
            return DenseVector(args[0])
        return DenseVector(args)

    @staticmethod
    def sparse(size, *args):
        return SparseVector(size, *args)


class VectorUDT:  # schema placeholder only
    pass
