"""Shared data loading for the Spark examples.

The reference examples read ``mnist_train.csv`` from disk
(examples/simple_dnn.py:14 in the reference).  When the csv exists we do
the same; when it does not (this image has no network to fetch MNIST),
we synthesize an MNIST-SHAPED DataFrame with the same column layout
(label in ``_c0``, pixels in ``_c1..c784``) so every example still runs
end to end.
"""
import os

import numpy as np


def load_mnist_df(spark, rows: int = 2000, csv: str = "examples/mnist_train.csv"):
    if os.path.exists(csv):
        return (spark.read.option("inferSchema", "true").csv(csv))
    rng = np.random.default_rng(12345)
    cols = ["_c%d" % i for i in range(785)]
    # class-templated pixels so training has signal to find
    templates = rng.random((10, 784), dtype=np.float32)
    data = []
    for i in range(rows):
        label = int(rng.integers(0, 10))
        px = np.clip(templates[label]
                     + rng.normal(0, 0.35, 784).astype(np.float32), 0, 1)
        data.append(tuple([label] + [float(v) for v in px]))
    print(f"[examples] {csv} not found — using a synthetic MNIST-shaped "
          f"DataFrame ({rows} rows)")
    return spark.createDataFrame(data, cols)
