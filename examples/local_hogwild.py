"""Async-PS Hogwild training WITHOUT Spark — the engine-level API.

Runs anywhere (CPU or MI355X).  Two local worker threads train an MLP on
synthetic blobs against a parameter-server process over the raw-tensor
TCP transport.  For multi-GPU RCCL runs use bench.py / the dist_engine
(torchrun) path instead.
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np

from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import gaussian_mlp
from sparkflow_amd.parallel.hogwild import HogwildSparkModel
from sparkflow_amd.parallel.ps import OptimizerSpec


def main():
    rng = np.random.default_rng(0)
    n = 1000
    rows = [(rng.normal(-2 if i % 2 == 0 else 2, 0.7, 10), i % 2)
            for i in range(n)]
    partitions = [rows[0::2], rows[1::2]]

    graph_json = build_graph(gaussian_mlp)
    trainer = HogwildSparkModel(
        tensorflowGraph=graph_json,
        iters=50,
        tfInput="x:0",
        tfLabel="y:0",
        optimizer=OptimizerSpec("adam", 0.01),
        master_url="127.0.0.1:5000",
        port=5000,
        verbose=1,
        mini_batch=64,
        mini_stochastic_iters=2,
    )
    weights = trainer.train(partitions)
    print("trained %d weight tensors" % len(weights))


if __name__ == "__main__":
    main()
