#!/usr/bin/env python3
"""Inference (transform-path) throughput: the reference's
``SparkAsyncDLModel.transform`` runs each partition as one batched
forward (reference ml_util.py:54-83); this measures that forward on the
MI355X HIP kernels.

  python benchmarks/inference_bench.py [--model mnist_mlp] [--batch 8192]
"""
import argparse
import json
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="mnist_mlp")
    ap.add_argument("--batch", type=int, default=8192)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=40)
    args = ap.parse_args()

    from sparkflow_amd import models
    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.utils.serialization import module_from_json

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    module = module_from_json(build_graph(getattr(models, args.model)))
    module = module.to(device).eval()

    rng = np.random.default_rng(0)
    dim = 3 * 224 * 224 if args.model == "resnet18_shape" else 784
    x = torch.as_tensor(rng.random((args.batch, dim), dtype=np.float32),
                        device=device)

    with torch.no_grad():
        for _ in range(args.warmup):
            module.predict(x) if hasattr(module, "predict") else module(x)
        if device != "cpu":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            module.predict(x) if hasattr(module, "predict") else module(x)
        if device != "cpu":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0

    print(json.dumps({
        "metric": "inference_samples_per_sec",
        "value": args.steps * args.batch / dt,
        "unit": "samples/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {"model": args.model, "global_batch": args.batch,
                   "seq_len": None, "parallelism": "inference"},
    }))


if __name__ == "__main__":
    main()
