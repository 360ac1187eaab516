#!/usr/bin/env python3
"""Per-shape NHWC conv forward microbenchmark: ms + effective TFLOP/s
for each ResNet-18 layer shape, per kernel variant.

Usage (GPU box):
    python benchmarks/conv_micro.py [--mbs 300] [--iters 30]
    SPARKFLOW_CONV_BIG=0 python benchmarks/conv_micro.py   # 64x64 tile

Prints one JSON line per shape so runs are diffable.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

# (name, H, W, Cin, Cout, K, stride, pad)
RESNET18 = [
    ("conv1_7x7s2", 224, 224, 3, 64, 7, 2, 3),
    ("l1_3x3", 56, 56, 64, 64, 3, 1, 1),
    ("l2_in_3x3s2", 56, 56, 64, 128, 3, 2, 1),
    ("l2_3x3", 28, 28, 128, 128, 3, 1, 1),
    ("l3_in_3x3s2", 28, 28, 128, 256, 3, 2, 1),
    ("l3_3x3", 14, 14, 256, 256, 3, 1, 1),
    ("l4_in_3x3s2", 14, 14, 256, 512, 3, 2, 1),
    ("l4_3x3", 7, 7, 512, 512, 3, 1, 1),
    ("l2_down_1x1s2", 56, 56, 64, 128, 1, 2, 0),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mbs", type=int, default=300)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    from sparkflow_amd import ops

    assert ops.available() and torch.cuda.is_available()
    dev = torch.device("cuda:0")
    mode = os.environ.get("SPARKFLOW_CONV_BIG", "1")
    for name, H, W, Cin, Cout, K, stride, pad in RESNET18:
        N = args.mbs
        x = torch.randn(N, H, W, Cin, device=dev)
        w = torch.randn(Cout, K, K, Cin, device=dev) * 0.05
        b = torch.empty(0, device=dev)
        Ho = (H + 2 * pad - K) // stride + 1
        Wo = (W + 2 * pad - K) // stride + 1
        flop = 2.0 * N * Ho * Wo * Cout * K * K * Cin
        for _ in range(args.warmup):
            y = ops.hip.conv_nhwc_fwd(x, w, b, stride, pad, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            y = ops.hip.conv_nhwc_fwd(x, w, b, stride, pad, 0)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / args.iters * 1000.0
        print(json.dumps({
            "shape": name, "mbs": N, "M": N * Ho * Wo,
            "K": K * K * Cin, "Cout": Cout, "ms": round(ms, 3),
            "tflops": round(flop / (ms / 1000.0) / 1e12, 1),
            "conv_big": mode,
        }))
        del x, w, y
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
