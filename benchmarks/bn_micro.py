#!/usr/bin/env python3
"""Per-shape NHWC BatchNorm kernel microbenchmark (fwd pair + bwd pair),
for the v4-vs-scalar A/B (SPARKFLOW_BN_V4=0 forces the scalar kernels)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

SHAPES = [(300, 112, 112, 64), (300, 56, 56, 64), (300, 28, 28, 128),
          (300, 14, 14, 256), (300, 7, 7, 512)]


def main():
    from sparkflow_amd import ops

    assert ops.available()
    dev = "cuda:0"
    mode = os.environ.get("SPARKFLOW_BN_V4", "1")
    for (N, H, W, C) in SHAPES:
        x = torch.randn(N, H, W, C, device=dev)
        dy = torch.randn_like(x)
        g = torch.rand(C, device=dev) + 0.5
        b = torch.randn(C, device=dev)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)
        e = torch.empty(0, device=dev)
        for _ in range(5):
            y, sm, si = ops.hip.bn2d_nhwc_fwd(x, g, b, rm, rv, 0.1, 1e-5,
                                              True, e)
            ops.hip.bn2d_nhwc_bwd(x, dy, y, g, sm, si, True, False)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(20):
            y, sm, si = ops.hip.bn2d_nhwc_fwd(x, g, b, rm, rv, 0.1, 1e-5,
                                              True, e)
            ops.hip.bn2d_nhwc_bwd(x, dy, y, g, sm, si, True, False)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 20 * 1000.0
        gb = x.numel() * 4 * 7 / 1e9  # ~7 tensor passes fwd+bwd
        print(json.dumps({"shape": [N, H, W, C], "ms": round(ms, 3),
                          "eff_tb_s": round(gb / ms, 2), "v4": mode}))
        del x, dy, y
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
