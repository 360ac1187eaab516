import json, os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from sparkflow_amd import ops
SHAPES = [("l1_3x3", 56, 56, 64, 64, 3, 1, 1),
          ("l2_in_s2", 56, 56, 64, 128, 3, 2, 1),
          ("l2_3x3", 28, 28, 128, 128, 3, 1, 1),
          ("l3_3x3", 14, 14, 256, 256, 3, 1, 1),
          ("l4_3x3", 7, 7, 512, 512, 3, 1, 1)]
mode = os.environ.get("SPARKFLOW_CONV_SQ", "1")
for name, H, W, Cin, Cout, K, st, pd in SHAPES:
    N = 300
    x = torch.randn(N, H, W, Cin, device="cuda")
    Ho = (H + 2*pd - K)//st + 1; Wo = (W + 2*pd - K)//st + 1
    dy = torch.randn(N, Ho, Wo, Cout, device="cuda")
    flop = 2.0 * N * Ho * Wo * Cout * K * K * Cin
    for _ in range(5): ops.hip.conv_nhwc_wgrad(x, dy, K, K, st, pd)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(20): ops.hip.conv_nhwc_wgrad(x, dy, K, K, st, pd)
    torch.cuda.synchronize()
    ms = (time.perf_counter()-t0)/20*1000
    print(json.dumps({"shape": name, "ms": round(ms,3),
                      "tflops": round(flop/(ms/1000)/1e12,1), "sq": mode}))
    del x, dy; torch.cuda.empty_cache()
