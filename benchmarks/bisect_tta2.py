"""Replicate time_to_acc.py's exact round/eval loop to find what
degrades GPU convergence vs the plain run_steps loop."""
import sys
import time

import numpy as np
import torch

sys.path.insert(0, "/root/repo")
from benchmarks.time_to_acc import make_teacher, make_teacher_task
from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_mlp
from sparkflow_amd.parallel.dist_engine import build_engine
from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig
from sparkflow_amd.utils.serialization import unpack_params

mode = sys.argv[1] if len(sys.argv) > 1 else "full"

torch.manual_seed(4242)
cfg = TrainConfig(iters=8000, mini_batch=300, mini_stochastic_iters=1,
                  shuffle=False, seed=7)
engine = build_engine(build_graph(mnist_mlp), OptimizerSpec("adam", 0.003),
                      cfg)
device = engine.device
t = make_teacher()
x, y = make_teacher_task(np.random.default_rng(1000), 60000, t)
ex, ey = make_teacher_task(np.random.default_rng(999), 2000, t)
engine.worker.stage(x, y.astype(np.float32).reshape(-1, 1))
ext = torch.as_tensor(ex, device=device)
eyt = torch.as_tensor(ey, device=device)

# warmup identical to harness
wt0 = time.perf_counter()
engine.worker.run_steps(3, 300)
torch.cuda.synchronize()
if mode != "noburn":
    a = torch.randn(4096, 4096, device=device)
    while time.perf_counter() - wt0 < 2.0:
        a = a @ a * 1e-6
        torch.cuda.synchronize()
    del a

for r in range(1, 401):
    engine.worker.run_steps(20, 300)
    torch.cuda.synchronize()
    if mode != "noeval":
        unpack_params(engine.worker.module, engine.ps.weights)
        with torch.no_grad():
            pred = engine.worker.module(ext).argmax(dim=1)
        acc = float((pred == eyt).float().mean())
    if r % 50 == 0:
        if mode == "noeval":
            unpack_params(engine.worker.module, engine.ps.weights)
            with torch.no_grad():
                acc = float((engine.worker.module(ext).argmax(1)
                             == eyt).float().mean())
        print(mode, "round", r, "acc", round(acc, 4), flush=True)
engine.close()
