import sys

import numpy as np
import torch

sys.path.insert(0, "/root/repo")
from benchmarks.time_to_acc import make_teacher, make_teacher_task
from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_mlp
from sparkflow_amd.parallel.dist_engine import build_engine
from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig
from sparkflow_amd.utils.serialization import module_from_json, unpack_params

mode = sys.argv[1]  # sep | unpack | fwd
torch.manual_seed(4242)
graph_json = build_graph(mnist_mlp)
cfg = TrainConfig(iters=8000, mini_batch=300, mini_stochastic_iters=1,
                  shuffle=False, seed=7)
engine = build_engine(graph_json, OptimizerSpec("adam", 0.003), cfg)
dev = engine.device
t = make_teacher()
x, y = make_teacher_task(np.random.default_rng(1000), 60000, t)
ex, ey = make_teacher_task(np.random.default_rng(999), 2000, t)
engine.worker.stage(x, y.astype(np.float32).reshape(-1, 1))
ext = torch.as_tensor(ex, device=dev)
eyt = torch.as_tensor(ey, device=dev)
em = module_from_json(graph_json).to(dev).eval()

for r in range(1, 13):
    engine.worker.run_steps(500, 300)
    torch.cuda.synchronize()
    acc = None
    if mode == "sep":
        unpack_params(em, engine.ps.weights)
        with torch.no_grad():
            acc = float((em(ext).argmax(1) == eyt).float().mean())
    elif mode == "unpack":
        unpack_params(engine.worker.module, engine.ps.weights)
    elif mode == "fwd":
        with torch.no_grad():
            engine.worker.module(ext)
    torch.cuda.synchronize()
    if acc is None:
        unpack_params(em, engine.ps.weights)
        with torch.no_grad():
            acc = float((em(ext).argmax(1) == eyt).float().mean())
    print(mode, r * 500, round(acc, 4), flush=True)
engine.close()
