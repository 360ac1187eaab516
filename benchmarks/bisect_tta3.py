import sys

import numpy as np
import torch

sys.path.insert(0, "/root/repo")
from benchmarks.time_to_acc import make_teacher, make_teacher_task
from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_mlp
from sparkflow_amd.parallel.dist_engine import build_engine
from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig
from sparkflow_amd.utils.serialization import unpack_params

mode = sys.argv[1]  # k20 | k20sync | k100sync | k3000
torch.manual_seed(4242)
cfg = TrainConfig(iters=8000, mini_batch=300, mini_stochastic_iters=1,
                  shuffle=False, seed=7)
engine = build_engine(build_graph(mnist_mlp), OptimizerSpec("adam", 0.003),
                      cfg)
t = make_teacher()
x, y = make_teacher_task(np.random.default_rng(1000), 60000, t)
ex, ey = make_teacher_task(np.random.default_rng(999), 2000, t)
engine.worker.stage(x, y.astype(np.float32).reshape(-1, 1))
ext = torch.as_tensor(ex, device=engine.device)
eyt = torch.as_tensor(ey, device=engine.device)

k = {"k20": 20, "k20sync": 20, "k100sync": 100, "k3000": 3000}[mode]
sync = mode.endswith("sync")
for _ in range(3000 // k):
    engine.worker.run_steps(k, 300)
    if sync:
        torch.cuda.synchronize()
torch.cuda.synchronize()
unpack_params(engine.worker.module, engine.ps.weights)
with torch.no_grad():
    acc = float((engine.worker.module(ext).argmax(1) == eyt).float().mean())
print(mode, "acc@3000", round(acc, 4), flush=True)
engine.close()
