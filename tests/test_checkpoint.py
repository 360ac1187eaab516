"""PS checkpoint/resume + metrics counters."""
import numpy as np
import torch

from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import gaussian_mlp
from sparkflow_amd.parallel.ps import OptimizerSpec, PSCore, TrainConfig
from sparkflow_amd.parallel.transport import LocalTransport
from sparkflow_amd.parallel.worker import Worker


def _train_some(ps, graph_json, steps=5, seed=0):
    cfg = TrainConfig(iters=steps, seed=seed)
    w = Worker(graph_json, LocalTransport(ps, "cpu"), cfg, device="cpu")
    rng = np.random.default_rng(seed)
    w.run(rng.normal(size=(32, 10)).astype(np.float32),
          rng.random((32, 1)).astype(np.float32))
    return w


def test_ps_checkpoint_resume(tmp_path):
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("adam", 0.01), device="cpu")
    _train_some(ps, graph_json, steps=5)
    assert ps.optimizer.step_count() == 5
    path = str(tmp_path / "ps.ckpt")
    ps.save(path)

    ps2 = PSCore(graph_json, OptimizerSpec("adam", 0.01), device="cpu")
    ps2.restore(path)
    torch.testing.assert_close(ps2.weights, ps.weights)
    torch.testing.assert_close(ps2.optimizer.m, ps.optimizer.m)
    torch.testing.assert_close(ps2.optimizer.v, ps.optimizer.v)
    assert ps2.optimizer.step_count() == 5

    # resumed training continues identically to uninterrupted training
    _train_some(ps, graph_json, steps=3, seed=1)
    _train_some(ps2, graph_json, steps=3, seed=1)
    torch.testing.assert_close(ps2.weights, ps.weights)


def test_worker_stats():
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("gradient_descent", 0.01),
                device="cpu")
    w = _train_some(ps, graph_json, steps=4)
    s = w.stats.summary()
    assert s["iters"] == 4
    assert s["samples"] == 4 * 32
    assert s["samples_per_sec"] > 0
    assert s["mean_iter_ms"] > 0
