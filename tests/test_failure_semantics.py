"""Failure-detection parity: error budget on the PS, push
failures printed-and-continued on the worker, PS teardown on train()
failure."""
import threading

import numpy as np
import pytest
import torch

from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import gaussian_mlp
from sparkflow_amd.parallel.ps import OptimizerSpec, PSCore, TrainConfig
from sparkflow_amd.parallel.transport import LocalTransport, WorkerTransport
from sparkflow_amd.parallel.worker import Worker


def test_ps_error_budget():
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("gradient_descent", 0.01),
                device="cpu", max_errors=3)
    bad = torch.zeros(7)  # wrong size -> apply failure
    ps.apply_gradients(bad)
    ps.apply_gradients(bad)
    assert ps.error_count == 2
    with pytest.raises(Exception, match="Too many failures during training"):
        ps.apply_gradients(bad)


def test_ps_good_updates_do_not_count():
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("gradient_descent", 0.01),
                device="cpu", max_errors=2)
    ps.apply_gradients(torch.randn(ps.numel))
    ps.apply_gradients(torch.randn(ps.numel))
    assert ps.error_count == 0
    assert ps.update_count == 2


class _FailingPushTransport(WorkerTransport):
    """Pull works; every push raises (the reference's 'Timeout error from
    partition' path, HogwildSparkModel.py:68-71)."""

    def __init__(self, ps):
        self.ps = ps

    def pull(self):
        return self.ps.read_weights()

    def push(self, flat_grads):
        raise ConnectionError("injected push failure")


def test_worker_push_failure_prints_and_continues(capsys):
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("gradient_descent", 0.01),
                device="cpu")
    w = Worker(graph_json, _FailingPushTransport(ps),
               TrainConfig(iters=3, seed=0), device="cpu")
    rng = np.random.default_rng(0)
    w.run(rng.normal(size=(16, 10)).astype(np.float32),
          rng.random((16, 1)).astype(np.float32))  # must NOT raise
    out = capsys.readouterr().out
    assert out.count("Timeout error from partition") == 3
    assert ps.update_count == 0


def test_hogwild_concurrent_threads_cpu():
    """Concurrent racy applies (the Hogwild mode's intended race) leave
    finite weights and account every update."""
    graph_json = build_graph(gaussian_mlp)
    ps = PSCore(graph_json, OptimizerSpec("adam", 0.01), device="cpu")

    def run(seed):
        w = Worker(graph_json, LocalTransport(ps, "cpu"),
                   TrainConfig(iters=10, seed=seed), device="cpu")
        rng = np.random.default_rng(seed)
        w.run(rng.normal(size=(32, 10)).astype(np.float32),
              rng.random((32, 1)).astype(np.float32))

    threads = [threading.Thread(target=run, args=(s,)) for s in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert ps.update_count == 40
    assert torch.isfinite(ps.weights).all()


def test_train_stops_server_on_failure(tcp_port):
    """train() must terminate the PS process on both success and failure
    (reference HogwildSparkModel.py:246-272)."""
    from sparkflow_amd.parallel.hogwild import HogwildSparkModel

    graph_json = build_graph(gaussian_mlp)
    model = HogwildSparkModel(
        tensorflowGraph=graph_json, iters=2, tfInput="x", tfLabel="y",
        optimizer=OptimizerSpec("gradient_descent", 0.01),
        master_url="127.0.0.1:%d" % tcp_port, port=tcp_port,
        worker_device="cpu", ps_device="cpu")
    server = model.server
    assert server.is_alive()

    class Boom(Exception):
        pass

    rows = [(np.zeros(10, dtype=np.float32), 0)] * 4

    # inject a failure inside the local training path
    orig = model._train_local

    def boom(parts):
        raise Boom("injected")

    model._train_local = boom
    with pytest.raises(Boom):
        model.train([rows])
    server.join(timeout=10)
    assert not server.is_alive()
