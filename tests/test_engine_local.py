"""End-to-end async-PS training on the local (threaded workers + spawned
PS process over TCP) path — the no-Spark analog of the reference's
Spark-local[2] integration tests (reference tests/dl_runner.py)."""
import numpy as np
import pytest

from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.ml_util import convert_weights_to_json, torch_set_weights
from sparkflow_amd.models import gaussian_mlp, tabular_autoencoder, xor_mlp
from sparkflow_amd.parallel.hogwild import HogwildSparkModel
from sparkflow_amd.parallel.ps import OptimizerSpec
from sparkflow_amd.utils.serialization import module_from_json

import torch


def _gaussian_data(n=400, seed=12345):
    rng = np.random.default_rng(seed)
    half = n // 2
    x0 = rng.normal(loc=-2.0, scale=0.7, size=(half, 10))
    x1 = rng.normal(loc=2.0, scale=0.7, size=(half, 10))
    rows = [(x0[i], 0) for i in range(half)] + [(x1[i], 1) for i in range(half)]
    rng.shuffle(rows)
    return rows


def _partitions(rows, k):
    return [rows[i::k] for i in range(k)]


def _accuracy(graph_json, weights, rows):
    module = module_from_json(graph_json)
    torch_set_weights(module, weights)
    module.eval()
    x = torch.tensor(np.asarray([r[0] for r in rows]), dtype=torch.float32)
    y = np.asarray([r[1] for r in rows])
    with torch.no_grad():
        pred = (module(x).numpy().reshape(-1) > 0.5).astype(int)
    return (pred == y).mean()


def _train(rows, graph_json, tcp_port, **kw):
    defaults = dict(
        tensorflowGraph=graph_json,
        iters=30,
        tfInput="x",
        tfLabel="y",
        optimizer=OptimizerSpec("adam", 0.01),
        master_url="127.0.0.1:%d" % tcp_port,
        port=tcp_port,
        worker_device="cpu",
        ps_device="cpu",
    )
    defaults.update(kw)
    model = HogwildSparkModel(**defaults)
    return model.train(_partitions(rows, 2))


def test_hogwild_gaussians_converge(tcp_port):
    rows = _gaussian_data()
    graph_json = build_graph(gaussian_mlp)
    weights = _train(rows, graph_json, tcp_port)
    assert len(weights) > 0
    acc = _accuracy(graph_json, weights, rows)
    assert acc > 0.8, acc


def test_lock_mode_converges(tcp_port):
    rows = _gaussian_data(n=200)
    graph_json = build_graph(gaussian_mlp)
    weights = _train(rows, graph_json, tcp_port, acquire_lock=True, iters=20)
    acc = _accuracy(graph_json, weights, rows)
    assert acc > 0.8, acc


def test_stochastic_minibatch_mode(tcp_port):
    rows = _gaussian_data(n=200)
    graph_json = build_graph(gaussian_mlp)
    weights = _train(rows, graph_json, tcp_port, iters=15,
                     mini_batch=32, mini_stochastic_iters=3)
    acc = _accuracy(graph_json, weights, rows)
    assert acc > 0.75, acc


def test_sequential_minibatch_mode(tcp_port):
    rows = _gaussian_data(n=200)
    graph_json = build_graph(gaussian_mlp)
    weights = _train(rows, graph_json, tcp_port, iters=10, mini_batch=50)
    acc = _accuracy(graph_json, weights, rows)
    assert acc > 0.75, acc


def test_partition_shuffles(tcp_port):
    rows = _gaussian_data(n=200)
    graph_json = build_graph(gaussian_mlp)
    weights = _train(rows, graph_json, tcp_port, iters=10,
                     partition_shuffles=2)
    acc = _accuracy(graph_json, weights, rows)
    assert acc > 0.75, acc


def test_autoencoder_unsupervised(tcp_port):
    rng = np.random.default_rng(0)
    rows = [rng.random(784).astype(np.float32) for _ in range(64)]
    graph_json = build_graph(tabular_autoencoder)

    model = HogwildSparkModel(
        tensorflowGraph=graph_json,
        iters=10,
        tfInput="x",
        tfLabel=None,  # unsupervised
        optimizer=OptimizerSpec("adam", 0.01),
        master_url="127.0.0.1:%d" % tcp_port,
        port=tcp_port,
        mini_batch=16,
        mini_stochastic_iters=2,
        worker_device="cpu",
        ps_device="cpu",
    )
    weights = model.train(_partitions(rows, 2))
    assert len(weights) == 4  # 2 layers x (W, b)
    for w in weights:
        assert np.isfinite(w).all()


def test_verbose_and_loss_callback(tcp_port, capsys):
    rows = _gaussian_data(n=100)
    graph_json = build_graph(gaussian_mlp)
    seen = []

    model = HogwildSparkModel(
        tensorflowGraph=graph_json,
        iters=3,
        tfInput="x",
        tfLabel="y",
        optimizer=OptimizerSpec("gradient_descent", 0.01),
        master_url="127.0.0.1:%d" % tcp_port,
        port=tcp_port,
        verbose=1,
        loss_callback=lambda loss, it, pid: seen.append((loss, it, pid)),
        worker_device="cpu",
        ps_device="cpu",
    )
    model.train(_partitions(rows, 2))
    assert len(seen) == 6  # 2 partitions x 3 iters
    out = capsys.readouterr().out
    assert "Partition Id" in out


def test_final_weights_json_roundtrip(tcp_port):
    rows = _gaussian_data(n=100)
    graph_json = build_graph(xor_mlp)
    rows2 = [(r[0][:2], r[1]) for r in rows]
    weights = _train(rows2, graph_json, tcp_port, iters=5)
    js = convert_weights_to_json(weights)
    assert isinstance(js, str) and js.startswith("[")


def test_multi_worker_engine_cpu():
    """workers_per_gpu>1: W concurrent Hogwild workers share one PS
    (reference: several partitions hitting the threaded Flask PS)."""
    import numpy as np

    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.models import mnist_mlp
    from sparkflow_amd.parallel.dist_engine import build_engine
    from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig

    cfg = TrainConfig(iters=1, mini_batch=32, mini_stochastic_iters=1,
                      seed=11)
    eng = build_engine(build_graph(mnist_mlp),
                       OptimizerSpec("adam", 0.001), cfg, device="cpu",
                       workers_per_gpu=3)
    assert len(eng.workers) == 3
    rng = np.random.default_rng(0)
    x = rng.random((256, 784), dtype=np.float32)
    y = rng.integers(0, 10, (256, 1)).astype(np.float32)
    eng.stage_all(x, y)
    samples = eng.run_steps(4, 32)
    assert samples == 4 * 32 * 3
    assert eng.ps.update_count == 12
    # per-worker seeds differ -> different minibatch streams
    assert eng.workers[0].rng.integers(0, 1 << 30) != \
        eng.workers[1].rng.integers(0, 1 << 30)
    eng.close()


def test_idx_pool_refreshes_on_wrap():
    """The device-side minibatch index pool is REDRAWN once its
    _IDX_POOL_CAP pre-drawn minibatches are consumed, so long runs do
    not cycle the same batches forever (round-1 advisor finding)."""
    import torch

    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.models import mnist_mlp
    from sparkflow_amd.parallel.dist_engine import build_engine
    from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig

    cfg = TrainConfig(iters=1, mini_batch=8, mini_stochastic_iters=1,
                      seed=3)
    eng = build_engine(build_graph(mnist_mlp),
                       OptimizerSpec("sgd", 0.01), cfg, device="cpu")
    w = eng.worker
    rng = np.random.default_rng(0)
    w.stage(rng.random((64, 784), dtype=np.float32),
            rng.integers(0, 10, (64, 1)).astype(np.float32))
    cap = w._IDX_POOL_CAP
    # fabricate the graph-capture state (graphs are GPU-only; the pool
    # bookkeeping is device-independent)
    w._graph_state = {
        "idx_pool": torch.zeros(cap, 8, dtype=torch.int64),
        "ctr": torch.zeros(1, dtype=torch.int64),
    }
    w._fill_idx_pool(64, 8)
    p0 = w._graph_state["idx_pool"].clone()
    w.maybe_refresh_pool(100)
    assert w._pool_used == 100
    torch.testing.assert_close(w._graph_state["idx_pool"], p0)  # no wrap
    w._graph_state["ctr"].fill_(37)
    w.maybe_refresh_pool(cap)  # would wrap -> redraw + rewind
    assert not torch.equal(w._graph_state["idx_pool"], p0)
    assert int(w._graph_state["ctr"].item()) == 0
    # every redrawn row is still a sample-without-replacement minibatch
    row = w._graph_state["idx_pool"][0]
    assert len(set(row.tolist())) == 8 and int(row.max()) < 64
    eng.close()


def test_idx_pool_background_draw():
    """Installing a pool kicks the NEXT draw on a background thread, so
    a wrap joins a finished thread instead of paying the host RNG in
    the step loop."""
    import torch

    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.models import mnist_mlp
    from sparkflow_amd.parallel.dist_engine import build_engine
    from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig

    cfg = TrainConfig(iters=1, mini_batch=8, mini_stochastic_iters=1,
                      seed=9)
    eng = build_engine(build_graph(mnist_mlp),
                       OptimizerSpec("sgd", 0.01), cfg, device="cpu")
    w = eng.worker
    rng = np.random.default_rng(1)
    w.stage(rng.random((64, 784), dtype=np.float32),
            rng.integers(0, 10, (64, 1)).astype(np.float32))
    cap = w._IDX_POOL_CAP
    w._graph_state = {"idx_pool": torch.zeros(cap, 8, dtype=torch.int64),
                      "ctr": torch.zeros(1, dtype=torch.int64)}
    w._fill_idx_pool(64, 8)
    t = getattr(w, "_pool_thread", None)
    assert t is not None  # draw already running
    t.join(timeout=60)
    assert w._pool_next is not None
    p0 = w._graph_state["idx_pool"].clone()
    w.maybe_refresh_pool(cap + 1)  # wrap -> installs the prefetched pool
    assert not torch.equal(w._graph_state["idx_pool"], p0)
    eng.close()
