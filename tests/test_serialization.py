import json

import numpy as np
import pytest
import torch

from sparkflow_amd.graph_utils import build_graph, build_adam_config
from sparkflow_amd.ml_util import (
    calculate_weights,
    convert_json_to_weights,
    convert_weights_to_json,
    handle_batch,
    handle_features,
    handle_shuffle,
    torch_get_weights,
    torch_set_weights,
)
from sparkflow_amd.models import mnist_mlp, xor_mlp
from sparkflow_amd.utils.serialization import (
    flat_size,
    flat_to_list,
    list_to_flat,
    module_from_json,
    pack_grads,
    pack_params,
    unpack_params,
)


def test_build_graph_roundtrip():
    g = build_graph(mnist_mlp)
    m1 = module_from_json(g)
    m2 = module_from_json(g)
    w1 = torch_get_weights(m1)
    w2 = torch_get_weights(m2)
    assert len(w1) == 6  # 3 layers x (W, b)
    for a, b in zip(w1, w2):
        np.testing.assert_array_equal(a, b)
    x = torch.randn(4, 784)
    y1, y2 = m1(x), m2(x)
    torch.testing.assert_close(y1, y2)


def test_weights_json_codec():
    m = module_from_json(build_graph(xor_mlp))
    ws = torch_get_weights(m)
    js = convert_weights_to_json(ws)
    back = convert_json_to_weights(js)
    assert json.loads(js)  # valid json list-of-lists
    for a, b in zip(ws, back):
        np.testing.assert_allclose(a, b, rtol=1e-6)
    torch_set_weights(m, back)
    for a, b in zip(torch_get_weights(m), ws):
        np.testing.assert_allclose(a, b, rtol=1e-6)


def test_flat_pack_roundtrip():
    m = module_from_json(build_graph(xor_mlp))
    n = flat_size(m)
    flat = pack_params(m)
    assert flat.numel() == n
    m2 = module_from_json(build_graph(xor_mlp))
    unpack_params(m2, flat)
    torch.testing.assert_close(pack_params(m2), flat)
    lst = flat_to_list(m, flat)
    torch.testing.assert_close(list_to_flat(lst), flat)


def test_pack_grads():
    m = module_from_json(build_graph(xor_mlp))
    x = torch.randn(8, 2)
    out = m(x)
    loss = m.compute_loss(out, torch.rand(8, 1))
    loss.backward()
    g = pack_grads(m)
    assert g.numel() == flat_size(m)
    assert torch.isfinite(g).all()
    assert g.abs().sum() > 0


def test_handle_features_supervised():
    data = [(np.ones(3), 1), (np.zeros(3), 0.0)]
    feats, labels = handle_features(data, is_supervised=True)
    assert feats.shape == (2, 3)
    assert labels.shape == (2, 1)  # scalar labels wrapped to [y]


def test_handle_batch_modes():
    x = np.arange(20, dtype=np.float32).reshape(10, 2)
    y = np.arange(10).reshape(10, 1)
    fx, fy = handle_batch(x, y, -1)
    assert fx.shape == (10, 2)
    sx, sy = handle_batch(x, y, 4, idx=2)
    np.testing.assert_array_equal(sx, x[2:6])
    rx, ry = handle_batch(x, y, 4, rng=np.random.default_rng(0))
    assert rx.shape == (4, 2)
    # reference clamp: mbs > n -> n-1
    cx, _ = handle_batch(x, y, 100, rng=np.random.default_rng(0))
    assert cx.shape == (9, 2)


def test_handle_shuffle():
    x = np.arange(10, dtype=np.float32).reshape(10, 1)
    sx, _ = handle_shuffle(x, None, rng=np.random.default_rng(0))
    assert sorted(sx.reshape(-1).tolist()) == x.reshape(-1).tolist()


def test_calculate_weights():
    a = [np.ones((2, 2)), np.zeros(3)]
    b = [3 * np.ones((2, 2)), 2 * np.ones(3)]
    avg = calculate_weights([a, b])
    np.testing.assert_allclose(avg[0], 2 * np.ones((2, 2)))
    np.testing.assert_allclose(avg[1], np.ones(3))


def test_optimizer_config_builders():
    cfg = json.loads(build_adam_config(learning_rate=0.1))
    assert cfg["learning_rate"] == 0.1
    assert cfg["beta1"] == 0.9
