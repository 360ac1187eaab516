import math

import numpy as np
import pytest
import torch

from sparkflow_amd.parallel.optimizers import REGISTRY, make_optimizer


ALL_NAMES = [
    "adam", "rmsprop", "momentum", "adadelta", "adagrad", "gradient_descent",
    "adagrad_da", "ftrl", "proximal_adagrad", "proximal_gradient_descent",
]


def test_registry_complete():
    assert sorted(REGISTRY) == sorted(ALL_NAMES)


def test_unknown_name_falls_back_to_sgd():
    opt = make_optimizer("nope", 4, "cpu", learning_rate=0.5)
    assert opt.name == "gradient_descent"


@pytest.mark.parametrize("name", ALL_NAMES)
def test_optimizers_descend_quadratic(name):
    """Loss f(p)=||p - target||^2 must decrease under every optimizer."""
    torch.manual_seed(0)
    n = 32
    target = torch.randn(n)
    p = torch.zeros(n)
    opt = make_optimizer(name, n, "cpu", learning_rate=0.05)
    loss0 = float(((p - target) ** 2).sum())
    for _ in range(200):
        grad = 2 * (p - target)
        opt.apply(p, grad)
    loss1 = float(((p - target) ** 2).sum())
    # TF-semantics adadelta self-scales from sqrt(eps) and moves very slowly
    # on early steps — require monotone progress only.
    factor = 0.995 if name == "adadelta" else 0.9
    assert loss1 < loss0 * factor, f"{name}: {loss0} -> {loss1}"


def test_sgd_exact():
    p = torch.ones(4)
    opt = make_optimizer("gradient_descent", 4, "cpu", learning_rate=0.1)
    opt.apply(p, torch.ones(4))
    torch.testing.assert_close(p, torch.full((4,), 0.9))


def test_adam_matches_tf_formula():
    n = 8
    rng = np.random.default_rng(0)
    p = torch.as_tensor(rng.normal(size=n), dtype=torch.float32)
    p_ref = p.clone().numpy().astype(np.float64)
    opt = make_optimizer("adam", n, "cpu", learning_rate=0.01)
    m = np.zeros(n)
    v = np.zeros(n)
    b1, b2, eps, lr = 0.9, 0.999, 1e-8, 0.01
    for t in range(1, 6):
        g = rng.normal(size=n)
        gt = torch.as_tensor(g, dtype=torch.float32)
        opt.apply(p, gt)
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        lr_t = lr * math.sqrt(1 - b2 ** t) / (1 - b1 ** t)
        p_ref -= lr_t * m / (np.sqrt(v) + eps)
    np.testing.assert_allclose(p.numpy(), p_ref, atol=1e-5)


def test_momentum_matches_formula():
    n = 4
    p = torch.zeros(n)
    opt = make_optimizer("momentum", n, "cpu", learning_rate=0.1,
                         options={"learning_rate": 0.1, "momentum": 0.9})
    g = torch.ones(n)
    opt.apply(p, g)   # acc=1, p=-0.1
    opt.apply(p, g)   # acc=1.9, p=-0.29
    torch.testing.assert_close(p, torch.full((n,), -0.29), atol=1e-6, rtol=0)


def test_rmsprop_matches_formula():
    n = 4
    lr, decay, eps = 0.1, 0.9, 1e-10
    p = torch.zeros(n)
    opt = make_optimizer("rmsprop", n, "cpu", learning_rate=lr)
    g = torch.ones(n)
    ms = 0.0
    ref = 0.0
    mom = 0.0
    for _ in range(3):
        opt.apply(p, g)
        ms = decay * ms + (1 - decay) * 1.0
        mom = 0.0 * mom + lr * 1.0 / math.sqrt(ms + eps)
        ref -= mom
    np.testing.assert_allclose(p.numpy(), np.full(n, ref), atol=1e-5)


def test_adagrad_initial_accumulator():
    n = 2
    p = torch.zeros(n)
    opt = make_optimizer("adagrad", n, "cpu", learning_rate=0.1)
    g = torch.ones(n)
    opt.apply(p, g)
    # accum = 0.1 + 1 = 1.1; p = -0.1/sqrt(1.1)
    np.testing.assert_allclose(p.numpy(), np.full(n, -0.1 / math.sqrt(1.1)),
                               atol=1e-6)


def test_default_momentum_option_injected():
    opt = make_optimizer("momentum", 2, "cpu", learning_rate=0.1, options=None)
    assert opt.momentum == 0.9
