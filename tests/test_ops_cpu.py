"""CPU-path numerics of the op wrappers vs plain torch autograd."""
import numpy as np
import pytest
import torch

from sparkflow_amd.ops import functional as F


def _grads(inputs, out):
    out.backward()
    return [i.grad.clone() for i in inputs]


@pytest.mark.parametrize("act", ["none", "relu", "sigmoid"])
def test_linear_act_matches_torch(act):
    torch.manual_seed(0)
    x = torch.randn(5, 7, requires_grad=True)
    w = torch.randn(7, 3, requires_grad=True)
    b = torch.randn(3, requires_grad=True)

    act_fn = {"none": lambda t: t, "relu": torch.relu,
              "sigmoid": torch.sigmoid}[act]
    x2 = x.detach().requires_grad_()
    w2 = w.detach().requires_grad_()
    b2 = b.detach().requires_grad_()

    y = F.linear_act(x, w, b, act)
    y2 = act_fn(x2 @ w2 + b2)
    torch.testing.assert_close(y, y2)

    proj = torch.randn_like(y)
    g1 = torch.autograd.grad((y * proj).sum(), [x, w, b])
    g2 = torch.autograd.grad((y2 * proj).sum(), [x2, w2, b2])
    for a, b_ in zip(g1, g2):
        torch.testing.assert_close(a, b_, atol=1e-5, rtol=1e-4)


def test_softmax_xent_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(6, 10, requires_grad=True)
    labels = torch.randint(0, 10, (6,))
    loss = F.softmax_xent(logits, labels)
    ref_logits = logits.detach().requires_grad_()
    ref = torch.nn.functional.cross_entropy(ref_logits, labels)
    torch.testing.assert_close(loss, ref)
    g1 = torch.autograd.grad(loss, logits)[0]
    g2 = torch.autograd.grad(ref, ref_logits)[0]
    torch.testing.assert_close(g1, g2, atol=1e-6, rtol=1e-5)


def test_softmax_xent_onehot():
    torch.manual_seed(0)
    logits = torch.randn(4, 5, requires_grad=True)
    onehot = torch.zeros(4, 5)
    onehot[torch.arange(4), torch.tensor([1, 2, 3, 0])] = 1.0
    loss = F.softmax_xent(logits, onehot)
    ref = torch.nn.functional.cross_entropy(logits.detach(),
                                            torch.tensor([1, 2, 3, 0]))
    torch.testing.assert_close(loss, ref)


def test_mse_matches_torch():
    torch.manual_seed(0)
    pred = torch.randn(6, 3, requires_grad=True)
    tgt = torch.randn(6, 3)
    loss = F.mse_loss(pred, tgt)
    ref_pred = pred.detach().requires_grad_()
    ref = torch.nn.functional.mse_loss(ref_pred, tgt)
    torch.testing.assert_close(loss, ref)
    g1 = torch.autograd.grad(loss, pred)[0]
    g2 = torch.autograd.grad(ref, ref_pred)[0]
    torch.testing.assert_close(g1, g2)


def test_conv2d_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(2, 3, 8, 8, requires_grad=True)
    w = torch.randn(5, 3, 3, 3, requires_grad=True)
    b = torch.randn(5, requires_grad=True)
    y = F.conv2d_act(x, w, b, stride=1, pad=1, act="relu")
    proj = torch.randn_like(y)
    g1 = torch.autograd.grad((y * proj).sum(), [x, w, b])

    x2 = x.detach().requires_grad_()
    w2 = w.detach().requires_grad_()
    b2 = b.detach().requires_grad_()
    y2 = torch.relu(torch.nn.functional.conv2d(x2, w2, b2, padding=1))
    torch.testing.assert_close(y, y2)
    g2 = torch.autograd.grad((y2 * proj).sum(), [x2, w2, b2])
    for a, b_ in zip(g1, g2):
        torch.testing.assert_close(a, b_, atol=1e-5, rtol=1e-4)


def test_maxpool_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(2, 4, 8, 8, requires_grad=True)
    y = F.maxpool2d(x, 2, 2)
    proj = torch.randn_like(y)
    g1 = torch.autograd.grad((y * proj).sum(), [x])[0]
    x2 = x.detach().requires_grad_()
    y2 = torch.nn.functional.max_pool2d(x2, 2, 2)
    torch.testing.assert_close(y, y2)
    g2 = torch.autograd.grad((y2 * proj).sum(), [x2])[0]
    torch.testing.assert_close(g1, g2)


def test_argmax_rows():
    y = torch.tensor([[0.1, 0.9], [0.8, 0.2]])
    out = F.argmax_rows(y)
    assert out.dtype == torch.float32
    np.testing.assert_array_equal(out.numpy(), [1.0, 0.0])


def test_grad_only_cpu_keeps_value():
    """On CPU the loss value stays exact even inside grad_only() (the
    value-skip is a GPU-kernel optimization only)."""
    import torch

    from sparkflow_amd.ops import functional as F

    torch.manual_seed(3)
    logits = torch.randn(32, 10, requires_grad=True)
    labels = torch.randint(0, 10, (32,))
    with F.grad_only():
        loss = F.softmax_xent(logits, labels)
    ref = torch.nn.functional.cross_entropy(logits.detach(), labels)
    torch.testing.assert_close(loss.detach(), ref, atol=1e-5, rtol=1e-5)
    loss.backward(gradient=torch.ones(()))
    assert torch.isfinite(logits.grad).all()


def test_batchnorm2d_module_cpu():
    """snn.BatchNorm2d CPU fallback matches torch batch_norm + relu and
    updates running stats."""
    import torch

    from sparkflow_amd import nn as snn

    torch.manual_seed(4)
    bn = snn.BatchNorm2d(6, fuse_relu=True)
    x = torch.randn(4, 6, 5, 5)
    y = bn(x)
    ref_bn = torch.nn.BatchNorm2d(6)
    ref = torch.relu(ref_bn(x))
    torch.testing.assert_close(y, ref, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(bn.running_mean, ref_bn.running_mean,
                               atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(bn.running_var, ref_bn.running_var,
                               atol=1e-6, rtol=1e-5)
    # eval mode uses running stats
    bn.eval()
    y2 = bn(x)
    ref_bn.eval()
    torch.testing.assert_close(y2, torch.relu(ref_bn(x)), atol=1e-5,
                               rtol=1e-5)
