import torch

from sparkflow_amd.models import (
    mnist_autoencoder,
    mnist_cnn,
    mnist_mlp,
    resnet18_shape,
    tabular_autoencoder,
    xor_mlp,
)


def _fwd_bwd(model, x, target):
    out = model(x)
    loss = model.compute_loss(out, target)
    loss.backward()
    assert torch.isfinite(loss)
    for p in model.parameters():
        if p.requires_grad:
            assert p.grad is not None
            assert torch.isfinite(p.grad).all()
    return out, loss


def test_mnist_mlp():
    m = mnist_mlp()
    x = torch.randn(4, 784)
    y = torch.randint(0, 10, (4,))
    out, _ = _fwd_bwd(m, x, y)
    assert out.shape == (4, 10)
    pred = m.predict(x)
    assert pred.shape == (4,)
    assert pred.dtype == torch.float32


def test_mnist_cnn_from_flat_rows():
    m = mnist_cnn()
    x = torch.randn(2, 784)
    y = torch.randint(0, 10, (2,))
    out, _ = _fwd_bwd(m, x, y)
    assert out.shape == (2, 10)


def test_autoencoders_unsupervised():
    for builder in (mnist_autoencoder, tabular_autoencoder):
        m = builder()
        x = torch.rand(3, 784)
        out, _ = _fwd_bwd(m, x, x)
        assert out.shape == (3, 784)


def test_xor_mlp_scalar_head():
    m = xor_mlp()
    x = torch.randn(5, 2)
    y = torch.rand(5, 1)
    out, _ = _fwd_bwd(m, x, y)
    assert out.shape == (5, 1)
    assert (out >= 0).all() and (out <= 1).all()  # sigmoid head


def test_resnet18_shape_small_batch():
    m = resnet18_shape(num_classes=10)
    x = torch.randn(2, 3, 224, 224)
    y = torch.randint(0, 10, (2,))
    out, _ = _fwd_bwd(m, x, y)
    assert out.shape == (2, 10)


def test_keep_dropout_inference_inert_both_ways():
    """Dropout is inert at prediction for BOTH to_keep_dropout values —
    the reference's flag (ml_util.py:70-71) only chooses whether the fed
    constant is read as keep-prob (1.0) or drop-rate (0.0); neither
    setting drops units at inference."""
    import numpy as np
    import torch

    from sparkflow_amd import nn as snn
    from sparkflow_amd.ml_util import predict_batch

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.drop = snn.Dropout(0.5)

        def forward(self, x):
            return self.drop(x)

    m = M()
    x = np.ones((64, 32), dtype=np.float32)
    out_eval = predict_batch(m, x, device=torch.device("cpu"))
    np.testing.assert_array_equal(out_eval, x)  # dropout inert
    out_keep = predict_batch(m, x, device=torch.device("cpu"),
                             keep_dropout=True)
    np.testing.assert_array_equal(out_keep, x)  # still inert
