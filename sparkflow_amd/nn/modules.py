"""NN modules backed by the CDNA4 HIP ops.

These are the building blocks for user model-builder functions (the
``build_graph`` convention).  They mirror the TF-1.x layers the reference
examples use (tf.layers.dense / conv2d / max_pooling2d — reference
examples/simple_dnn.py:16-18, examples/cnn_example.py:13-21) with TF's
default glorot-uniform weight init and zero bias init, so convergence
behavior matches.

Weight layouts are chosen for the kernels, not for TF compatibility:
Linear keeps W as [in, out] (row-major GEMM B-operand), Conv2d keeps
[Cout, Cin, Kh, Kw] (NCHW implicit-GEMM).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from sparkflow_amd.ops import functional as F


def _glorot(shape, fan_in, fan_out, generator=None):
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    return torch.empty(*shape).uniform_(-limit, limit, generator=generator)


class Linear(nn.Module):
    def __init__(self, in_features: int, out_features: int,
                 activation: str = "none", bias: bool = True):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.activation = activation
        self.weight = nn.Parameter(
            _glorot((in_features, out_features), in_features, out_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None

    def forward(self, x):
        return F.linear_act(x, self.weight, self.bias, self.activation)

    def extra_repr(self):
        return f"{self.in_features}->{self.out_features}, act={self.activation}"


class Conv2d(nn.Module):
    """``layout="nchw"`` (default) keeps weights [Cout, Cin, Kh, Kw] and
    NCHW tensors; ``layout="nhwc"`` keeps weights [Cout, Kh, Kw, Cin] and
    channel-last tensors (the ResNet-config fast path — channel-
    contiguous float4 im2col reads)."""

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding="same", activation: str = "none",
                 bias: bool = True, layout: str = "nchw"):
        super().__init__()
        k = kernel_size
        if padding == "same":
            if stride != 1:
                raise ValueError("padding='same' requires stride=1")
            padding = (k - 1) // 2
        self.stride, self.padding, self.activation = stride, int(padding), activation
        self.layout = layout
        fan_in = in_channels * k * k
        fan_out = out_channels * k * k
        shape = ((out_channels, k, k, in_channels) if layout == "nhwc"
                 else (out_channels, in_channels, k, k))
        self.weight = nn.Parameter(_glorot(shape, fan_in, fan_out))
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None

    def forward(self, x):
        if self.layout == "nhwc":
            return F.conv2d_act_nhwc(x, self.weight, self.bias, self.stride,
                                     self.padding, self.activation)
        return F.conv2d_act(x, self.weight, self.bias, self.stride,
                            self.padding, self.activation)


class MaxPool2d(nn.Module):
    def __init__(self, kernel_size: int, stride: int = None, padding: int = 0,
                 layout: str = "nchw"):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride or kernel_size
        self.padding = padding
        self.layout = layout

    def forward(self, x):
        if self.layout == "nhwc":
            from sparkflow_amd import ops as _ops

            if x.is_cuda and _ops.available():
                return F.maxpool2d_nhwc(x, self.kernel_size, self.stride,
                                        self.padding)
            y = F.maxpool2d(x.permute(0, 3, 1, 2).contiguous(),
                            self.kernel_size, self.stride, self.padding)
            return y.permute(0, 2, 3, 1).contiguous()
        return F.maxpool2d(x, self.kernel_size, self.stride, self.padding)


class Flatten(nn.Module):
    def forward(self, x):
        return x.reshape(x.shape[0], -1)


class BatchNorm2d(nn.Module):
    """BatchNorm2d on the custom CDNA4 kernels (training mode), with an
    optional fused trailing ReLU (``fuse_relu=True`` replaces a separate
    bn -> relu pair).  Eval/CPU fall back to torch batch_norm."""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, fuse_relu: bool = False,
                 layout: str = "nchw"):
        super().__init__()
        self.num_features = num_features
        self.eps, self.momentum, self.fuse_relu = eps, momentum, fuse_relu
        self.layout = layout
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def forward(self, x, residual=None):
        if self.layout == "nhwc":
            return F.batchnorm2d_act_nhwc(x, self.weight, self.bias,
                                          self.running_mean,
                                          self.running_var, self.momentum,
                                          self.eps, self.fuse_relu,
                                          self.training, residual=residual)
        y = F.batchnorm2d_act(x, self.weight, self.bias,
                              self.running_mean, self.running_var,
                              self.momentum, self.eps,
                              self.fuse_relu and residual is None,
                              self.training)
        if residual is not None:  # NCHW path: plain add (+relu)
            y = y + residual
            if self.fuse_relu:
                y = torch.relu(y)
        return y

    def extra_repr(self):
        return f"{self.num_features}, fuse_relu={self.fuse_relu}"


class Dropout(nn.Module):
    """Inverted dropout; inert in eval mode (the inference path — parity
    with the reference's keep-prob feed at prediction, ml_util.py:70-71)."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p

    def forward(self, x):
        if not self.training or self.p <= 0:
            return x
        keep = 1.0 - self.p
        mask = (torch.rand_like(x) < keep).to(x.dtype) / keep
        return x * mask


class SoftmaxCrossEntropy(nn.Module):
    """Mean softmax cross-entropy (one-hot or index labels)."""

    def forward(self, logits, labels):
        return F.softmax_xent(logits, labels)


class MSE(nn.Module):
    def forward(self, pred, target):
        return F.mse_loss(pred, target)
