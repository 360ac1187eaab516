"""MLP / autoencoder model families (reference examples/simple_dnn.py,
examples/autoencoder_example.py, tests/dl_runner.py model builders)."""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from sparkflow_amd import nn as snn
from sparkflow_amd.ops import functional as F


class ClassifierMLP(nn.Module):
    """Dense stack with ReLU hidden layers, logits out, softmax-xent loss,
    argmax prediction (reference simple_dnn.py:13-21 shape)."""

    def __init__(self, dims: List[int], hidden_activation: str = "relu"):
        super().__init__()
        layers = []
        for i in range(len(dims) - 1):
            act = hidden_activation if i < len(dims) - 2 else "none"
            layers.append(snn.Linear(dims[i], dims[i + 1], activation=act))
        self.layers = nn.ModuleList(layers)
        self.loss_fn = snn.SoftmaxCrossEntropy()

    def forward(self, x):
        for layer in self.layers:
            x = layer(x)
        return x

    def compute_loss(self, out, target):
        return self.loss_fn(out, target)

    @torch.no_grad()
    def predict(self, x):
        return F.argmax_rows(self.forward(x))


class RegressorMLP(nn.Module):
    """Dense stack with a sigmoid scalar head + MSE loss (reference
    tests/dl_runner.py:46-53 create_model shape)."""

    def __init__(self, dims: List[int], hidden_activation: str = "relu",
                 out_activation: str = "sigmoid"):
        super().__init__()
        layers = []
        for i in range(len(dims) - 1):
            act = hidden_activation if i < len(dims) - 2 else out_activation
            layers.append(snn.Linear(dims[i], dims[i + 1], activation=act))
        self.layers = nn.ModuleList(layers)
        self.loss_fn = snn.MSE()

    def forward(self, x):
        for layer in self.layers:
            x = layer(x)
        return x

    def compute_loss(self, out, target):
        return self.loss_fn(out, target.reshape(out.shape).to(out.dtype))


class DropoutRegressorMLP(nn.Module):
    """RegressorMLP with an inter-layer Dropout — exercises the
    reference's tf_dropout / to_keep_dropout surface (ml_util.py:70-71):
    dropout is active in training and inert at prediction for both
    flag values."""

    def __init__(self, dims: List[int], p: float = 0.3):
        super().__init__()
        layers = []
        for i in range(len(dims) - 1):
            act = "relu" if i < len(dims) - 2 else "sigmoid"
            layers.append(snn.Linear(dims[i], dims[i + 1], activation=act))
        self.layers = nn.ModuleList(layers)
        self.drop = snn.Dropout(p)
        self.loss_fn = snn.MSE()

    def forward(self, x):
        for i, layer in enumerate(self.layers):
            x = layer(x)
            if i < len(self.layers) - 1:
                x = self.drop(x)
        return x

    def compute_loss(self, out, target):
        return self.loss_fn(out, target.reshape(out.shape).to(out.dtype))


class Autoencoder(nn.Module):
    """Unsupervised dense AE; loss is MSE against the input batch
    (reference examples/autoencoder_example.py:9-15 — ``tfLabel=None``)."""

    def __init__(self, dims: List[int], hidden_activation: str = "relu",
                 out_activation: str = "sigmoid"):
        super().__init__()
        layers = []
        for i in range(len(dims) - 1):
            act = hidden_activation if i < len(dims) - 2 else out_activation
            layers.append(snn.Linear(dims[i], dims[i + 1], activation=act))
        self.layers = nn.ModuleList(layers)
        self.loss_fn = snn.MSE()

    def forward(self, x):
        for layer in self.layers:
            x = layer(x)
        return x

    def compute_loss(self, out, target):
        # target is the input batch (unsupervised convention)
        return self.loss_fn(out, target.reshape(out.shape).to(out.dtype))
