"""Canonical model builders (pass these to ``build_graph``).

These correspond to the models of the reference examples and of
BASELINE.json's five benchmark configs.
"""
from __future__ import annotations

from sparkflow_amd.models.cnn import MNISTCNN, ResNet18Shape
from sparkflow_amd.models.mlp import (
    Autoencoder,
    ClassifierMLP,
    DropoutRegressorMLP,
    RegressorMLP,
)


def mnist_mlp():
    """MNIST 784-256-256-10 MLP (reference examples/simple_dnn.py:13-21)."""
    return ClassifierMLP([784, 256, 256, 10])


def mnist_cnn():
    """MNIST CNN (reference examples/cnn_example.py:10-21 shape)."""
    return MNISTCNN()


def mnist_cnn_nhwc():
    """Channel-last variant of the MNIST CNN."""
    return MNISTCNN(layout="nhwc")


def mnist_autoencoder():
    """784-256-128-256-784 AE (reference examples/autoencoder_example.py)."""
    return Autoencoder([784, 256, 128, 256, 784])


def tabular_autoencoder(dim: int = 784, hidden: int = 32):
    """BASELINE.json config 4: AE 784-32-784 on synthetic tabular data."""
    return Autoencoder([dim, hidden, dim])


def resnet18_shape(num_classes: int = 1000):
    """BASELINE.json config 5: ResNet-18-shape CNN on synthetic 3x224x224."""
    return ResNet18Shape(num_classes)


def xor_mlp(in_dim: int = 2, hidden: int = 256):
    """2-in MLP with sigmoid scalar head + MSE (reference
    tests/dl_runner.py:46-53 create_model)."""
    return RegressorMLP([in_dim, hidden, 1])


def gaussian_mlp(in_dim: int = 10, hidden: int = 256):
    """10-in MLP (reference tests/dl_runner.py:56-63 create_random_model)."""
    return RegressorMLP([in_dim, hidden, 1])


def dropout_mlp(in_dim: int = 10, hidden: int = 64, p: float = 0.3):
    """gaussian_mlp + inter-layer Dropout (the reference's tf_dropout
    surface, ml_util.py:70-71)."""
    return DropoutRegressorMLP([in_dim, hidden, 1], p=p)
