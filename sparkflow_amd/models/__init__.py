from sparkflow_amd.models.mlp import (  # noqa: F401
    Autoencoder,
    ClassifierMLP,
    DropoutRegressorMLP,
    RegressorMLP,
)
from sparkflow_amd.models.cnn import MNISTCNN, ResNet18Shape  # noqa: F401
from sparkflow_amd.models.zoo import (  # noqa: F401
    mnist_mlp,
    mnist_cnn,
    mnist_cnn_nhwc,
    mnist_autoencoder,
    tabular_autoencoder,
    resnet18_shape,
    xor_mlp,
    gaussian_mlp,
    dropout_mlp,
)
