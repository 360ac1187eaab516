from sparkflow_amd.nn.modules import (  # noqa: F401
    BatchNorm2d,
    Linear,
    Conv2d,
    MaxPool2d,
    Flatten,
    Dropout,
    SoftmaxCrossEntropy,
    MSE,
)
