"""CNN model families.

MNISTCNN mirrors reference examples/cnn_example.py:13-21 (5x5x32 conv ->
2x2 pool -> 3x3x64 conv -> 2x2 pool -> flatten -> dense 10).

ResNet18Shape is the BASELINE.json config-5 model: a ResNet-18-shaped CNN
for synthetic 3x224x224 inputs (large-batch PS stress).  Convs/GEMMs and batch-norm
all run on the CDNA4 HIP kernels (snn.BatchNorm2d fuses the trailing
ReLU where one follows).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from sparkflow_amd import nn as snn
from sparkflow_amd.ops import functional as F


class MNISTCNN(nn.Module):
    """Input: flat [B, 784] (Spark vector rows), reshaped to NCHW (or
    channel-last with layout="nhwc")."""

    def __init__(self, num_classes: int = 10, layout: str = "nchw"):
        super().__init__()
        self.layout = layout
        self.conv1 = snn.Conv2d(1, 32, 5, padding="same", activation="relu",
                                layout=layout)
        self.pool1 = snn.MaxPool2d(2, 2, layout=layout)
        self.conv2 = snn.Conv2d(32, 64, 3, padding="same", activation="relu",
                                layout=layout)
        self.pool2 = snn.MaxPool2d(2, 2, layout=layout)
        self.fc = snn.Linear(7 * 7 * 64, num_classes)
        self.loss_fn = snn.SoftmaxCrossEntropy()

    def forward(self, x):
        if x.dim() == 2:
            x = x.reshape(-1, 1, 28, 28)
        if self.layout == "nhwc":
            x = x.permute(0, 2, 3, 1).contiguous()
        x = self.pool1(self.conv1(x))
        x = self.pool2(self.conv2(x))
        x = x.reshape(x.shape[0], -1)
        return self.fc(x)

    def compute_loss(self, out, target):
        return self.loss_fn(out, target)

    @torch.no_grad()
    def predict(self, x):
        return F.argmax_rows(self.forward(x))


class _BasicBlock(nn.Module):
    def __init__(self, cin, cout, stride, layout="nhwc"):
        super().__init__()
        self.conv1 = snn.Conv2d(cin, cout, 3, stride=stride, padding=1,
                                activation="none", bias=False,
                                layout=layout)
        self.bn1 = snn.BatchNorm2d(cout, fuse_relu=True, layout=layout)
        self.conv2 = snn.Conv2d(cout, cout, 3, stride=1, padding=1,
                                activation="none", bias=False,
                                layout=layout)
        # bn2 fuses the block boundary: out = relu(bn2(conv2) + identity)
        # (one kernel instead of bn + add + relu)
        self.bn2 = snn.BatchNorm2d(cout, fuse_relu=True, layout=layout)
        if stride != 1 or cin != cout:
            self.down_conv = snn.Conv2d(cin, cout, 1, stride=stride,
                                        padding=0, activation="none",
                                        bias=False, layout=layout)
            self.down_bn = snn.BatchNorm2d(cout, layout=layout)
        else:
            self.down_conv = None

    def forward(self, x):
        idt = x
        y = self.bn1(self.conv1(x))  # fused bn+relu
        if self.down_conv is not None:
            idt = self.down_bn(self.down_conv(x))
        return self.bn2(self.conv2(y), residual=idt)


class ResNet18Shape(nn.Module):
    """ResNet-18 layer shapes for 3x224x224 synthetic inputs.

    Internally channel-last (NHWC): one permute at the input, then every
    conv/bn runs on the channel-contiguous kernel family."""

    def __init__(self, num_classes: int = 1000, layout: str = "nhwc"):
        super().__init__()
        self.layout = layout
        self.stem_conv = snn.Conv2d(3, 64, 7, stride=2, padding=3,
                                    activation="none", bias=False,
                                    layout=layout)
        self.stem_bn = snn.BatchNorm2d(64, fuse_relu=True, layout=layout)
        self.stem_pool = snn.MaxPool2d(3, 2, padding=1,
                                       layout=layout)  # 112 -> 56
        stages = []
        cin = 64
        for cout, stride in [(64, 1), (64, 1), (128, 2), (128, 1),
                             (256, 2), (256, 1), (512, 2), (512, 1)]:
            stages.append(_BasicBlock(cin, cout, stride, layout=layout))
            cin = cout
        self.stages = nn.ModuleList(stages)
        self.fc = snn.Linear(512, num_classes)
        self.loss_fn = snn.SoftmaxCrossEntropy()

    def forward(self, x):
        if x.dim() == 2:
            x = x.reshape(-1, 3, 224, 224)
        if self.layout == "nhwc":
            x = x.permute(0, 2, 3, 1).contiguous()
            sdims = (1, 2)
        else:
            sdims = (2, 3)
        x = self.stem_bn(self.stem_conv(x))  # fused bn+relu
        x = self.stem_pool(x)
        for block in self.stages:
            x = block(x)
        x = x.mean(dim=sdims)
        return self.fc(x)

    def compute_loss(self, out, target):
        return self.loss_fn(out, target)

    @torch.no_grad()
    def predict(self, x):
        return F.argmax_rows(self.forward(x))
