"""CIFAR-shape DenseNet family (pre-activation, GroupNorm(32)).

Architecture parity with reference Net/Densenet.py: bottleneck layers
GN->ReLU->1x1 (4k) -> GN->ReLU->3x3 (k) with dense channel concat,
transitions GN->ReLU->1x1 (0.5x) -> avg_pool(2), final GN->ReLU->
avg_pool(4)->linear.  DenseNet-121 here is the flagship bench model
(BASELINE.json: images/sec at global batch 512).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as FD
from ..ops.layers import Conv2d, GroupNormAct, Linear

_GN = 32


class _DenseLayer(nn.Module):
    def __init__(self, cin, growth):
        super().__init__()
        mid = 4 * growth
        self.norm1 = GroupNormAct(_GN, cin, relu=True)
        self.conv1 = Conv2d(cin, mid, 1)
        self.norm2 = GroupNormAct(_GN, mid, relu=True)
        self.conv2 = Conv2d(mid, growth, 3, padding=1)

    def forward(self, x):
        fresh = self.conv2(self.norm2(self.conv1(self.norm1(x))))
        return torch.cat([fresh, x], dim=1)


class _Transition(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.norm = GroupNormAct(_GN, cin, relu=True)
        self.conv = Conv2d(cin, cout, 1)

    def forward(self, x):
        return FD.avg_pool2d(self.conv(self.norm(x)), 2)


class DenseNet(nn.Module):
    def __init__(self, depths, growth=12, reduction=0.5, num_classes=10):
        super().__init__()
        ch = 2 * growth
        self.stem = Conv2d(3, ch, 3, padding=1)
        body = []
        for i, depth in enumerate(depths):
            for _ in range(depth):
                body.append(_DenseLayer(ch, growth))
                ch += growth
            if i < len(depths) - 1:
                cout = int(math.floor(ch * reduction))
                body.append(_Transition(ch, cout))
                ch = cout
        self.body = nn.Sequential(*body)
        self.final_norm = GroupNormAct(_GN, ch, relu=True)
        self.head = Linear(ch, num_classes)

    def forward(self, x):
        out = self.body(self.stem(x))
        out = FD.avg_pool2d(self.final_norm(out), 4).flatten(1)
        return self.head(out)


def DenseNet121(num_classes=10):
    return DenseNet((6, 12, 24, 16), growth=32, num_classes=num_classes)


def DenseNet169(num_classes=10):
    return DenseNet((6, 12, 32, 32), growth=32, num_classes=num_classes)


def DenseNet201(num_classes=10):
    return DenseNet((6, 12, 48, 32), growth=32, num_classes=num_classes)


def DenseNet161(num_classes=10):
    return DenseNet((6, 12, 36, 24), growth=48, num_classes=num_classes)
