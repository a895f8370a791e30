"""CIFAR-shape DenseNet family (pre-activation, GroupNorm(32)).

Architecture parity with reference Net/Densenet.py: bottleneck layers
GN->ReLU->1x1 (4k) -> GN->ReLU->3x3 (k) with dense channel concat,
transitions GN->ReLU->1x1 (0.5x) -> avg_pool(2), final GN->ReLU->
avg_pool(4)->linear.  DenseNet-121 here is the flagship bench model
(BASELINE.json: images/sec at global batch 512).

MI355X execution note: the dense concat (reference torch.cat at
Net/Densenet.py:20) is VIRTUAL on the GPU path — the residual stream is
carried as a list of per-layer outputs and the fused GroupNorm kernel
reads the segments in place (ops/functional.group_norm_act_cat), so no
concat copies are materialized in forward or backward.  Numerics are
identical to the materialized concat (the CPU debug path still cats).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops import functional as FD
from ..ops.layers import Conv2d, GroupNormAct, Linear

_GN = 32


class CatGroupNormAct(nn.Module):
    """GroupNorm(+ReLU) over a list of channel segments (virtual concat,
    newest segment first — the reference's cat([out, x], 1) order)."""

    def __init__(self, num_groups, num_channels, eps=1e-5, relu=False):
        super().__init__()
        self.num_groups, self.eps, self.relu = num_groups, eps, relu
        self.weight = nn.Parameter(torch.ones(num_channels))
        self.bias = nn.Parameter(torch.zeros(num_channels))

    def forward(self, segs):
        if isinstance(segs, torch.Tensor):
            segs = [segs]
        return FD.group_norm_act_cat(segs, self.num_groups, self.weight,
                                     self.bias, self.eps, self.relu)


class _DenseLayer(nn.Module):
    def __init__(self, cin, growth):
        super().__init__()
        mid = 4 * growth
        self.norm1 = CatGroupNormAct(_GN, cin, relu=True)
        self.conv1 = Conv2d(cin, mid, 1)
        self.norm2 = GroupNormAct(_GN, mid, relu=True)
        self.conv2 = Conv2d(mid, growth, 3, padding=1)

    def forward(self, segs):
        h = self.conv1(self.norm1(segs))
        return self.conv2(self.norm2(h))  # the fresh growth channels


class _Transition(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.norm = CatGroupNormAct(_GN, cin, relu=True)
        self.conv = Conv2d(cin, cout, 1)

    def forward(self, segs):
        return FD.avg_pool2d(self.conv(self.norm(segs)), 2)


class DenseNet(nn.Module):
    def __init__(self, depths, growth=12, reduction=0.5, num_classes=10):
        super().__init__()
        ch = 2 * growth
        self.stem = Conv2d(3, ch, 3, padding=1)
        self.blocks = nn.ModuleList()
        self.transitions = nn.ModuleList()
        for i, depth in enumerate(depths):
            block = nn.ModuleList()
            for _ in range(depth):
                block.append(_DenseLayer(ch, growth))
                ch += growth
            self.blocks.append(block)
            if i < len(depths) - 1:
                cout = int(math.floor(ch * reduction))
                self.transitions.append(_Transition(ch, cout))
                ch = cout
        self.final_norm = CatGroupNormAct(_GN, ch, relu=True)
        self.head = Linear(ch, num_classes)

    def forward(self, x):
        segs = [self.stem(x)]
        for i, block in enumerate(self.blocks):
            trans = self.transitions[i] if i < len(self.transitions) else None
            segs = FD.dense_block(block, trans, segs)
        out = self.final_norm(segs)
        out = FD.avg_pool2d(out, 4).flatten(1)
        return self.head(out)


def DenseNet121(num_classes=10):
    return DenseNet((6, 12, 24, 16), growth=32, num_classes=num_classes)


def DenseNet169(num_classes=10):
    return DenseNet((6, 12, 32, 32), growth=32, num_classes=num_classes)


def DenseNet201(num_classes=10):
    return DenseNet((6, 12, 48, 32), growth=32, num_classes=num_classes)


def DenseNet161(num_classes=10):
    return DenseNet((6, 12, 36, 24), growth=48, num_classes=num_classes)
