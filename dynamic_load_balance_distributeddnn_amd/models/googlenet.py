"""CIFAR-shape GoogLeNet (Inception) with GroupNorm.

Architecture parity with reference Net/GoogleNet.py, EXCEPT one deliberate
fix: the reference's 5x5-reduce branch places GroupNorm(8, n5x5red) BEFORE
its 1x1 conv (Net/GoogleNet.py:29-30), which is a confirmed shape bug —
the GN sees in_planes channels but is sized for n5x5red, so the model
crashes on first forward.  We implement the corrected order
(conv -> GN -> ReLU, matching the working 3x3 branch at
Net/GoogleNet.py:18-25), as SURVEY.md §2.3 mandates.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as FD
from ..ops.layers import Conv2d, GroupNormAct, Linear


def _cbr(cin, cout, k, groups_gn=8):
    """conv(k, bias) -> GN -> fused ReLU (GoogLeNet convs carry bias)."""
    return nn.Sequential(
        Conv2d(cin, cout, k, padding=k // 2, bias=True),
        GroupNormAct(groups_gn, cout, relu=True),
    )


class _MaxPool(nn.Module):
    def __init__(self, k, stride, pad):
        super().__init__()
        self.k, self.stride, self.pad = k, stride, pad

    def forward(self, x):
        return FD.max_pool2d(x, self.k, self.stride, self.pad)


class _Inception(nn.Module):
    def __init__(self, cin, n1x1, n3x3red, n3x3, n5x5red, n5x5, pool_planes):
        super().__init__()
        self.branch1 = _cbr(cin, n1x1, 1)
        self.branch2 = nn.Sequential(
            _cbr(cin, n3x3red, 1),
            Conv2d(n3x3red, n3x3, 3, padding=1, bias=True),
            GroupNormAct(16, n3x3, relu=True),
        )
        # "5x5" branch = 1x1 reduce + two stacked 3x3 (reference layout),
        # with the GN placed AFTER the reduce conv (bug fix, see module doc).
        self.branch3 = nn.Sequential(
            _cbr(cin, n5x5red, 1),
            _cbr(n5x5red, n5x5, 3),
            _cbr(n5x5, n5x5, 3),
        )
        self.branch4_pool = _MaxPool(3, 1, 1)
        self.branch4 = nn.Sequential(
            _cbr(cin, pool_planes, 1),
        )

    def forward(self, x):
        # one Function per module on GPU (ops/googblock.py): branch
        # data-grads accumulate in the conv epilogues instead of three
        # autograd adds; composes per-branch elsewhere
        return FD.inception(self, x)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.stem = _cbr(3, 192, 3)
        self.inc3 = nn.Sequential(
            _Inception(192, 64, 96, 128, 16, 32, 32),
            _Inception(256, 128, 128, 192, 32, 96, 64),
        )
        self.pool = _MaxPool(3, 2, 1)
        self.inc4 = nn.Sequential(
            _Inception(480, 192, 96, 208, 16, 48, 64),
            _Inception(512, 160, 112, 224, 24, 64, 64),
            _Inception(512, 128, 128, 256, 24, 64, 64),
            _Inception(512, 112, 144, 288, 32, 64, 64),
            _Inception(528, 256, 160, 320, 32, 128, 128),
        )
        self.inc5 = nn.Sequential(
            _Inception(832, 256, 160, 320, 32, 128, 128),
            _Inception(832, 384, 192, 384, 48, 128, 128),
        )
        self.head = Linear(1024, num_classes)

    def forward(self, x):
        out = self.inc3(self.stem(x))
        out = self.inc4(self.pool(out))
        out = self.inc5(self.pool(out))
        out = FD.avg_pool2d(out, 8).flatten(1)
        return self.head(out)
