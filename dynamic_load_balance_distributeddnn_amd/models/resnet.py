"""CIFAR-shape ResNet family with GroupNorm(32).

Architecture parity with reference Net/Resnet.py (ResNet-18/34/50/101/152,
3x3 stem, 4 stages, avg_pool2d(4), linear head; GroupNorm not BatchNorm —
batch-size-independent normalization is required by DBS's heterogeneous
per-rank batches).  Implementation is our own: config-driven stages over
the ops.layers wrappers so the GPU path runs the gfx950 kernels.
"""

from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as FD
from ..ops.layers import Conv2d, GroupNormAct, Linear

_GN = 32


def _proj(cin, cout, stride):
    """1x1 projection shortcut (conv + GN, no activation)."""
    return nn.Sequential(
        Conv2d(cin, cout, 1, stride=stride),
        GroupNormAct(_GN, cout),
    )


class _Basic(nn.Module):
    expansion = 1

    def __init__(self, cin, width, stride):
        super().__init__()
        cout = width * self.expansion
        self.a = nn.Sequential(
            Conv2d(cin, width, 3, stride=stride, padding=1),
            GroupNormAct(_GN, width, relu=True),
            Conv2d(width, width, 3, padding=1),
        )
        self.norm_out = GroupNormAct(_GN, cout)  # params for the fused tail
        self.proj = _proj(cin, cout, stride) if (stride != 1 or cin != cout) else None

    def forward(self, x):
        out = self.a(x)
        res = self.proj(x) if self.proj is not None else x
        # relu(GN(out) + res) in one kernel (ops.functional)
        return FD.group_norm_add_act(out, res, _GN, self.norm_out.weight,
                                     self.norm_out.bias, self.norm_out.eps)


class _Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride):
        super().__init__()
        cout = width * self.expansion
        self.a = nn.Sequential(
            Conv2d(cin, width, 1),
            GroupNormAct(_GN, width, relu=True),
            Conv2d(width, width, 3, stride=stride, padding=1),
            GroupNormAct(_GN, width, relu=True),
            Conv2d(width, cout, 1),
        )
        self.norm_out = GroupNormAct(_GN, cout)  # params for the fused tail
        self.proj = _proj(cin, cout, stride) if (stride != 1 or cin != cout) else None

    def forward(self, x):
        # one Function per block on GPU: junction add fused into conv1's
        # data-grad epilogue, direct-arena weight grads, batched
        # dgamma/dbeta (ops/resblock.py); composes per-layer elsewhere
        return FD.res_bottleneck(self, x)


class ResNet(nn.Module):
    def __init__(self, block, depths, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(
            Conv2d(3, 64, 3, padding=1),
            GroupNormAct(_GN, 64, relu=True),
        )
        stages, cin = [], 64
        for i, (width, depth) in enumerate(zip((64, 128, 256, 512), depths)):
            blocks = []
            for j in range(depth):
                stride = 2 if (i > 0 and j == 0) else 1
                blocks.append(block(cin, width, stride))
                cin = width * block.expansion
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.head = Linear(cin, num_classes)

    def forward(self, x):
        out = self.stages(self.stem(x))
        out = FD.avg_pool2d(out, 4).flatten(1)
        return self.head(out)


def ResNet18(num_classes=10):
    return ResNet(_Basic, (2, 2, 2, 2), num_classes)


def ResNet34(num_classes=10):
    return ResNet(_Basic, (3, 4, 6, 3), num_classes)


def ResNet50(num_classes=10):
    return ResNet(_Bottleneck, (3, 4, 6, 3), num_classes)


def ResNet101(num_classes=10):
    return ResNet(_Bottleneck, (3, 4, 23, 3), num_classes)


def ResNet152(num_classes=10):
    return ResNet(_Bottleneck, (3, 8, 36, 3), num_classes)
