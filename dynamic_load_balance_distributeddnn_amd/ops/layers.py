"""Module wrappers over ops.functional — the zoo builds from these.

These are thin parameter holders; the execution path (gfx950 HIP kernel
vs CPU torch composition) is decided in ops/functional.py.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from . import functional as FD


class Conv2d(nn.Module):
    def __init__(self, in_ch, out_ch, kernel_size, stride=1, padding=0,
                 groups=1, bias=False):
        super().__init__()
        if isinstance(kernel_size, int):
            kernel_size = (kernel_size, kernel_size)
        self.stride, self.padding, self.groups = stride, padding, groups
        self.weight = nn.Parameter(
            torch.empty(out_ch, in_ch // groups, *kernel_size))
        # match nn.Conv2d's kaiming-uniform init so models are numerically
        # comparable to stock torch versions under a fixed seed
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if bias:
            fan_in = in_ch // groups * kernel_size[0] * kernel_size[1]
            bound = 1 / math.sqrt(fan_in)
            self.bias = nn.Parameter(torch.empty(out_ch).uniform_(-bound, bound))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return FD.conv2d(x, self.weight, self.bias, self.stride,
                         self.padding, self.groups)


class GroupNormAct(nn.Module):
    """GroupNorm with optional fused ReLU.

    The zoo keeps GroupNorm (never BatchNorm) because DBS varies per-rank
    batch sizes and BatchNorm statistics would diverge across ranks
    (reference Net/Resnet.py:11 et al.) — load-bearing, preserved here.
    """

    def __init__(self, num_groups, num_channels, eps=1e-5, relu=False):
        super().__init__()
        self.num_groups, self.eps, self.relu = num_groups, eps, relu
        self.weight = nn.Parameter(torch.ones(num_channels))
        self.bias = nn.Parameter(torch.zeros(num_channels))

    def forward(self, x):
        return FD.group_norm_act(x, self.num_groups, self.weight, self.bias,
                                 self.eps, self.relu)


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if bias:
            bound = 1 / math.sqrt(in_features)
            self.bias = nn.Parameter(torch.empty(out_features).uniform_(-bound, bound))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return FD.linear(x, self.weight, self.bias)


class LayerNorm(nn.Module):
    def __init__(self, normalized_shape, eps=1e-5):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(*self.normalized_shape))
        self.bias = nn.Parameter(torch.zeros(*self.normalized_shape))

    def forward(self, x):
        return FD.layer_norm(x, self.normalized_shape, self.weight, self.bias,
                             self.eps)
