"""Autograd bindings over the gfx950 kernels (_dlb_kernels).

Tensors cross into kernels in NHWC ([N, HW, C] contiguous) — the natural
layout for channels_last CV activations: ``permute(0,2,3,1)`` is a free
view on a channels_last tensor.
"""

from __future__ import annotations

import torch

from . import ext


def _to_nhwc3(x: torch.Tensor) -> torch.Tensor:
    n, c, h, w = x.shape
    x3 = x.permute(0, 2, 3, 1).reshape(n, h * w, c)
    return x3 if x3.is_contiguous() else x3.contiguous()


class _GroupNormAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, num_groups, weight, bias, eps, relu):
        n, c, h, w = x.shape
        x3 = _to_nhwc3(x)
        y3, mean, rstd = ext().gn_fwd(x3, weight, bias, num_groups, eps, relu)
        ctx.save_for_backward(x3, weight, bias, mean, rstd)
        ctx.gn_dims = (n, c, h, w, num_groups, relu)
        return y3.view(n, h, w, c).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dz):
        x3, weight, bias, mean, rstd = ctx.saved_tensors
        n, c, h, w, groups, relu = ctx.gn_dims
        dz3 = _to_nhwc3(dz)
        dx3, dgamma, dbeta = ext().gn_bwd(x3, dz3, weight, bias, mean, rstd,
                                          groups, relu)
        dx = dx3.view(n, h, w, c).permute(0, 3, 1, 2)
        return dx, None, dgamma, dbeta, None, None


def group_norm_act(x, num_groups, weight, bias, eps=1e-5, relu=False):
    return _GroupNormAct.apply(x, num_groups, weight, bias, eps, relu)


def gn_native_ok(x, num_groups, weight) -> bool:
    """Shape/dtype envelope the fused GN kernel covers (everything the
    zoo produces on the GPU path)."""
    return (x.dim() == 4 and x.dtype == torch.bfloat16
            and weight.dtype == torch.float32
            and x.shape[1] % 8 == 0 and num_groups <= 64
            and x.shape[1] % num_groups == 0)
