"""Flat fused SGD-with-momentum.

The reference's optimizer is torch.optim.SGD(lr, momentum=0.9)
(dbs.py:369), stepped per-tensor — up to 362 tensors for DenseNet-121.
Here parameters are re-homed into ONE flat fp32 arena laid out identically
to GradientSynchronizer's gradient arena, and momentum is a third flat
buffer, so the whole update is a single HBM-bound HIP kernel
(ops/csrc/sgd.hip) per step: v = mu*v + g ; p -= lr*v.

Math matches torch SGD with dampening=0, nesterov=False exactly.
CPU (debug mode) runs the same flat update with three torch ops.
"""

from __future__ import annotations

import torch

from .grad_sync import GradientSynchronizer

__all__ = ["FlatSGD"]


class FlatSGD:
    def __init__(self, sync: GradientSynchronizer, lr: float,
                 momentum: float = 0.9):
        self.sync = sync
        self.momentum = momentum
        self.param_groups = [{"lr": lr}]  # LR-policy-compatible surface

        total = sync.arena.numel()
        device = sync.arena.device
        self.param_arena = torch.empty(total, dtype=torch.float32, device=device)
        self.momentum_buf = torch.zeros(total, dtype=torch.float32, device=device)
        from .grad_sync import _strided_view

        # bf16 weight mirror: the SGD kernel writes updated params as
        # bf16 into this arena in the same pass, and conv forwards read
        # weights from per-param views of it — removing the ~120
        # per-tensor fp32->bf16 cast kernels per step (profiles/).
        self.bf16_mirror = (torch.empty(total, dtype=torch.bfloat16,
                                        device=device)
                            if self.param_arena.is_cuda else None)
        with torch.no_grad():
            for p in sync.params:
                off, n = sync.offsets[id(p)]
                view = _strided_view(self.param_arena, off, p)
                view.copy_(p.data)
                p.data = view  # re-home the parameter into the arena
                # conv (4D) and matmul (2D) weights get bf16 mirror
                # views — the compute kernels read weights as bf16
                if self.bf16_mirror is not None and p.dim() in (2, 4):
                    p._dlb_bf16 = _strided_view(self.bf16_mirror, off, p)
            self.refresh_mirror()

    @torch.no_grad()
    def refresh_mirror(self) -> None:
        """Re-sync the bf16 weight mirror after any out-of-band write to
        the parameter arena (initial sync, checkpoint load)."""
        if self.bf16_mirror is not None:
            self.bf16_mirror.copy_(self.param_arena)

    @torch.no_grad()
    def step(self) -> None:
        lr = float(self.param_groups[0]["lr"])
        g = self.sync.arena
        if g.is_cuda:
            from ..ops import ext

            ext().sgd_momentum(self.param_arena, g, self.momentum_buf,
                               lr, self.momentum, self.bf16_mirror)
        else:
            self.momentum_buf.mul_(self.momentum).add_(g)
            self.param_arena.add_(self.momentum_buf, alpha=-lr)

    def zero_grad(self, set_to_none: bool = False) -> None:
        self.sync.zero()
