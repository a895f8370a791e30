"""One-cycle learning-rate policy (reference adjust_learning_rate,
dbs.py:193-215).

Behavioral parity note: the reference documents a full one-cycle schedule
(warm-up 30%, plateau, decay 30%) but its warm-up branch is commented out
(dbs.py:206-208), so ONLY the final-30% linear decay is live.  We
reproduce the live behavior; the full policy is available behind
``full=True`` for users who want the documented schedule.
"""

from __future__ import annotations

__all__ = ["one_cycle_lr"]


def one_cycle_lr(base_lr: float, epoch: int, epoch_size: int,
                 full: bool = False) -> float:
    if full and epoch < 0.3 * epoch_size:
        return 0.01 * base_lr + (0.99 * base_lr / (0.3 * epoch_size)) * epoch
    if 0.7 * epoch_size <= epoch < epoch_size:
        # reference's live branch, including its (epoch - 0.7*epoch) form
        return base_lr - (0.99 * base_lr / (0.3 * epoch_size)) * (epoch - 0.7 * epoch)
    return base_lr


def apply_lr(optimizer, lr: float) -> None:
    for group in optimizer.param_groups:
        group["lr"] = lr
