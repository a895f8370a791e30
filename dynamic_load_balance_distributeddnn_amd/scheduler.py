"""DBS partition solver + per-epoch time exchange.

This is the paper's contribution rebuilt with exact integer accounting.

Reference semantics (cited into /root/reference):
- ``get_size`` (dbs.py:458-476): next epoch's batch share of rank *i* is
  proportional to ``partition[i] / nodes_time[i]`` — a rank that was twice as
  slow per-sample gets half the samples, so all ranks finish an iteration at
  the same wall time.  The reference then float-truncates both the per-rank
  dataset shard (dataloader.py:43) and the per-rank batch size
  (dataloader.py:45,114), which can skew per-rank iteration counts by ±1 and
  deadlock the per-iteration all-reduce.
- ``time_allreduce`` (dbs.py:479-499): a hand-rolled gloo ring that leaves
  every rank with the rank-ordered vector of pure-compute times.

This rebuild:
- ``solve_partition`` returns integer per-rank batch sizes that sum EXACTLY
  to the global batch (largest-remainder rounding, deterministic, replicated
  on every rank), each >= a floor so no rank starves.
- iteration counts are common by construction: every rank runs
  ``steps_per_epoch`` iterations of its own ``batch[i]`` samples.
- the time exchange is a single ``all_gather`` of one float per rank
  (RCCL over xGMI on GPU, gloo on CPU) — same contract, one collective.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import numpy as np
import torch
import torch.distributed as dist

__all__ = [
    "solve_partition",
    "exchange_times",
    "straggler_idle_pct",
    "DBSScheduler",
]


def straggler_idle_pct(nodes_time: np.ndarray) -> float:
    """Straggler idle percentage — the second half of the BASELINE metric.

    With synchronous SGD every rank waits for the slowest, so the fraction
    of whole-node compute capacity lost to stragglers over a window is

        idle% = 100 * Σ_rank (max_t − t_rank) / (N * max_t)

    computed from the rank-ordered pure-compute vector the DBS loop
    already exchanges (the reference's node_time, dbs.py:425).  0 means
    perfectly balanced; DBS drives this toward 0 while a fixed split
    under a straggler pins it high.
    """
    t = np.asarray(nodes_time, dtype=np.float64)
    if t.size == 0:
        return 0.0
    m = float(t.max())
    if m <= 0:
        return 0.0
    return float(100.0 * (m - t).sum() / (t.size * m))


def solve_partition(
    nodes_time: np.ndarray,
    partition_frac: np.ndarray,
    global_batch: int,
    min_per_rank: int = 1,
) -> np.ndarray:
    """One DBS step: new integer per-rank batch sizes from last epoch's times.

    The continuous target is ``frac[i] ∝ partition_frac[i] / nodes_time[i]``
    (the reference's cons_k formulation at dbs.py:459-463 — per-sample speed
    estimated from last epoch's share and time).  Rounding is
    largest-remainder so the result sums exactly to ``global_batch``;
    ties broken by rank index for cross-rank determinism.

    Args:
        nodes_time: rank-ordered pure compute seconds of last epoch, > 0.
        partition_frac: last epoch's batch fractions (sum ≈ 1).
        global_batch: total samples per iteration across all ranks.
        min_per_rank: floor for any rank's batch (keeps every rank
            participating so the weighted all-reduce stays well-defined).

    Returns:
        int64 array of per-rank batch sizes, ``sum == global_batch``.
    """
    nodes_time = np.asarray(nodes_time, dtype=np.float64)
    partition_frac = np.asarray(partition_frac, dtype=np.float64)
    n = len(nodes_time)
    if global_batch < n * min_per_rank:
        raise ValueError(
            f"global_batch={global_batch} cannot give {min_per_rank} "
            f"sample(s) to each of {n} ranks"
        )
    if np.any(nodes_time <= 0):
        # Degenerate timing (first epoch, clock glitch): fall back to equal.
        speed = np.ones(n)
    else:
        # per-sample speed ∝ share/time; new share ∝ speed
        speed = partition_frac / nodes_time
        if not np.all(np.isfinite(speed)) or speed.sum() <= 0:
            speed = np.ones(n)

    target = speed / speed.sum() * global_batch
    # Largest-remainder with a per-rank floor.
    floor = np.maximum(np.floor(target).astype(np.int64), min_per_rank)
    # If floors overshoot (extreme skew + min_per_rank), walk back from the
    # largest allocations deterministically.
    while floor.sum() > global_batch:
        over = int(np.argmax(np.where(floor > min_per_rank, floor, -1)))
        floor[over] -= 1
    remainder = global_batch - int(floor.sum())
    if remainder > 0:
        frac_part = target - np.floor(target)
        # stable order: biggest fractional part first, then lowest rank
        order = np.lexsort((np.arange(n), -frac_part))
        for k in range(remainder):
            floor[order[k % n]] += 1
    assert floor.sum() == global_batch
    return floor


def exchange_times(my_time: float, device: torch.device | str = "cpu") -> np.ndarray:
    """All-gather each rank's pure-compute time; returns rank-ordered vector.

    Replaces the reference's size-1 isend/recv ring (dbs.py:479-499) with one
    all_gather_into_tensor.  On the RCCL backend the 4-byte payload rides
    xGMI; on gloo it stays on CPU.  Every rank returns the identical
    ``[t_0, ..., t_{n-1}]`` — the invariant the solver's determinism needs.
    """
    world = dist.get_world_size()
    backend = dist.get_backend()
    dev = torch.device(device) if backend != "gloo" else torch.device("cpu")
    send = torch.tensor([float(my_time)], dtype=torch.float32, device=dev)
    out = torch.empty(world, dtype=torch.float32, device=dev)
    dist.all_gather_into_tensor(out, send)
    return out.cpu().numpy().astype(np.float64)


@dataclass
class DBSScheduler:
    """Replicated-deterministic per-epoch feedback loop.

    Every rank holds identical state and runs identical updates from the
    identical all-gathered time vector — no coordinator, mirroring the
    reference's design (dbs.py:385-426) with exact integer splits.
    """

    world_size: int
    global_batch: int
    enabled: bool = True
    min_per_rank: int = 1
    batches: np.ndarray = field(init=False)

    def __post_init__(self) -> None:
        base = self.global_batch // self.world_size
        extra = self.global_batch % self.world_size
        self.batches = np.full(self.world_size, base, dtype=np.int64)
        self.batches[:extra] += 1  # exact even-ish start, sums to global_batch

    @property
    def fractions(self) -> np.ndarray:
        return self.batches.astype(np.float64) / float(self.global_batch)

    @property
    def weights(self) -> np.ndarray:
        """Gradient-averaging weights = exact batch share (dbs.py:293)."""
        return self.fractions

    def step(self, nodes_time: np.ndarray) -> np.ndarray:
        """Feed last epoch's rank-ordered times; returns new batch sizes."""
        if self.enabled:
            self.batches = solve_partition(
                nodes_time, self.fractions, self.global_batch, self.min_per_rank
            )
        return self.batches
