"""Exact-sum dataset partitioning for dynamic per-rank batch sizes.

Replaces the reference's DataPartitioner/Partition (dataloader.py:12-49)
whose float truncation (`int(frac*len)`, `batch_size*frac`) can skew
per-rank iteration counts and deadlock the synchronous all-reduce
(SURVEY.md §3.5).  Here iteration counts are common by construction:

CV:  steps = data_len // global_batch (identical on all ranks); rank *i*
     owns a contiguous slice of a seed-shuffled index list of
     ``steps * batch[i]`` samples and steps through it in batches of
     ``batch[i]``.

LM:  rows R = total_tokens // global_batch (identical on all ranks);
     rank *i* consumes ``R * batch[i]`` tokens, batchified to
     ``[R, batch[i]]``; BPTT windows over R-1 rows give a common step
     count.  Matches the reference's batchify reshape
     (dataloader.py:166-173) with exact integer widths.
"""

from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset, Subset

__all__ = ["partition_cv", "partition_lm", "batchify", "bptt_batch",
           "GlobalBatchStream"]


def _shard_bounds(batches: np.ndarray, steps: int) -> np.ndarray:
    """Cumulative sample offsets per rank: rank i owns [off[i], off[i+1])."""
    per_rank = batches.astype(np.int64) * steps
    return np.concatenate([[0], np.cumsum(per_rank)])


def partition_cv(
    dataset: Dataset,
    batches: np.ndarray,
    rank: int,
    seed: int,
    epoch: int,
    num_workers: int = 0,
):
    """Shard a map-style dataset for this epoch's integer batch split.

    Returns (loader, steps_per_epoch).  Every rank's loader yields exactly
    ``steps_per_epoch`` batches of size ``batches[rank]``.
    """
    batches = np.asarray(batches, dtype=np.int64)
    global_batch = int(batches.sum())
    data_len = len(dataset)
    steps = data_len // global_batch
    if steps == 0:
        raise ValueError(f"dataset of {data_len} samples < global batch {global_batch}")

    # Deterministic global shuffle (same on every rank), fresh each epoch.
    g = torch.Generator().manual_seed(seed * 100_003 + epoch)
    perm = torch.randperm(data_len, generator=g)
    bounds = _shard_bounds(batches, steps)
    my_idx = perm[bounds[rank]: bounds[rank + 1]]

    loader = DataLoader(
        Subset(dataset, my_idx.tolist()),
        batch_size=int(batches[rank]),
        shuffle=False,  # the global permutation already shuffles
        drop_last=True,
        num_workers=num_workers,
        pin_memory=torch.cuda.is_available(),
    )
    return loader, steps


class GlobalBatchStream:
    """Iteration-granularity CV data source (the `-dbsi` mode).

    The per-epoch sharding above fixes each rank's slice for the whole
    epoch, so the batch split can only change at epoch boundaries — the
    reference's cadence (dbs.py:385-395).  The north star asks for
    re-partitioning *every iteration*; this stream makes that sound:
    one replicated global permutation per epoch, step ``s`` consumes
    indices ``[s*GB, (s+1)*GB)``, and each rank takes the contiguous
    sub-slice its CURRENT split assigns.  The split may change between
    any two iterations without skewing sample coverage (each sample is
    seen exactly once per epoch) or iteration counts
    (``steps = len(dataset) // GB`` on every rank regardless of split).
    """

    def __init__(self, dataset: Dataset, global_batch: int, seed: int,
                 epoch: int):
        self.dataset = dataset
        self.global_batch = int(global_batch)
        data_len = len(dataset)
        self.steps = data_len // self.global_batch
        if self.steps == 0:
            raise ValueError(
                f"dataset of {data_len} samples < global batch {global_batch}")
        g = torch.Generator().manual_seed(seed * 100_003 + epoch)
        self.perm = torch.randperm(data_len, generator=g)

    def batch(self, step: int, batches: np.ndarray, rank: int):
        """(inputs, targets) for ``rank`` at global step ``step`` under the
        current integer split ``batches`` (sum == global_batch)."""
        batches = np.asarray(batches, dtype=np.int64)
        off = int(batches[:rank].sum())
        base = step * self.global_batch
        idx = self.perm[base + off: base + off + int(batches[rank])]
        tensors = getattr(self.dataset, "tensors", None)
        if tensors is not None:  # TensorDataset fast path (synthetic data)
            return tuple(t[idx] for t in tensors)
        items = [self.dataset[int(i)] for i in idx]
        return torch.utils.data.default_collate(items)


def batchify(tokens: torch.Tensor, width: int) -> torch.Tensor:
    """[rows, width] column-major reshape of a token stream (LM layout)."""
    rows = tokens.numel() // width
    return tokens[: rows * width].view(width, rows).t().contiguous()


def bptt_batch(source: torch.Tensor, i: int, bptt: int):
    """(data, flat target) = rows [i, i+L) / [i+1, i+1+L) — reference
    utils.py:7-11 semantics."""
    seq_len = min(bptt, source.size(0) - 1 - i)
    return source[i: i + seq_len], source[i + 1: i + 1 + seq_len].reshape(-1)


def partition_lm(
    tokens: torch.Tensor,
    batches: np.ndarray,
    rank: int,
    bptt: int = 35,
):
    """Shard a token stream into this rank's [R, batch[rank]] sheet.

    Returns (sheet, steps_per_epoch).  R = tokens // global_batch is common
    to all ranks, so the BPTT step count is too.
    """
    batches = np.asarray(batches, dtype=np.int64)
    global_batch = int(batches.sum())
    rows = tokens.numel() // global_batch
    if rows < 2:
        raise ValueError("token stream too short for this global batch")
    bounds = _shard_bounds(batches, rows)
    my_tokens = tokens[bounds[rank]: bounds[rank + 1]]
    sheet = batchify(my_tokens, int(batches[rank]))
    steps = (rows - 1 + bptt - 1) // bptt  # number of BPTT windows
    return sheet, steps
