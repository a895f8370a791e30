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

__all__ = ["partition_cv", "partition_lm", "batchify", "bptt_batch"]


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
