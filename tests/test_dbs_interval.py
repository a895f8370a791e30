"""Iteration-granularity DBS (`-dbsi`) and the straggler idle % metric.

The north star asks for re-partitioning "every iteration"
(BASELINE.json); the reference's cadence is per-epoch
(/root/reference/dbs.py:385-390).  `-dbsi N` provides the
iteration-granularity mode; these tests pin that the partition responds
MID-EPOCH to an injected straggle, and that the GlobalBatchStream's
coverage/step-count invariants hold under mid-epoch split changes.
"""

import time

import numpy as np
import pytest
import torch

from dynamic_load_balance_distributeddnn_amd.data import GlobalBatchStream
from dynamic_load_balance_distributeddnn_amd.scheduler import \
    straggler_idle_pct
from tests.test_distributed_cpu import run_distributed


# ---------------------------------------------------------------- idle %
def test_idle_pct_balanced_is_zero():
    assert straggler_idle_pct(np.array([2.0, 2.0, 2.0, 2.0])) == 0.0


def test_idle_pct_one_straggler():
    # 3 ranks at 1s waiting on one at 2s: each fast rank idles 1s of the
    # 2s window -> 3s idle over 8 rank-seconds = 37.5%
    assert straggler_idle_pct(np.array([1.0, 1.0, 1.0, 2.0])) == pytest.approx(37.5)


def test_idle_pct_degenerate():
    assert straggler_idle_pct(np.array([])) == 0.0
    assert straggler_idle_pct(np.array([0.0, 0.0])) == 0.0
    assert straggler_idle_pct(np.array([3.0])) == 0.0


# ------------------------------------------------------------- the stream
def test_stream_covers_every_sample_once_under_split_changes():
    xs = torch.arange(100, dtype=torch.float32).unsqueeze(1)
    ys = torch.arange(100)
    ds = torch.utils.data.TensorDataset(xs, ys)
    stream = GlobalBatchStream(ds, global_batch=10, seed=1, epoch=0)
    assert stream.steps == 10

    # split changes midway; union over ranks must still cover each global
    # batch slice exactly, and per-step rank batch sizes follow the split
    seen = []
    for s in range(stream.steps):
        split = np.array([7, 3]) if s < 5 else np.array([2, 8])
        for rank in range(2):
            x, y = stream.batch(s, split, rank)
            assert x.shape[0] == split[rank]
            seen.extend(y.tolist())
    assert sorted(seen) == list(range(100))


def test_stream_deterministic_across_ranks():
    ds = torch.utils.data.TensorDataset(torch.randn(64, 3), torch.arange(64))
    a = GlobalBatchStream(ds, 8, seed=5, epoch=3)
    b = GlobalBatchStream(ds, 8, seed=5, epoch=3)
    assert torch.equal(a.perm, b.perm)
    c = GlobalBatchStream(ds, 8, seed=5, epoch=4)
    assert not torch.equal(a.perm, c.perm)


def test_stream_generic_dataset_collate():
    class L(torch.utils.data.Dataset):
        def __len__(self):
            return 20

        def __getitem__(self, i):
            return torch.full((2,), float(i)), i

    stream = GlobalBatchStream(L(), 5, seed=0, epoch=0)
    x, y = stream.batch(0, np.array([3, 2]), 0)
    assert x.shape == (3, 2) and y.shape == (3,)


# ------------------------------------------- mid-epoch response (2 ranks)
def _interval_worker(rank, world):
    from dynamic_load_balance_distributeddnn_amd.cli import get_parser
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = get_parser().parse_args(
        ["-d", "true", "-ws", "2", "-b", "16", "-e", "1",
         "-ds", "mnist", "-m", "mnistnet", "-dbsi", "2"])
    tr = Trainer(args, rank, world, torch.device("cpu"), logger=None)
    if rank == 1:  # induced straggler: every forward costs +60 ms
        tr.model.register_forward_hook(lambda *a: time.sleep(0.06))
    tr.train_epoch(0)
    return tr.sched.batches.tolist()


def test_partition_shifts_mid_epoch(free_port, monkeypatch):
    """One epoch only: per-epoch cadence cannot shift the split during
    epoch 0 (the solver sees times only at epoch end), so any shift here
    proves the iteration-granularity path reacted mid-epoch."""
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.002")
    res = run_distributed(_interval_worker, 2, free_port)
    assert res[0] == res[1]            # replicated-deterministic
    assert sum(res[0]) == 16           # exact-sum invariant held
    assert res[0][1] < res[0][0]       # straggler lost batch share
