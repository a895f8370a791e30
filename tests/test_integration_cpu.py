"""End-to-end CPU integration: the dbs.py CLI, LM engine over gloo, and
the fault-injection -> partition-shift loop."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from tests.test_distributed_cpu import run_distributed

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_dbs_cli_end_to_end(tmp_path, free_port):
    """`python dbs.py` (the frozen reference CLI) runs 2 CPU workers for
    one epoch and writes the reference's artifact layout."""
    env = dict(os.environ, MASTER_PORT=str(free_port), MASTER_ADDR="127.0.0.1",
               DLB_SYNTH_SCALE="0.002")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "dbs.py"), "-d", "true",
         "-ws", "2", "-b", "16", "-e", "1", "-ds", "mnist", "-m", "mnistnet"],
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    logs = os.listdir(tmp_path / "logs")
    assert len(logs) == 2  # one per rank
    stats = os.listdir(tmp_path / "statis")
    assert len(stats) == 1
    rec = np.load(tmp_path / "statis" / stats[0], allow_pickle=True).item()
    assert rec["epoch"] == [0]
    assert len(rec["partition"][0]) == 2

    # idempotency guard: second invocation skips
    out2 = subprocess.run(
        [sys.executable, os.path.join(REPO, "dbs.py"), "-d", "true",
         "-ws", "2", "-b", "16", "-e", "1", "-ds", "mnist", "-m", "mnistnet"],
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=120)
    assert "skipping" in out2.stdout


def _lm_worker(rank, world):
    from dynamic_load_balance_distributeddnn_amd.cli import get_parser
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = get_parser().parse_args(
        ["-d", "true", "-ws", "2", "-b", "8", "-e", "1",
         "-ds", "wikitext2", "-m", "transformer"])
    tr = Trainer(args, rank, world, torch.device("cpu"), logger=None)
    compute, sync, loss = tr.train_epoch(0)
    val_loss, _ = tr.validate_epoch(0)
    return dict(loss=loss, val=val_loss, batches=tr.sched.batches.tolist())


def test_transformer_two_worker_gloo(free_port, monkeypatch):
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.002")
    res = run_distributed(_lm_worker, 2, free_port)
    assert res[0]["batches"] == res[1]["batches"]
    assert sum(res[0]["batches"]) == 8
    for r in range(2):
        assert np.isfinite(res[r]["loss"])


def _ft_worker(rank, world):
    import numpy as np

    from dynamic_load_balance_distributeddnn_amd.scheduler import (
        DBSScheduler, exchange_times)
    from dynamic_load_balance_distributeddnn_amd.utils import FaultInjector

    # rank 1 is forced into a slow phase (chance=1, fixed seed)
    fi = FaultInjector(enabled=(rank == 1), chance=1.0, rank=rank, seed=7)
    sched = DBSScheduler(world, global_batch=64)
    history = []
    for epoch in range(6):
        injected = sum(fi.maybe_wait(epoch, 20) for _ in range(20))
        compute = 0.05 + injected
        sched.step(exchange_times(compute))
        history.append(sched.batches.tolist())
    return history


def test_fault_injection_shifts_partition(free_port):
    """The reference's -ft robustness demo, made deterministic: the
    injected straggler must end with a smaller batch share."""
    res = run_distributed(_ft_worker, 2, free_port)
    final = res[0][-1]
    assert final == res[1][-1]
    assert final[1] < final[0]
    assert sum(final) == 64
