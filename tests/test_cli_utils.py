"""CLI surface, filename schema, LR policy, fault injector, recorder."""

import os

import numpy as np

from dynamic_load_balance_distributeddnn_amd.cli import (base_filename,
                                                         get_parser)
from dynamic_load_balance_distributeddnn_amd.utils import (FaultInjector,
                                                           StatsRecorder)
from dynamic_load_balance_distributeddnn_amd.utils.lr_policy import \
    one_cycle_lr


def test_default_flags_match_reference():
    args = get_parser().parse_args([])
    assert args.debug is True
    assert args.world_size == 4
    assert args.batch_size == 64
    assert args.learning_rate == 0.01
    assert args.epoch_size == 10
    assert args.dataset == "wikitext2"
    assert args.dynamic_batch_size is True
    assert args.gpu == 0
    assert args.model == "transformer"
    assert args.fault_tolerance is False
    assert args.fault_tolerance_chance == 0.1
    assert args.one_cycle_policy is False
    assert args.disable_enhancements is False


def test_gpu_map_parsing():
    args = get_parser().parse_args(["-gpu", "0,0,1,2"])
    assert args.gpu == [0, 0, 1, 2]
    args = get_parser().parse_args(["-gpu", "3"])
    assert args.gpu == 3


def test_base_filename_schema():
    """Byte-parity with reference dbs.py:54-61."""
    args = get_parser().parse_args(
        ["-m", "densenet", "-ds", "cifar10", "-d", "false", "-ws", "4",
         "-b", "512", "-lr", "0.01", "-e", "10"])
    name = base_filename(args)
    assert name == ("densenet-cifar10-debug0-n4-bs512-lr0.0100-ep10-dbs1-"
                    "ft0-ftc0.100000-node{}-ocp0")
    args = get_parser().parse_args(["-de", "true"])
    assert base_filename(args).startswith("puredbs=")


def test_one_cycle_only_final_decay_live():
    """Parity with the reference's live behavior (warm-up commented out,
    dbs.py:206-212)."""
    lr = 0.1
    assert one_cycle_lr(lr, 0, 10) == lr          # no warm-up
    assert one_cycle_lr(lr, 5, 10) == lr          # plateau
    assert one_cycle_lr(lr, 8, 10) < lr           # final 30% decays
    # full (documented) policy has warm-up
    assert one_cycle_lr(lr, 0, 10, full=True) < lr


def test_fault_injector_deterministic_and_bounded():
    fi = FaultInjector(enabled=True, chance=1.0, rank=0, seed=42)
    d = fi.maybe_wait(0, steps_per_epoch=1000)
    assert d == 0.0  # roll happens at first call; waiting starts next epoch
    assert fi.slow_until_epoch >= 4
    assert 5 <= fi.extra_per_epoch <= 10
    d = fi.maybe_wait(1, steps_per_epoch=10_000)
    assert d > 0
    # disabled injector is a no-op
    assert FaultInjector(False, 1.0, 0).maybe_wait(0, 10) == 0.0


def test_fault_injector_rolls_once_per_epoch():
    fi = FaultInjector(enabled=True, chance=0.0, rank=0, seed=1)
    fi.maybe_wait(0, 10)
    first = fi._last_rolled_epoch
    fi.maybe_wait(0, 10)
    assert fi._last_rolled_epoch == first == 0


def test_recorder_layout(tmp_path):
    rec = StatsRecorder("m-ds-node{}-x", output_dir=str(tmp_path / "statis"))
    rec.append(epoch=0, train_loss=1.0, train_time=2.0, sync_time=0.1,
               val_loss=0.5, accuracy=90.0, partition=np.array([0.5, 0.5]),
               node_time=np.array([1.0, 1.0]), wallclock_time=3.0)
    path = rec.save()
    assert os.path.basename(path) == "m-ds-node0-x.npy"
    loaded = np.load(path, allow_pickle=True).item()
    assert loaded["epoch"] == [0]
    # the reference's 9 lists plus the straggler_idle_pct extension
    assert set(loaded.keys()) == {
        "epoch", "train_loss", "train_time", "sync_time", "val_loss",
        "accuracy", "partition", "node_time", "wallclock_time",
        "straggler_idle_pct"}


def test_checkpoint_roundtrip(tmp_path, monkeypatch):
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.002")
    import torch

    from dynamic_load_balance_distributeddnn_amd.cli import get_parser
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = get_parser().parse_args(
        ["-d", "true", "-ws", "1", "-b", "8", "-e", "1", "-ds", "mnist",
         "-m", "mnistnet"])
    tr = Trainer(args, 0, 1, torch.device("cpu"), logger=None)
    tr.train_epoch(0)
    path = str(tmp_path / "ck.pt")
    tr.save_checkpoint(path)
    ref = tr.optimizer.param_arena.clone()

    tr2 = Trainer(args, 0, 1, torch.device("cpu"), logger=None)
    assert not torch.allclose(tr2.optimizer.param_arena, ref)
    tr2.load_checkpoint(path)
    assert torch.allclose(tr2.optimizer.param_arena, ref)
    assert (tr2.sched.batches == tr.sched.batches).all()
