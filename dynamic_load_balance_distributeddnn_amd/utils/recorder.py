"""Per-rank logging + rank-0 .npy stats recorder.

Artifact layout parity (the north star freezes it):
- logs:   ./logs/<base_filename % rank>.log  (dbs_logging.py:27-29)
- stats:  ./statis/<base_filename % 0>.npy — a pickled dict of 9 lists
          (dbs.py:316-326, 440-442).  Unlike the reference we mkdir
          ./statis (the reference crashes if it is absent).

Log line format embeds world size / lr / dbs / ft metadata like the
reference's LoggerAdapter (dbs_logging.py:10-33).
"""

from __future__ import annotations

import logging
import os
import socket

import numpy as np

__all__ = ["init_logger", "StatsRecorder"]

FIELDS = ("epoch", "train_loss", "train_time", "sync_time", "val_loss",
          "accuracy", "partition", "node_time", "wallclock_time",
          # extension beyond the reference's 9 lists: the BASELINE
          # metric's "straggler idle %" per epoch, derived from node_time
          # (scheduler.straggler_idle_pct) — the 9 reference lists above
          # keep their exact names and order.
          "straggler_idle_pct")


def init_logger(args, rank: int, base_filename: str, output_dir: str = "./logs"):
    os.makedirs(output_dir, exist_ok=True)
    extra = {
        "world_size": args.world_size,
        "lr": args.learning_rate,
        "dbs": "enabled" if args.dynamic_batch_size else "disabled",
        "ft": "enabled" if args.fault_tolerance else "disabled",
    }
    logger = logging.getLogger(f"{socket.gethostname()}.r{rank}")
    for h in logger.handlers[:]:
        logger.removeHandler(h)
    logger.setLevel(logging.DEBUG)
    logger.propagate = False
    fmt = logging.Formatter(
        "%(asctime)s [%(world_size)s:%(lr)s:dbs_%(dbs)s:ft_%(ft)s] "
        "[%(filename)s:%(lineno)d] %(levelname)s %(message)s")
    sh = logging.StreamHandler()
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    fh = logging.FileHandler(
        os.path.join(output_dir, base_filename.format(str(rank)) + ".log"), "w+")
    fh.setFormatter(fmt)
    logger.addHandler(fh)
    return logging.LoggerAdapter(logger, extra)


class StatsRecorder:
    """Rank-0 training statistics, saved as ./statis/<key>.npy."""

    def __init__(self, base_filename: str, output_dir: str = "./statis"):
        self.base_filename = base_filename
        self.output_dir = output_dir
        self.data = {k: [] for k in FIELDS}

    def append(self, **kwargs) -> None:
        for k, v in kwargs.items():
            self.data[k].append(v)

    def save(self, rank: int = 0) -> str:
        os.makedirs(self.output_dir, exist_ok=True)
        path = os.path.join(self.output_dir,
                            self.base_filename.format(str(rank)) + ".npy")
        np.save(path, self.data)  # pickled dict, same as reference
        return path
