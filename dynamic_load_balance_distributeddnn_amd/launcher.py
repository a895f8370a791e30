"""Process launcher + distributed runtime bring-up.

Reference layout (dbs.py:511-544): fork ``world_size`` local processes,
rendezvous over env vars at 127.0.0.1, bind each rank to a GPU from the
``-gpu`` map (several ranks may share one GPU — that oversubscription is
the reference's straggler simulator and is preserved).

MI355X changes: the process group backend is ``nccl`` (= RCCL over xGMI)
whenever the run is on GPUs; ``gloo`` only in `-d true` CPU debug mode
(the reference used gloo even for GPU runs — dbs.py:511).
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from .cli import base_filename
from .engine import Trainer
from .utils import init_logger

__all__ = ["launch", "worker_entry", "resolve_device", "init_process_group"]


def resolve_device(args, rank: int) -> torch.device:
    if args.debug or not torch.cuda.is_available():
        return torch.device("cpu")
    gpu = args.gpu
    if isinstance(gpu, list):
        idx = gpu[rank % len(gpu)]
    else:
        idx = int(gpu)
    return torch.device(f"cuda:{idx}")


def init_process_group(args, rank: int, world_size: int,
                       master_port: str | None = None) -> torch.device:
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    if master_port is not None:
        os.environ["MASTER_PORT"] = str(master_port)
    else:
        os.environ.setdefault("MASTER_PORT", "29500")
    device = resolve_device(args, rank)
    # RCCL requires one rank per GPU; the reference's straggler trick maps
    # several ranks onto one GPU (-gpu 0,0,0,1 — README.md:23-28), which
    # must then rendezvous over gloo (the reference's only backend).
    if isinstance(args.gpu, list):
        distinct = len({args.gpu[r % len(args.gpu)] for r in range(world_size)})
    else:
        distinct = 1  # '-gpu 0' with ws>1: every rank on one GPU
    oversubscribed = world_size > 1 and distinct < world_size
    backend = "gloo" if (device.type == "cpu" or oversubscribed) else "nccl"
    if device.type == "cuda":
        torch.cuda.set_device(device)
    dist.init_process_group(backend, rank=rank, world_size=world_size)
    return device


def worker_entry(rank: int, args, master_port: str | None = None) -> None:
    device = init_process_group(args, rank, args.world_size, master_port)
    name = base_filename(args)
    logger = init_logger(args, rank, name)
    logger.info(f"Rank {rank}/{args.world_size} up on {device} "
                f"(backend {dist.get_backend()})")
    try:
        trainer = Trainer(args, rank, args.world_size, device, logger)
        trainer.run(name)
    finally:
        dist.destroy_process_group()


def launch(args, master_port: str | None = None) -> None:
    """Spawn world_size worker processes and join them."""
    if args.world_size == 1:
        worker_entry(0, args, master_port)
        return
    mp.start_processes(
        worker_entry,
        args=(args, master_port),
        nprocs=args.world_size,
        start_method="spawn",
        join=True,
    )
