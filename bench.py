#!/usr/bin/env python3
"""Flagship benchmark: DenseNet-121, global batch 512, bf16, synthetic
CIFAR-10-shape data — images/sec over the whole node (BASELINE.json
headline metric).

Single process:   python bench.py --steps 30 --warmup 10
Multi-GPU (driver): python -m torch.distributed.run --nnodes=1
    --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...

The global batch stays fixed at 512 as GPUs scale (strong scaling, the
reference's own experiment shape: DBS splits ONE global batch across
workers).  Every timed step runs the full training iteration: zero, bf16
forward, loss, backward, weighted bucketed RCCL all-reduce, fused SGD
step.  Rank 0 prints one JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch
import torch.distributed as dist
import torch.nn.functional as F


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=None,
                   help="world size (informational; env WORLD_SIZE wins)")
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="densenet",
                   choices=["densenet", "resnet50", "resnet", "regnetx200",
                            "regnet", "googlenet", "transformer", "mnistnet"])
    p.add_argument("--global-batch", type=int, default=512)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--device", default=None, help="override (e.g. cpu for debug)")
    p.add_argument("--no-graphs", action="store_true",
                   help="disable hipGraph capture of the training step")
    p.add_argument("--engine", action="store_true",
                   help="measure the ENGINE-level step (Trainer epoch: "
                        "dataloader, H2D, DBS bookkeeping, hipEvent "
                        "sensor) instead of the pre-staged kernel-"
                        "throughput step — the trainer's steady state")
    p.add_argument("--no-channels-last", action="store_true",
                   help="disable NHWC layout (NHWC is the GPU default; the "
                        "gfx950 kernels are NHWC-native)")
    return p.parse_args()


def build(model_name, num_classes=10):
    from dynamic_load_balance_distributeddnn_amd import models as M

    table = {
        "densenet": lambda: M.DenseNet121(num_classes),
        "resnet50": lambda: M.ResNet50(num_classes),
        "resnet": lambda: M.ResNet101(num_classes),
        "regnetx200": lambda: M.RegNetX_200MF(num_classes),
        "regnet": lambda: M.RegNetY_400MF(num_classes),
        "googlenet": lambda: M.GoogLeNet(num_classes),
        "mnistnet": lambda: M.MnistNet(),
        "transformer": lambda: M.build_model("transformer"),
    }
    return table[model_name]()


def engine_bench(args, rank, world, device):
    """Trainer-in-the-loop measurement: epochs with the real data path.

    Complements the default pre-staged step (which isolates kernel
    throughput): this includes the synthetic-dataset loader, H2D, the
    per-epoch repartition, and the hipEvent DBS sensor."""
    from dynamic_load_balance_distributeddnn_amd.cli import get_parser
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer
    from dynamic_load_balance_distributeddnn_amd.scheduler import \
        straggler_idle_pct

    # size the synthetic dataset so one epoch is exactly --steps steps
    os.environ["DLB_SYNTH_SCALE"] = str(
        args.steps * args.global_batch / 50_000.0)
    model_flag = {"densenet": "densenet", "resnet": "resnet",
                  "regnet": "regnet", "googlenet": "googlenet",
                  "mnistnet": "mnistnet",
                  "transformer": "transformer"}.get(args.model, "densenet")
    ds = ("wikitext2" if model_flag == "transformer"
          else ("mnist" if model_flag == "mnistnet" else "cifar10"))
    targs = get_parser().parse_args(
        ["-d", "false" if device.type == "cuda" else "true",
         "-ws", str(world), "-b", str(args.global_batch),
         "-e", "2", "-ds", ds, "-m", model_flag])
    tr = Trainer(targs, rank, world, device, logger=None)
    tr.train_epoch(0)  # warmup epoch (hipGraph capture happens here)
    steps = tr.steps_per_epoch
    if world > 1:
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    compute_s, _, _ = tr.train_epoch(1)
    if device.type == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    idle = 0.0
    if world > 1:
        t = torch.tensor([elapsed],
                         device=device if device.type == "cuda" else "cpu")
        allt = torch.empty(world, dtype=t.dtype, device=t.device)
        dist.all_gather_into_tensor(allt, t)
        per = allt.cpu().numpy()
        elapsed = float(per.max())
        idle = straggler_idle_pct(per)
    if rank == 0:
        items = (args.global_batch * 35 if model_flag == "transformer"
                 else args.global_batch)
        print(json.dumps({
            "metric": ("tokens_per_sec" if model_flag == "transformer"
                       else "images_per_sec"),
            "value": round(items * steps / elapsed, 2),
            "unit": ("tokens/s" if model_flag == "transformer"
                     else "images/s"),
            "n_gpus": world, "steps": steps, "warmup": steps,
            "ms_per_step": round(elapsed / steps * 1e3, 3),
            "higher_is_better": True, "scaling": "strong",
            "vs_baseline": None, "dtype": args.dtype, "data": "synthetic",
            "straggler_idle_pct": round(idle, 3), "mode": "engine",
            "config": {"model": model_flag,
                       "global_batch": args.global_batch,
                       "parallelism": f"dbs-dp{world}"},
        }))
    if world > 1:
        dist.destroy_process_group()


def main():
    args = get_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus or 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank % max(1, torch.cuda.device_count())}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        # DLB_BENCH_BACKEND=gloo lets a 1-GPU box exercise the
        # multi-rank bench path (RCCL refuses 2 ranks on one GPU);
        # the driver's real SCALE runs use the default (RCCL).
        backend = os.environ.get(
            "DLB_BENCH_BACKEND",
            "nccl" if device.type == "cuda" else "gloo")
        dist.init_process_group(backend, rank=rank, world_size=world)

    if args.engine:
        return engine_bench(args, rank, world, device)

    from dynamic_load_balance_distributeddnn_amd.models import LM_CONFIG
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.parallel.optim import FlatSGD
    from dynamic_load_balance_distributeddnn_amd.scheduler import DBSScheduler

    torch.manual_seed(1234)
    is_lm = args.model == "transformer"
    model = build(args.model).to(device)
    channels_last = (device.type == "cuda" and not is_lm
                     and not args.no_channels_last)
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    sched = DBSScheduler(world, args.global_batch)
    my_batch = int(sched.batches[rank])

    sync = GradientSynchronizer(model, defer=is_lm)
    sync.set_weight(float(sched.weights[rank]))
    opt = FlatSGD(sync, lr=0.01, momentum=0.9)

    # synthetic per-rank batch, pre-staged on device
    g = torch.Generator().manual_seed(1234 + rank)
    if is_lm:
        bptt = LM_CONFIG["bptt"]
        ntokens = LM_CONFIG["ntokens"]
        x = torch.randint(0, ntokens, (bptt, my_batch), generator=g).to(device)
        y = torch.randint(0, ntokens, (bptt * my_batch,), generator=g).to(device)
        items_per_step = args.global_batch * bptt  # tokens
        metric, unit = "tokens_per_sec", "tokens/s"
    else:
        shape = (1, 28, 28) if args.model == "mnistnet" else (3, 32, 32)
        x = torch.randn(my_batch, *shape, generator=g).to(device)
        y = torch.randint(0, 10, (my_batch,), generator=g).to(device)
        if channels_last:
            x = x.to(memory_format=torch.channels_last)
        items_per_step = args.global_batch
        metric, unit = "images_per_sec", "images/s"

    amp = (torch.autocast("cuda", dtype=torch.bfloat16)
           if args.dtype == "bf16" and device.type == "cuda"
           else torch.autocast("cpu", enabled=False))
    criterion = F.nll_loss if is_lm else F.cross_entropy

    def step():
        sync.zero()
        with amp:
            if is_lm and device.type == "cuda":
                from dynamic_load_balance_distributeddnn_amd.ops import \
                    functional as FD
                h = model.forward_features(x)
                loss = FD.lm_loss(h, model.decoder.weight,
                                  model.decoder.bias, y)
            else:
                out = model(x)
                if is_lm:
                    out = out.reshape(-1, ntokens)
                loss = criterion(out, y)
        loss.backward()
        if is_lm:
            torch.nn.utils.clip_grad_norm_(model.parameters(), 0.25)
        sync.finish()
        opt.step()
        return loss

    model.train()
    for _ in range(args.warmup):
        step()

    # hipGraph capture: the zoo's steps are hundreds of small kernels, so
    # replaying one captured graph removes launch overhead (decisive at
    # small per-rank batches in the strong-scaling sweep).  Collectives
    # and dynamic shapes fall back to eager if capture fails.
    run_step = step
    # capture only single-rank: RCCL collectives inside hipGraph capture
    # are not validated on this stack, and a capture hang on one rank
    # would stall the whole job
    if device.type == "cuda" and world == 1 and not args.no_graphs:
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    step()
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step()
            run_step = graph.replay
        except Exception as e:  # capture unsupported (e.g. RCCL op) -> eager
            if rank == 0:
                print(f"# hipGraph capture unavailable ({type(e).__name__}); "
                      "running eager", flush=True)
            run_step = step
        if rank == 0 and run_step is not step:
            print("# hipGraph capture active", flush=True)

    if world > 1:
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # Gather every rank's elapsed: MAX defines the job (slowest rank),
    # and the spread gives the straggler idle % — the second half of the
    # BASELINE metric (Σ(max−t_i)/(N·max), see scheduler.straggler_idle_pct).
    idle_pct = 0.0
    if world > 1:
        t = torch.tensor([elapsed],
                         device=device if device.type == "cuda" else "cpu")
        allt = torch.empty(world, dtype=t.dtype, device=t.device)
        dist.all_gather_into_tensor(allt, t)
        per_rank = allt.cpu().numpy()
        elapsed = float(per_rank.max())
        from dynamic_load_balance_distributeddnn_amd.scheduler import \
            straggler_idle_pct
        idle_pct = straggler_idle_pct(per_rank)

    if rank == 0:
        value = items_per_step * args.steps / elapsed
        print(json.dumps({
            "metric": metric,
            "value": round(value, 2),
            "unit": unit,
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "straggler_idle_pct": round(idle_pct, 3),
            "config": {
                "model": "DenseNet-121" if args.model == "densenet" else args.model,
                "global_batch": args.global_batch,
                "seq_len": LM_CONFIG["bptt"] if is_lm else None,
                "parallelism": f"dbs-dp{world}",
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
