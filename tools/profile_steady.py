"""Steady-state per-kernel breakdown of the flagship step (torch profiler).

rocprofv3 --stats aggregates a whole process including MIOpen's find-mode
tuning sweeps; this scopes to post-warmup steps only.
Usage (GPU box): python tools/profile_steady.py [--channels-last] [--model densenet]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="densenet")
    ap.add_argument("--global-batch", type=int, default=512)
    ap.add_argument("--channels-last", action="store_true", default=True)
    ap.add_argument("--nchw", action="store_true")
    ap.add_argument("--rows", type=int, default=30)
    args = ap.parse_args()

    import bench as B
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.parallel.optim import FlatSGD

    torch.manual_seed(0)
    dev = torch.device("cuda")
    model = B.build(args.model).to(dev)
    if args.channels_last and not args.nchw:
        model = model.to(memory_format=torch.channels_last)
    sync = GradientSynchronizer(model)
    opt = FlatSGD(sync, lr=0.01)
    x = torch.randn(args.global_batch, 3, 32, 32, device=dev)
    if args.channels_last and not args.nchw:
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (args.global_batch,), device=dev)

    def step():
        sync.zero()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        sync.finish()
        opt.step()

    for _ in range(12):  # warmup incl. MIOpen find
        step()
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CUDA]) as prof:
        for _ in range(5):
            step()
        torch.cuda.synchronize()
    print(prof.key_averages().table(
        sort_by="self_cuda_time_total", row_limit=args.rows, max_name_column_width=80))


if __name__ == "__main__":
    main()
