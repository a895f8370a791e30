"""Standalone GroupNorm-backward microbench over the DenseNet-121 bs512
norm population (the top round-2 optimization target — ROADMAP item 1).

Usage (on a GPU box):
    python tools/gn_bwd_bench.py [--iters 50] [--accumulate]

Prints per-shape kernel time and effective HBM throughput assuming the
kernel's 6 logical tensor passes (x and dz twice each, dx write, dx read
in accumulate mode), so variants can be compared against the ~8 TB/s
roof.  Env knobs DLB_GN_TARGET_BWD etc. apply.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from dynamic_load_balance_distributeddnn_amd.ops import ext  # noqa: E402

# (N, HW, [segment widths]) — every norm1/norm2/transition/final GN of
# DenseNet-121 at global batch 512, grouped by distinct shape with its
# per-step call count.
def dn121_norm_population(N=512):
    pop = []
    ch, hw = 64, 1024
    for bi, depth in enumerate([6, 12, 24, 16]):
        for k in range(depth):
            segs = [32] * k + [ch]
            pop.append((N, hw, segs, 1))          # norm1 (virtual concat)
            pop.append((N, hw, [128], 1))         # norm2
        cin = ch + 32 * depth
        if bi < 3:
            pop.append((N, hw, [32] * depth + [ch], 1))  # transition norm
            ch, hw = cin // 2, hw // 4
    pop.append((N, hw, [32] * 16 + [ch], 1))      # final norm
    return pop


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--accumulate", action="store_true")
    args = ap.parse_args()
    E = ext()
    G, eps = 32, 1e-5
    tot_ms = 0.0
    tot_gb = 0.0
    for (N, HW, widths, calls) in dn121_norm_population(args.batch):
        segs = [torch.randn(N, HW, c, device="cuda").bfloat16()
                for c in widths]
        C = sum(widths)
        ga = torch.randn(C, device="cuda").float()
        be = torch.randn(C, device="cuda").float()
        y, mean, rstd = E.gn_fwd(segs, ga, be, G, eps, True)
        dz = torch.randn_like(y)
        acc = [torch.zeros_like(s) for s in segs] if args.accumulate else None
        kw = dict(dx_accum=acc) if acc else {}
        for _ in range(3):
            E.gn_bwd(segs, dz, ga, be, mean, rstd, G, True, **kw)
        torch.cuda.synchronize()
        s_ev, e_ev = torch.cuda.Event(True), torch.cuda.Event(True)
        s_ev.record()
        for _ in range(args.iters):
            E.gn_bwd(segs, dz, ga, be, mean, rstd, G, True, **kw)
        e_ev.record()
        torch.cuda.synchronize()
        ms = s_ev.elapsed_time(e_ev) / args.iters
        passes = 7 if args.accumulate else 6
        gb = N * HW * C * 2 * passes / 1e9
        tot_ms += ms * calls
        tot_gb += gb * calls
        print(f"N{N} HW{HW:5d} C{C:4d} segs{len(widths):3d}: "
              f"{ms*1e3:8.1f} us  {gb/ms*1000:6.0f} GB/s")
    print(f"\npopulation total: {tot_ms:.2f} ms/step, "
          f"avg {tot_gb/tot_ms*1000:.0f} GB/s (roof ~8000)")


if __name__ == "__main__":
    main()
