"""Time gn_bwd (or gn_fwd) for ONE shape — chunking diagnostics.

    python tools/gn_shape.py --hw 64 --c 128 [--fwd] [--iters 50]

DLB_GN_TARGET / DLB_GN_TARGET_BWD control the chunk count.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from dynamic_load_balance_distributeddnn_amd.ops import ext  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=512)
    ap.add_argument("--hw", type=int, default=64)
    ap.add_argument("--c", type=int, default=128)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--fwd", action="store_true")
    args = ap.parse_args()
    E = ext()
    N, HW, C, G = args.n, args.hw, args.c, 32
    x = torch.randn(N, HW, C, device="cuda").bfloat16()
    ga = torch.randn(C, device="cuda").float()
    be = torch.randn(C, device="cuda").float()
    y, mean, rstd = E.gn_fwd([x], ga, be, G, 1e-5, True)
    dz = torch.randn_like(y)

    def run():
        if args.fwd:
            E.gn_fwd([x], ga, be, G, 1e-5, True)
        else:
            E.gn_bwd([x], dz, ga, be, mean, rstd, G, True)

    for _ in range(5):
        run()
    torch.cuda.synchronize()
    s_ev, e_ev = torch.cuda.Event(True), torch.cuda.Event(True)
    s_ev.record()
    for _ in range(args.iters):
        run()
    e_ev.record()
    torch.cuda.synchronize()
    ms = s_ev.elapsed_time(e_ev) / args.iters
    passes = 2.5 if args.fwd else 6  # fwd: x read 2x + y write (0.5 ratio)
    gb = N * HW * C * 2 * passes / 1e9
    print(f"{'fwd' if args.fwd else 'bwd'} N{N} HW{HW} C{C}: {ms*1e3:8.1f} us"
          f"  {gb/ms*1000:6.0f} GB/s")


if __name__ == "__main__":
    main()
