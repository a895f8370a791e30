"""Per-shape conv kernel timing vs the MIOpen path (GPU box tool).

Times fwd / bwd-data / wrw for the zoo's dominant conv shapes.
"""

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F
from dynamic_load_balance_distributeddnn_amd.ops import ext

SHAPES = [  # (name, N, Ci, H, Co, R, stride, pad)
    ("dn-3x3-s1 C128->32 @32", 512, 128, 32, 32, 3, 1, 1),
    ("dn-3x3-s1 C128->32 @16", 512, 128, 16, 32, 3, 1, 1),
    ("dn-1x1 C256->128 @32", 512, 256, 32, 128, 1, 1, 0),
    ("dn-1x1 C512->128 @16", 512, 512, 16, 128, 1, 1, 0),
    ("rn-3x3 C64->64 @32", 512, 64, 32, 64, 3, 1, 1),
    ("rn-1x1 C512->2048 @4", 512, 512, 4, 2048, 1, 1, 0),
]


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    import time
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    for name, N, Ci, H, Co, R, stride, pad in SHAPES:
        x = torch.randn(N, Ci, H, H, device="cuda").bfloat16() \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(Co, Ci, R, R, device="cuda") * 0.05).bfloat16() \
            .to(memory_format=torch.channels_last)
        OH = (H + 2 * pad - R) // stride + 1
        dy = torch.randn(N, Co, OH, OH, device="cuda").bfloat16() \
            .to(memory_format=torch.channels_last)
        t_fwd = timeit(lambda: ext().conv_fwd(x, w, None, stride, pad))
        t_bwd = timeit(lambda: ext().conv_bwd_data(dy, w, H, H, stride, pad))
        t_wrw = timeit(lambda: ext().conv_wrw(x, dy, R, R, stride, pad))

        wf = w.float()
        t_mf = timeit(lambda: F.conv2d(x, w, None, stride=stride, padding=pad))
        t_mw = timeit(lambda: torch.nn.grad.conv2d_weight(
            x, (Co, Ci, R, R), dy, stride=stride, padding=pad))
        t_md = timeit(lambda: torch.nn.grad.conv2d_input(
            (N, Ci, H, H), w, dy, stride=stride, padding=pad))

        print(f"{name:28s} fwd {t_fwd:7.1f} (miopen {t_mf:7.1f}) "
              f"bwd {t_bwd:7.1f} ({t_md:7.1f}) wrw {t_wrw:7.1f} ({t_mw:7.1f})")


if __name__ == "__main__":
    main()
