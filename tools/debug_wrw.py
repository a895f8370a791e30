import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from dynamic_load_balance_distributeddnn_amd.ops import ext

torch.manual_seed(0)
for (N, Ci, H, Co, R, stride, pad) in [(2, 8, 8, 32, 3, 1, 1),
                                       (8, 128, 32, 32, 3, 1, 1),
                                       (2, 8, 8, 32, 3, 1, 0),
                                       (2, 8, 8, 32, 1, 1, 0)]:
    x = torch.randn(N, Ci, H, H, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    OH = (H + 2 * pad - R) // stride + 1
    dy = torch.randn(N, Co, OH, OH, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    ref = torch.nn.grad.conv2d_weight(
        x.float(), (Co, Ci, R, R), dy.float(), stride=stride, padding=pad)
    dwf = ext().conv_wrw(x, dy, R, R, stride, pad)
    got = dwf.view(Co, R, R, Ci).permute(0, 3, 1, 2)
    diff = (got - ref).abs()
    print(f"shape N{N} Ci{Ci} H{H} Co{Co} R{R} s{stride} p{pad}: "
          f"maxdiff={diff.max().item():.4f} ref_std={ref.std().item():.3f}")
    if diff.max() > 0.5:
        # where is it wrong? aggregate error by (r,s), by co, by ci
        print("  err by (r,s):", diff.sum(dim=(0, 1)).cpu().numpy().round(1))
        print("  err by co[:8]:", diff.sum(dim=(1, 2, 3))[:8].cpu().numpy().round(1))
        print("  err by ci[:8]:", diff.sum(dim=(0, 2, 3))[:8].cpu().numpy().round(1))
