import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dynamic_load_balance_distributeddnn_amd.ops import ext

torch.manual_seed(0)
E = ext()

def check(N, HW, seg_widths, Co, G=32, relu=True):
    segs = [torch.randn(N, HW, c, device="cuda").bfloat16() for c in seg_widths]
    C = sum(seg_widths)
    gamma = torch.randn(C, device="cuda").float()
    beta = torch.randn(C, device="cuda").float()
    w = torch.randn(Co, C, 1, 1, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    dy = torch.randn(N, Co, 1, HW, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last)

    # unfused reference: gn_fwd -> conv_fwd / conv_wrw
    y_ref3, mean, rstd = E.gn_fwd(segs, gamma, beta, G, 1e-5, relu)
    # conv via kernel on packed y (4D view)
    y4 = y_ref3.view(N, 1, HW, C).permute(0, 3, 1, 2)
    h_ref = E.conv_fwd(y4, w, None, 1, 0)           # [N, Co, 1, HW]
    dw_ref = E.conv_wrw(y4, dy, 1, 1, 1, 0)         # [Co, C]

    m2, r2 = E.gn_stats(segs, G, 1e-5)
    em = (m2 - mean).abs().max().item(); er = (r2 - rstd).abs().max().item()

    h = E.gn_conv1x1_fwd(segs, m2, r2, gamma, beta, relu, w)  # [N, HW, Co]
    h4 = h.view(N, 1, HW, Co).permute(0, 3, 1, 2)
    eh = (h4.float() - h_ref.float()).abs().max().item()
    hrel = (h4.float() - h_ref.float()).norm().item() / (h_ref.float().norm().item() + 1e-9)

    dw = E.gn_conv1x1_wrw(segs, m2, r2, gamma, beta, relu, dy)
    ew = (dw - dw_ref).abs().max().item()
    wrel = (dw - dw_ref).norm().item() / (dw_ref.norm().item() + 1e-9)
    print(f"N{N} HW{HW} C{C} segs{len(seg_widths)} Co{Co}: stats {em:.2e}/{er:.2e} "
          f"fwd max {eh:.2e} rel {hrel:.2e}  wrw max {ew:.2e} rel {wrel:.2e}")
    return hrel < 2e-2 and wrel < 2e-2

ok = True
ok &= check(16, 1024, [64], 128)
ok &= check(16, 1024, [32]*3 + [64], 128)          # multi-seg
ok &= check(8, 256, [32]*11 + [128], 192)          # 12 segs, C=480
ok &= check(8, 64, [32]*24 + [256], 512)           # block3-ish, C=1024
ok &= check(4, 64, [32]*16 + [512], 96)            # C=1024, small Co
ok &= check(4, 64, [40, 32], 72, G=8)              # C=72 (tail), Co tail
print("ALL OK" if ok else "MISMATCH")
