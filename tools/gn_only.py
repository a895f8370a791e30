import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dynamic_load_balance_distributeddnn_amd.ops import ext
x = torch.randn(512, 1024, 128, device="cuda").bfloat16()   # [N,HW,C]
dz = torch.randn_like(x)
g = torch.ones(128, device="cuda"); b = torch.zeros(128, device="cuda")
y, mean, rstd = ext().gn_fwd(x, g, b, 32, 1e-5, True)
for _ in range(20):
    ext().gn_bwd(x, dz, g, b, mean, rstd, 32, True)
torch.cuda.synchronize()
