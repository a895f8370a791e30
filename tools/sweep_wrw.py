import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dynamic_load_balance_distributeddnn_amd.ops import ext

def t(fn, n=15):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

for (Ci,H,Co,R,pad) in [(128,32,32,3,1),(256,32,128,1,0)]:
    x=torch.randn(512,Ci,H,H,device="cuda").bfloat16().to(memory_format=torch.channels_last)
    dy=torch.randn(512,Co,H,H,device="cuda").bfloat16().to(memory_format=torch.channels_last)
    K=R*R*Ci
    for sp in ["1","8","32","64","128","256",""]:
        os.environ["DLB_WRW_SPLITS"]=sp
        if sp=="": os.environ.pop("DLB_WRW_SPLITS")
        us=t(lambda: ext().conv_wrw(x,dy,R,R,1,pad))
        print(f"Ci{Ci} Co{Co} R{R}: splits={sp or 'auto'} -> {us:.0f} us")
    # component timing at auto
    spl_part=torch.zeros(64,Co,K,device="cuda")
    print("  zeros64:", t(lambda: torch.zeros(64,Co,K,device="cuda")), "sum64:", t(lambda: spl_part.sum(0)))
