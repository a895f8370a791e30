import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dynamic_load_balance_distributeddnn_amd.ops import ext
x=torch.randn(512,256,32,32,device="cuda").bfloat16().to(memory_format=torch.channels_last)
dy=torch.randn(512,128,32,32,device="cuda").bfloat16().to(memory_format=torch.channels_last)
for _ in range(10):
    ext().conv_wrw(x,dy,1,1,1,0)
torch.cuda.synchronize()
