import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dynamic_load_balance_distributeddnn_amd.models.densenet import _DenseLayer, _Transition
from dynamic_load_balance_distributeddnn_amd.ops import denseblock as db

def build():
    torch.manual_seed(0)
    block = torch.nn.ModuleList([_DenseLayer(64+32*i, 32) for i in range(4)]).cuda()
    trans = _Transition(192, 96).cuda()
    for m in (*block, trans): m.to(memory_format=torch.channels_last)
    return block, trans

block, trans = build()
sd_b, sd_t = block.state_dict(), trans.state_dict()
torch.manual_seed(1)
x0 = torch.randn(16,64,16,16, device="cuda").to(memory_format=torch.channels_last).to(torch.bfloat16)

def run(use_fn):
    blk, tr = build()
    blk.load_state_dict(sd_b); tr.load_state_dict(sd_t)
    x = x0.clone().requires_grad_(True)
    if use_fn:
        out = db.dense_block_forward(blk, tr, [x])[0]
    else:
        segs = [x]
        for layer in blk: segs.insert(0, layer(segs))
        out = tr(segs)
    out.float().square().mean().backward()
    g = {n: p.grad.clone() for n,p in list(blk.named_parameters())+list(tr.named_parameters())}
    return out.detach(), x.grad.clone(), g

oa, dxa, ga = run(True)
ob, dxb, gb = run(False)
ob2, dxb2, gb2 = run(False)  # determinism probe
def cmp(tag, a, b):
    a, b = a.float().flatten(), b.float().flatten()
    cos = torch.nn.functional.cosine_similarity(a,b,dim=0).item()
    rel = (a-b).norm().item()/(b.norm().item()+1e-12)
    mx = (a-b).abs().max().item()
    print(f"{tag:45s} cos={cos:.6f} rel={rel:.4e} maxabs={mx:.4e}")
print("== determinism (per-layer path twice) ==")
cmp("out", ob, ob2); cmp("dx", dxb, dxb2)
worst=sorted(gb, key=lambda n: -(gb[n].float()-gb2[n].float()).norm().item()/(gb2[n].float().norm().item()+1e-12))[:3]
for n in worst: cmp(n, gb[n], gb2[n])
print("== blockfn vs per-layer ==")
cmp("out", oa, ob); cmp("dx", dxa, dxb)
for n in ga: cmp(n, ga[n], gb[n])
