import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn.functional as F
from dynamic_load_balance_distributeddnn_amd.ops import ext
def p(*a): print(*a, flush=True)

torch.manual_seed(2)
S, B, E, H = 35, 8, 200, 2
p("alloc...")
# contiguous separate tensors first
q = torch.randn(S, B, E, device="cuda").bfloat16().contiguous()
k = torch.randn(S, B, E, device="cuda").bfloat16().contiguous()
v = torch.randn(S, B, E, device="cuda").bfloat16().contiguous()
torch.cuda.synchronize(); p("inputs ready; strides", q.stride())
o, ps = ext().attn_fwd(q, k, v, H)
torch.cuda.synchronize(); p("contig fwd OK")

qkv = torch.randn(S, B, 3*E, device="cuda").bfloat16()
qc, kc, vc = qkv.chunk(3, -1)
torch.cuda.synchronize(); p("chunk strides", qc.stride(), qc.data_ptr()-qkv.data_ptr())
o2, ps2 = ext().attn_fwd(qc, kc, vc, H)
torch.cuda.synchronize(); p("chunk fwd OK")

ref_q, ref_k, ref_v = qkv.float().chunk(3, -1)
d = E//H
def split(t): return t.reshape(S, B*H, d).transpose(0,1)
ref = F.scaled_dot_product_attention(split(ref_q), split(ref_k), split(ref_v), is_causal=True)
ref = ref.transpose(0,1).reshape(S,B,E)
diff = (o2.float()-ref).abs()
p("chunk max diff", diff.max().item())

p("--- autograd path ---")
from dynamic_load_balance_distributeddnn_amd.ops import functional as FD
qkv2 = torch.randn(S, B, 3*E, device="cuda").bfloat16().requires_grad_()
qa, ka, va = qkv2.chunk(3, -1)
out = FD.causal_attention(qa, ka, va, H)
torch.cuda.synchronize(); p("apply fwd OK")
dz = torch.randn_like(out)
out.backward(dz)
torch.cuda.synchronize(); p("apply bwd OK; grad finite:", torch.isfinite(qkv2.grad).all().item())
