"""K10/K11/K15 elementwise kernels vs plain fp32 torch references."""

import math

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def _rel(a, b):
    return (a.float() - b.float()).norm().item() / max(b.norm().item(), 1e-9)


# ------------------------------------------------------------- K15 dropout
def test_dropout_mask_statistics_and_backward():
    from dynamic_load_balance_distributeddnn_amd.ops import native

    torch.manual_seed(0)
    x = torch.ones(1 << 16, device="cuda").bfloat16().requires_grad_()
    torch.manual_seed(5)
    y = native.dropout(x, 0.3)
    kept = (y != 0)
    frac = kept.float().mean().item()
    assert 0.67 < frac < 0.73
    # survivors scaled by 1/(1-p)
    assert torch.allclose(y[kept].float(),
                          torch.full_like(y[kept].float(), 1 / 0.7),
                          rtol=1e-2)
    y.sum().backward()
    # backward re-applies the same mask
    assert torch.equal((x.grad != 0), kept)


def test_dropout_deterministic_under_seed():
    from dynamic_load_balance_distributeddnn_amd.ops import native

    x = torch.randn(4096, device="cuda").bfloat16()
    torch.manual_seed(3)
    a = native.dropout(x, 0.5)
    torch.manual_seed(3)
    b = native.dropout(x, 0.5)
    assert torch.equal(a, b)


# ----------------------------------------------------------- K11 embedding
def test_embedding_scaled_fwd_bwd():
    from dynamic_load_balance_distributeddnn_amd.ops import native

    torch.manual_seed(1)
    V, d, T = 1000, 200, 700
    w = torch.randn(V, d, device="cuda").requires_grad_()
    idx = torch.randint(0, V, (35, 20), device="cuda")
    scale = math.sqrt(d)

    out = native.embedding_scaled(idx, w, scale)
    assert out.shape == (35, 20, d) and out.dtype == torch.bfloat16
    g = torch.randn_like(out)
    out.backward(g)

    w32 = w.detach().to(torch.bfloat16).float().requires_grad_()
    ref = F.embedding(idx, w32) * scale
    ref.backward(g.float())
    assert _rel(out, ref) < 1e-2
    assert _rel(w.grad, w32.grad) < 1e-2


# -------------------------------------------------------------- K10 SE mul
def test_se_mul_fwd_bwd():
    from dynamic_load_balance_distributeddnn_amd.ops import native

    torch.manual_seed(2)
    N, C, H, W = 8, 64, 16, 16
    x = torch.randn(N, C, H, W, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    gate = torch.randn(N, C, 1, 1, device="cuda").bfloat16() \
        .requires_grad_()
    y = native.se_mul(x, gate)
    g = torch.randn_like(y)
    y.backward(g)

    x32 = x.detach().float().requires_grad_()
    g32 = gate.detach().float().requires_grad_()
    ref = x32 * g32.sigmoid()
    ref.backward(g.float())
    assert _rel(y, ref) < 1e-2
    assert _rel(x.grad, x32.grad) < 1.5e-2
    assert _rel(gate.grad, g32.grad) < 1.5e-2
