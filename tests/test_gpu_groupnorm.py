"""Fused GroupNorm(+ReLU) kernel vs plain fp32 PyTorch reference."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")

SHAPES = [
    # (N, C, H, W, G, relu) — spans the zoo: Cg=1, Cg=3, big-C, GoogLeNet G=8
    (4, 64, 32, 32, 32, True),
    (4, 32, 32, 32, 32, False),   # Cg = 1
    (2, 24, 16, 16, 8, True),     # Cg = 3 (RegNetX-200MF stage 1)
    (3, 2208, 4, 4, 32, True),    # C > 8*block (DenseNet-161 tail)
    (2, 192, 32, 32, 8, False),   # GoogLeNet stem
    (5, 1024, 8, 8, 32, True),
]


def _native_gn(xb, G, gamma, beta, relu):
    from dynamic_load_balance_distributeddnn_amd.ops import native

    return native.group_norm_act(xb, G, gamma, beta, 1e-5, relu)


@needs_gpu
@pytest.mark.parametrize("shape", SHAPES)
def test_gn_forward_matches_fp32(shape):
    N, C, H, W, G, relu = shape
    torch.manual_seed(0)
    x = torch.randn(N, C, H, W, device="cuda")
    gamma = torch.randn(C, device="cuda") * 0.5 + 1.0
    beta = torch.randn(C, device="cuda") * 0.1
    xb = x.bfloat16()

    ref = F.group_norm(xb.float(), G, gamma, beta, 1e-5)
    if relu:
        ref = F.relu(ref)

    xcl = xb.float().bfloat16().to(memory_format=torch.channels_last)
    out = _native_gn(xcl, G, gamma, beta, relu)
    assert out.is_contiguous(memory_format=torch.channels_last)
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=3e-2)


@needs_gpu
@pytest.mark.parametrize("shape", SHAPES)
def test_gn_backward_matches_fp32(shape):
    N, C, H, W, G, relu = shape
    torch.manual_seed(1)
    x = torch.randn(N, C, H, W, device="cuda").bfloat16()
    gamma = (torch.randn(C, device="cuda") * 0.5 + 1.0)
    beta = torch.randn(C, device="cuda") * 0.1
    dz = torch.randn(N, C, H, W, device="cuda").bfloat16()

    # fp32 reference of the same composed op on the same quantized inputs
    x32 = x.float().requires_grad_()
    g32 = gamma.clone().requires_grad_()
    b32 = beta.clone().requires_grad_()
    ref = F.group_norm(x32, G, g32, b32, 1e-5)
    if relu:
        ref = F.relu(ref)
    ref.backward(dz.float())

    xcl = x.to(memory_format=torch.channels_last).requires_grad_()
    gk = gamma.clone().requires_grad_()
    bk = beta.clone().requires_grad_()
    out = _native_gn(xcl, G, gk, bk, relu)
    out.backward(dz.to(memory_format=torch.channels_last))

    torch.testing.assert_close(xcl.grad.float(), x32.grad,
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(gk.grad, g32.grad, rtol=2e-2, atol=2e-1)
    torch.testing.assert_close(bk.grad, b32.grad, rtol=2e-2, atol=2e-1)


@needs_gpu
def test_gn_autograd_gradcheck_small():
    """End-to-end through a tiny GroupNormAct module in bf16 training mode."""
    from dynamic_load_balance_distributeddnn_amd.ops.layers import \
        GroupNormAct

    torch.manual_seed(2)
    m = GroupNormAct(4, 16, relu=True).cuda()
    x = torch.randn(2, 16, 8, 8, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    y = m(x)
    y.sum().backward()
    assert torch.isfinite(x.grad.float()).all()
    assert m.weight.grad is not None and m.weight.grad.dtype == torch.float32


@needs_gpu
def test_gn_virtual_concat_matches_materialized():
    """Multi-segment GN == GN over torch.cat of the same segments."""
    from dynamic_load_balance_distributeddnn_amd.ops import functional as FD

    torch.manual_seed(5)
    segs32 = [torch.randn(4, c, 16, 16, device="cuda")
              for c in (32, 64, 128, 32)]
    G, C = 32, 256
    gamma = torch.randn(C, device="cuda") * 0.4 + 1
    beta = torch.randn(C, device="cuda") * 0.1
    dz = torch.randn(4, C, 16, 16, device="cuda").bfloat16()

    # materialized reference through the SAME single-tensor kernel
    segs_a = [s.bfloat16().to(memory_format=torch.channels_last)
              .requires_grad_() for s in segs32]
    cat_in = torch.cat(segs_a, 1)
    ref = FD.group_norm_act(cat_in, G, gamma, beta, relu=True)

    segs_b = [s.bfloat16().to(memory_format=torch.channels_last)
              .requires_grad_() for s in segs32]
    out = FD.group_norm_act_cat(segs_b, G, gamma, beta, relu=True)
    torch.testing.assert_close(out.float(), ref.float(), rtol=5e-3, atol=5e-3)

    ref.backward(dz)
    out.backward(dz)
    for a, b in zip(segs_a, segs_b):
        # LDS-atomic accumulation order differs between the two paths ->
        # occasional 1-ulp bf16 differences
        torch.testing.assert_close(a.grad.float(), b.grad.float(),
                                   rtol=5e-3, atol=5e-3)


@needs_gpu
def test_densenet_virtual_concat_trains():
    import torch.nn.functional as F2

    from dynamic_load_balance_distributeddnn_amd.models import DenseNet121

    torch.manual_seed(0)
    m = DenseNet121(10).cuda().to(memory_format=torch.channels_last)
    x = torch.randn(8, 3, 32, 32, device="cuda") \
        .to(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = F2.cross_entropy(m(x), torch.randint(0, 10, (8,), device="cuda"))
    loss.backward()
    assert torch.isfinite(loss)
    assert all(torch.isfinite(p.grad).all() for p in m.parameters())


def test_gn_add_relu_fused_matches_fp32():
    """relu(GN(x) + residual) fused (ResNet/RegNet junction) vs the
    fp32 torch composition, forward and all gradients."""
    import torch.nn.functional as F

    from dynamic_load_balance_distributeddnn_amd.ops import native

    torch.manual_seed(6)
    N, C, H, W = 16, 128, 8, 8
    x = torch.randn(N, C, H, W, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    r = torch.randn(N, C, H, W, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    g = torch.randn(C, device="cuda").requires_grad_()
    b = torch.randn(C, device="cuda").requires_grad_()

    y = native.group_norm_add_act(x, r, 32, g, b, 1e-5)
    dz = torch.randn_like(y)
    y.backward(dz)

    x32 = x.detach().float().requires_grad_()
    r32 = r.detach().float().requires_grad_()
    g32 = g.detach().clone().requires_grad_()
    b32 = b.detach().clone().requires_grad_()
    ref = F.relu(F.group_norm(x32, 32, g32, b32, 1e-5) + r32)
    ref.backward(dz.float())

    def rel(a, bb):
        return (a.float() - bb).norm().item() / max(bb.norm().item(), 1e-9)

    assert rel(y, ref) < 1.5e-2
    assert rel(x.grad, x32.grad) < 2e-2
    assert rel(r.grad, r32.grad) < 1.5e-2
    assert rel(g.grad, g32.grad) < 1.5e-2
    assert rel(b.grad, b32.grad) < 1.5e-2
