"""Grouped 3x3 conv kernels (RegNet K3) vs fp32 torch reference."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")

# (N, C, H, GW, stride) — RegNetX-200MF/400MF shapes
SHAPES = [(8, 24, 32, 8, 1), (8, 56, 32, 8, 1), (8, 152, 16, 8, 2),
          (8, 64, 32, 16, 1), (8, 160, 16, 16, 2)]


@needs_gpu
@pytest.mark.parametrize("shape", SHAPES)
def test_grouped_conv_fwd_bwd(shape):
    from dynamic_load_balance_distributeddnn_amd.ops import native

    N, C, H, GW, stride = shape
    groups = C // GW
    torch.manual_seed(0)
    x = torch.randn(N, C, H, H, device="cuda").bfloat16()
    w = (torch.randn(C, GW, 3, 3, device="cuda") / (3 * GW ** 0.5)).float()
    OH = (H + 2 - 3) // stride + 1
    dz = torch.randn(N, C, OH, OH, device="cuda").bfloat16()

    x32 = x.float().requires_grad_()
    w32 = w.bfloat16().float().requires_grad_()
    ref = F.conv2d(x32, w32, None, stride=stride, padding=1, groups=groups)
    ref.backward(dz.float())

    xn = x.to(memory_format=torch.channels_last).requires_grad_()
    wn = w.clone().requires_grad_()
    y = native.grouped_conv2d(xn, wn, stride)
    torch.testing.assert_close(y.float(), ref.detach(), rtol=3e-2, atol=1e-1)
    y.backward(dz.to(memory_format=torch.channels_last))
    torch.testing.assert_close(xn.grad.float(), x32.grad, rtol=5e-2, atol=2e-1)
    torch.testing.assert_close(wn.grad.float(), w32.grad, rtol=5e-2, atol=5e-1)


@needs_gpu
def test_regnet_step_native():
    import bench as B

    torch.manual_seed(0)
    model = B.build("regnetx200").cuda() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(16, 3, 32, 32, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (16,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = F.cross_entropy(model(x), y)
    loss.backward()
    assert torch.isfinite(loss)
