"""Manual RegNet block backward (ops/regblock.py) vs the per-layer
autograd composition — same kernels except the junction sum (rides
conv1's data-grad epilogue) and the SE gate math (fp32 matrix products
vs bf16 1x1 convs), so X-blocks agree tightly and Y-blocks to the SE
gate's bf16 rounding."""

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


def _mk(cin, cout, stride, gw, se_ratio, seed=0):
    from dynamic_load_balance_distributeddnn_amd.models.regnet import _Block
    torch.manual_seed(seed)
    return _Block(cin, cout, stride, gw, 1, se_ratio).cuda() \
        .to(memory_format=torch.channels_last)


def _run(m, x0, blockfn, monkeypatch):
    if blockfn:
        monkeypatch.delenv("DLB_NO_BLOCK_FN", raising=False)
    else:
        monkeypatch.setenv("DLB_NO_BLOCK_FN", "1")
    m.zero_grad(set_to_none=True)
    x = x0.clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(x)
    y.float().square().mean().backward()
    return y.detach(), x.grad.clone(), \
        {n: p.grad.clone() for n, p in m.named_parameters()}


@needs_gpu
@pytest.mark.parametrize("cin,cout,stride,gw,se",
                         [(64, 64, 1, 16, 0.0),    # X-block, identity skip
                          (64, 160, 2, 16, 0.0),   # X-block, proj + stride
                          (64, 64, 1, 16, 0.25),   # Y-block with SE
                          (64, 160, 2, 16, 0.25)])
def test_regblock_matches_per_layer(cin, cout, stride, gw, se, monkeypatch):
    from dynamic_load_balance_distributeddnn_amd.ops import regblock

    m = _mk(cin, cout, stride, gw, se)
    torch.manual_seed(1)
    x0 = torch.randn(16, cin, 16, 16, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    assert regblock.regblock_fn_ok(m, x0)

    ya, dxa, ga = _run(m, x0, True, monkeypatch)
    yb, dxb, gb = _run(m, x0, False, monkeypatch)

    ytol = 0.0 if se == 0.0 else 5e-2
    assert (ya.float() - yb.float()).abs().max() <= ytol
    assert (dxa.float() - dxb.float()).abs().max() <= 5e-2
    rtol = 1e-3 if se == 0.0 else 5e-2
    for n in ga:
        d = (ga[n].float() - gb[n].float()).abs().max().item()
        s = gb[n].float().abs().max().item() + 1e-6
        assert d <= rtol * s + 1e-4, (n, d, s)


@needs_gpu
def test_regblock_direct_arena_grads(monkeypatch):
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    m = _mk(64, 64, 1, 16, 0.25, seed=3)
    torch.manual_seed(4)
    x0 = torch.randn(8, 64, 8, 8, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)

    sync = GradientSynchronizer(m)
    sync.zero()
    x = x0.clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(x)
    y.float().square().mean().backward()
    sync.finish()
    ga = {n: p.grad.clone() for n, p in m.named_parameters()}
    sync.detach()

    monkeypatch.setenv("DLB_NO_BLOCK_FN", "1")
    m.zero_grad(set_to_none=True)
    x = x0.clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(x)
    y.float().square().mean().backward()
    gb = {n: p.grad.clone() for n, p in m.named_parameters()}

    for n in ga:
        d = (ga[n].float() - gb[n].float()).abs().max().item()
        s = gb[n].float().abs().max().item() + 1e-6
        assert d <= 3e-2 * s + 1e-4, (n, d, s)
