"""Manual Inception backward (ops/googblock.py) vs the per-branch
autograd composition — identical kernels except the 4-way junction sum
(pool-branch grad written fresh, conv grads accumulated in-epilogue vs
three autograd adds), so agreement is tight."""

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


def _mk(seed=0):
    from dynamic_load_balance_distributeddnn_amd.models.googlenet import \
        _Inception
    torch.manual_seed(seed)
    return _Inception(192, 64, 96, 128, 16, 32, 32).cuda() \
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
def test_inception_blockfn_matches_per_branch(monkeypatch):
    from dynamic_load_balance_distributeddnn_amd.ops import googblock

    m = _mk()
    torch.manual_seed(1)
    x0 = torch.randn(16, 192, 16, 16, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    assert googblock.inception_fn_ok(m, x0)

    ya, dxa, ga = _run(m, x0, True, monkeypatch)
    yb, dxb, gb = _run(m, x0, False, monkeypatch)

    assert torch.equal(ya, yb)  # forward is the identical kernel chain
    assert (dxa.float() - dxb.float()).abs().max() <= 3e-2
    for n in ga:
        d = (ga[n].float() - gb[n].float()).abs().max().item()
        s = gb[n].float().abs().max().item() + 1e-6
        assert d <= 2e-3 * s + 1e-4, (n, d, s)


@needs_gpu
def test_inception_direct_arena_grads(monkeypatch):
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    m = _mk(seed=3)
    torch.manual_seed(4)
    x0 = torch.randn(8, 192, 8, 8, device="cuda").bfloat16() \
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
        assert d <= 2e-3 * s + 1e-4, (n, d, s)
