"""Fused GN->1x1-conv kernels vs the unfused composition, plus the
memory reduction the fusion exists for."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from dynamic_load_balance_distributeddnn_amd.ops import ext
    return ext()


@pytest.mark.parametrize("N,HW,widths,Co", [
    (16, 1024, [64], 128),
    (16, 1024, [32] * 3 + [64], 128),       # multi-segment stream
    (8, 64, [32] * 24 + [256], 512),        # block3 shape, 25 segments
    (8, 16, [32] * 4 + [128], 128),         # last-block spatial (4x4)
])
def test_fused_matches_unfused(N, HW, widths, Co):
    E = _ext()
    torch.manual_seed(0)
    segs = [torch.randn(N, HW, c, device="cuda").bfloat16() for c in widths]
    C = sum(widths)
    gamma = torch.randn(C, device="cuda").float()
    beta = torch.randn(C, device="cuda").float()
    w = torch.randn(Co, C, 1, 1, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    dy = torch.randn(N, Co, 1, HW, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last)

    y3, mean, rstd = E.gn_fwd(segs, gamma, beta, 32, 1e-5, True)
    y4 = y3.view(N, 1, HW, C).permute(0, 3, 1, 2)
    h_ref = E.conv_fwd(y4, w, None, 1, 0)
    dw_ref = E.conv_wrw(y4, dy, 1, 1, 1, 0)

    m2, r2 = E.gn_stats(segs, 32, 1e-5)
    assert (m2 - mean).abs().max().item() < 1e-5
    assert (r2 - rstd).abs().max().item() < 1e-4

    h = E.gn_conv1x1_fwd(segs, m2, r2, gamma, beta, True, w)
    h4 = h.view(N, 1, HW, Co).permute(0, 3, 1, 2).float()
    rel = (h4 - h_ref.float()).norm().item() / (h_ref.float().norm().item() + 1e-9)
    assert rel < 1e-3, rel

    dw = E.gn_conv1x1_wrw(segs, m2, r2, gamma, beta, True, dy)
    relw = (dw - dw_ref).norm().item() / (dw_ref.norm().item() + 1e-9)
    assert relw < 1e-3, relw

    # out= destination variant (the direct-arena grad path)
    dst = torch.zeros(Co * C, device="cuda").view(Co, C)
    E.gn_conv1x1_wrw(segs, m2, r2, gamma, beta, True, dy, out=dst)
    assert torch.allclose(dst, dw)


def test_fusion_cuts_peak_memory():
    """The fused path never materializes/saves the normalized stream;
    measured 3.9 vs 7.0 GB at batch 512 — assert a healthy margin at a
    test-sized batch."""
    import os
    from dynamic_load_balance_distributeddnn_amd.models.densenet import \
        DenseNet121

    def peak(disable):
        if disable:
            os.environ["DLB_NO_BLOCK_FN"] = "1"
        else:
            os.environ.pop("DLB_NO_BLOCK_FN", None)
        try:
            import gc
            gc.collect()  # free other tests' leftovers (arena cycles)
            torch.cuda.empty_cache()
            torch.cuda.reset_peak_memory_stats()
            base = torch.cuda.memory_allocated()  # lingering tensors
            torch.manual_seed(0)
            model = DenseNet121().cuda().to(memory_format=torch.channels_last)
            x = torch.randn(64, 3, 32, 32, device="cuda") \
                .to(memory_format=torch.channels_last)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                model(x).float().sum().backward()
            return torch.cuda.max_memory_allocated() - base
        finally:
            os.environ.pop("DLB_NO_BLOCK_FN", None)
    fused, unfused = peak(False), peak(True)
    assert fused < 0.75 * unfused, (fused, unfused)
