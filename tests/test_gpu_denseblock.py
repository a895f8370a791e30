"""Manual dense-block backward (kernel-accumulated segment grads) vs the
autograd virtual-concat path — same kernels, so grads must agree to bf16
association-order rounding."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run(block_fn: bool):
    from dynamic_load_balance_distributeddnn_amd.models.densenet import \
        DenseNet

    if block_fn:
        os.environ.pop("DLB_NO_BLOCK_FN", None)
    else:
        os.environ["DLB_NO_BLOCK_FN"] = "1"
    try:
        torch.manual_seed(0)
        model = DenseNet((2, 3), growth=32, num_classes=10).cuda() \
            .to(memory_format=torch.channels_last)
        torch.manual_seed(1)
        x = torch.randn(16, 3, 32, 32, device="cuda") \
            .to(memory_format=torch.channels_last)
        y = torch.randint(0, 10, (16,), device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
        loss = torch.nn.functional.cross_entropy(out.float(), y)
        loss.backward()
        return (loss.item(),
                {n: p.grad.clone() for n, p in model.named_parameters()})
    finally:
        os.environ.pop("DLB_NO_BLOCK_FN", None)


def test_block_fn_matches_autograd_path():
    loss_a, grads_a = _run(block_fn=True)
    loss_b, grads_b = _run(block_fn=False)
    assert abs(loss_a - loss_b) < 1e-3 * max(1.0, abs(loss_b))
    assert set(grads_a) == set(grads_b)
    for name, ga in grads_a.items():
        gb = grads_b[name]
        ga, gb = ga.float().flatten(), gb.float().flatten()
        cos = torch.nn.functional.cosine_similarity(ga, gb, dim=0).item()
        rel = (ga - gb).norm().item() / (gb.norm().item() + 1e-12)
        assert cos > 0.999 and rel < 2e-2, (name, cos, rel)


def test_block_fn_engaged_on_flagship():
    """Guard against a silent fallback: the flagship model on the GPU
    path must route blocks through _DenseBlockFn."""
    import dynamic_load_balance_distributeddnn_amd.ops.denseblock as db
    from dynamic_load_balance_distributeddnn_amd.models.densenet import \
        DenseNet121

    calls = []
    orig = db.dense_block_forward

    def spy(block, transition, segs):
        calls.append(len(block))
        return orig(block, transition, segs)

    db.dense_block_forward = spy
    try:
        model = DenseNet121().cuda().to(memory_format=torch.channels_last)
        x = torch.randn(8, 3, 32, 32, device="cuda") \
            .to(memory_format=torch.channels_last)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            model(x).float().sum().backward()
    finally:
        db.dense_block_forward = orig
    assert calls == [6, 12, 24, 16]
