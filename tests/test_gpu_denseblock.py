"""Manual dense-block backward (kernel-accumulated segment grads) vs the
autograd virtual-concat path.

Both paths run the SAME gfx950 kernels; they differ only in the
association order of each segment's bf16 grad accumulation, so a single
block must agree tightly, while a full 4-block model shows the usual
bf16 depth drift (the same drift the CPU-fp32 comparison in
test_gpu_engine.py measures: stem cosine ~0.92 at DenseNet-121 depth).
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


def _grads(model):
    return {n: p.grad.clone() for n, p in model.named_parameters()}


def _cos(a, b):
    return torch.nn.functional.cosine_similarity(
        a.float().flatten(), b.float().flatten(), dim=0).item()


def test_single_block_matches_per_layer_path():
    """Depth-1 comparison: one block + transition, both paths, strict."""
    from dynamic_load_balance_distributeddnn_amd.models.densenet import \
        _DenseLayer, _Transition
    from dynamic_load_balance_distributeddnn_amd.ops import denseblock as db

    torch.manual_seed(0)
    block = torch.nn.ModuleList(
        [_DenseLayer(64 + 32 * i, 32) for i in range(4)])
    trans = _Transition(192, 96)
    ref = [torch.nn.ModuleList([_DenseLayer(64 + 32 * i, 32)
                                for i in range(4)]), _Transition(192, 96)]
    ref[0].load_state_dict(block.state_dict())
    ref[1].load_state_dict(trans.state_dict())
    block, trans = block.cuda(), trans.cuda()
    ref[0].cuda(), ref[1].cuda()
    for m in (*block, trans, *ref[0], ref[1]):
        m.to(memory_format=torch.channels_last)

    torch.manual_seed(1)
    x0 = torch.randn(16, 64, 16, 16, device="cuda") \
        .to(memory_format=torch.channels_last).to(torch.bfloat16)
    xa = x0.clone().requires_grad_(True)
    xb = x0.clone().requires_grad_(True)

    assert db.block_fn_ok(block, [xa])
    out_a = db.dense_block_forward(block, trans, [xa])[0]
    out_a.float().square().mean().backward()

    segs = [xb]
    for layer in ref[0]:
        segs.insert(0, layer(segs))
    out_b = ref[1](segs)
    out_b.float().square().mean().backward()

    # the GN stats merge uses LDS fp32 atomics, so even the per-layer
    # path is not bit-deterministic run to run; tolerances below are 3x
    # the measured run-to-run noise of the per-layer path itself
    # (tools/debug_blockfn.py: out rel 1.2e-3, dx rel 1.4e-2, params <2e-3)
    orel = (out_a.float() - out_b.float()).norm().item() \
        / (out_b.float().norm().item() + 1e-12)
    assert orel < 5e-3, orel
    assert _cos(xa.grad, xb.grad) > 0.999
    rel = (xa.grad.float() - xb.grad.float()).norm().item() \
        / (xb.grad.float().norm().item() + 1e-12)
    assert rel < 5e-2, rel
    pa = dict(list(block.named_parameters()) + list(trans.named_parameters()))
    pb = dict(list(ref[0].named_parameters()) + list(ref[1].named_parameters()))
    for name in pa:
        c = _cos(pa[name].grad, pb[name].grad)
        assert c > 0.998, (name, c)


def _run_model(block_fn: bool):
    from dynamic_load_balance_distributeddnn_amd.models.densenet import \
        DenseNet

    if block_fn:
        os.environ.pop("DLB_NO_BLOCK_FN", None)
    else:
        os.environ["DLB_NO_BLOCK_FN"] = "1"
    try:
        torch.manual_seed(0)
        # depths chosen so every width stays divisible by the 32 GN
        # groups: 64->128->T64 / ->192->T96 / ->192->T96 / ->160 final
        model = DenseNet((2, 4, 3, 2), growth=32, num_classes=10).cuda() \
            .to(memory_format=torch.channels_last)
        torch.manual_seed(1)
        x = torch.randn(16, 3, 32, 32, device="cuda") \
            .to(memory_format=torch.channels_last)
        y = torch.randint(0, 10, (16,), device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
        loss = torch.nn.functional.cross_entropy(out.float(), y)
        loss.backward()
        return loss.item(), _grads(model)
    finally:
        os.environ.pop("DLB_NO_BLOCK_FN", None)


def test_model_grads_match_with_depth_drift():
    loss_a, grads_a = _run_model(block_fn=True)
    loss_b, grads_b = _run_model(block_fn=False)
    # forward is kernel-identical
    assert abs(loss_a - loss_b) < 1e-3 * max(1.0, abs(loss_b))
    assert set(grads_a) == set(grads_b)
    # bf16 reassociation drift grows toward the input; bound per depth
    def floor_for(name):
        if name.startswith(("head", "final_norm", "blocks.3")):
            return 0.99
        if name.startswith(("blocks.2", "transitions.2")):
            return 0.98
        if name.startswith(("blocks.1", "transitions.1")):
            return 0.96
        return 0.92  # stem / blocks.0 / transitions.0
    for name, ga in grads_a.items():
        c = _cos(ga, grads_b[name])
        assert c > floor_for(name), (name, c)


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


@needs_gpu
def test_direct_arena_grads_match_module_path():
    """FlatSGD direct mode (arena grad views + DEFERRED batched
    dgamma/dbeta reduction) must produce the same gradients as the
    plain autograd/module path."""
    import torch.nn.functional as F

    from dynamic_load_balance_distributeddnn_amd.models import DenseNet121
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.parallel.optim import FlatSGD

    torch.manual_seed(4)
    x = torch.randn(16, 3, 32, 32, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (16,), device="cuda")

    def grads(direct):
        torch.manual_seed(11)
        model = DenseNet121(10).cuda().to(memory_format=torch.channels_last)
        sync = None
        if direct:
            sync = GradientSynchronizer(model)
            opt = FlatSGD(sync, lr=0.0)
            sync.zero()
        else:
            for p in model.parameters():
                p.grad = None
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        out = [p.grad.detach().float().clone()
               for p in model.parameters()], loss.item()
        if sync is not None:
            sync.detach()  # break the param->sink reference cycle
        return out

    gd, ld = grads(True)
    gm, lm = grads(False)
    assert abs(ld - lm) < 1e-2 * max(1.0, abs(lm))
    bad = 0
    for a, b in zip(gd, gm):
        ref = b.norm().item()
        if ref < 1e-8:
            continue
        if (a - b).norm().item() / ref > 2e-2:
            bad += 1
    assert bad == 0
