"""Block-Function gating: on CPU (or any non-envelope input) the ResNet
and RegNet block forwards must take the per-layer fallback, and the
envelope predicates must reject non-CUDA/non-bf16 inputs — the manual
backward (ops/resblock.py, ops/regblock.py) is a GPU-only path."""

import torch


def test_bottleneck_gate_rejects_cpu():
    from dynamic_load_balance_distributeddnn_amd.models.resnet import \
        _Bottleneck
    from dynamic_load_balance_distributeddnn_amd.ops import resblock

    m = _Bottleneck(64, 32, 1)
    x = torch.randn(2, 64, 8, 8)
    assert not resblock.bottleneck_fn_ok(m, x)
    y = m(x)  # falls back to the per-layer composition
    y.square().mean().backward()
    assert m.a[0].weight.grad is not None
    assert m.norm_out.weight.grad is not None


def test_regblock_gate_rejects_cpu():
    from dynamic_load_balance_distributeddnn_amd.models.regnet import _Block
    from dynamic_load_balance_distributeddnn_amd.ops import regblock

    m = _Block(64, 64, 1, 16, 1, 0.25)
    x = torch.randn(2, 64, 8, 8)
    assert not regblock.regblock_fn_ok(m, x)
    y = m(x)
    y.square().mean().backward()
    assert m.a[2].weight.grad is not None          # grouped conv
    assert m.se.reduce.weight.grad is not None     # SE path
    assert m.norm_out.weight.grad is not None


def test_blockfn_grads_match_reference_composition_cpu():
    """The CPU fallback must equal a hand-built torch composition of the
    same block (reference Net/Resnet.py:30-55 semantics)."""
    import torch.nn.functional as F
    from dynamic_load_balance_distributeddnn_amd.models.resnet import \
        _Bottleneck

    torch.manual_seed(0)
    m = _Bottleneck(32, 32, 1)
    x = torch.randn(2, 32, 8, 8, requires_grad=True)
    y = m(x)

    c1, n1, c2, n2, c3 = m.a
    h = F.relu(F.group_norm(F.conv2d(x, c1.weight), n1.num_groups,
                            n1.weight, n1.bias, n1.eps))
    h = F.relu(F.group_norm(F.conv2d(h, c2.weight, padding=1),
                            n2.num_groups, n2.weight, n2.bias, n2.eps))
    h = F.conv2d(h, c3.weight)
    res = m.proj(x) if m.proj is not None else x
    ref = F.relu(F.group_norm(h, m.norm_out.num_groups,
                              m.norm_out.weight, m.norm_out.bias,
                              m.norm_out.eps) + res)
    assert torch.allclose(y, ref, rtol=1e-5, atol=1e-6)
