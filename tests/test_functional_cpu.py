"""CPU fallback paths of the functional dispatch layer (the `-d true`
debug-mode compositions the GPU kernels mirror)."""

import math

import torch
import torch.nn.functional as F

from dynamic_load_balance_distributeddnn_amd.ops import functional as FD


def test_lm_loss_cpu_matches_criterion():
    torch.manual_seed(0)
    T, d, V = 50, 200, 300
    h = torch.randn(T, d)
    w = torch.randn(V, d) * 0.1
    b = torch.randn(V) * 0.01
    y = torch.randint(0, V, (T,))
    loss = FD.lm_loss(h, w, b, y)
    ref = F.nll_loss(F.log_softmax(F.linear(h, w, b).float(), -1), y)
    assert torch.allclose(loss, ref, rtol=1e-5, atol=1e-6)


def test_embedding_scaled_cpu():
    w = torch.randn(40, 16)
    idx = torch.randint(0, 40, (7, 3))
    out = FD.embedding_scaled(idx, w, math.sqrt(16))
    assert torch.allclose(out, F.embedding(idx, w) * 4.0)


def test_se_mul_cpu():
    x = torch.randn(2, 8, 4, 4)
    g = torch.randn(2, 8, 1, 1)
    assert torch.allclose(FD.se_mul(x, g), x * g.sigmoid())


def test_dropout_cpu_semantics():
    x = torch.ones(10_000)
    assert torch.equal(FD.dropout(x, 0.5, training=False), x)
    y = FD.dropout(x, 0.5, training=True)
    kept = (y != 0).float().mean().item()
    assert 0.45 < kept < 0.55
    assert torch.allclose(y[y != 0], torch.full_like(y[y != 0], 2.0))
