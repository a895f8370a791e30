"""Model zoo: parameter parity with the reference and forward shapes."""

import pytest
import torch

from dynamic_load_balance_distributeddnn_amd.models import (
    LM_CONFIG, DenseNet121, GoogLeNet, MnistNet, RegNetX_200MF,
    RegNetY_400MF, ResNet18, ResNet50, ResNet101, TransformerModel,
    build_model)


def nparams(m):
    return sum(p.numel() for p in m.parameters())


# Reference param counts measured in SURVEY.md §2.3
@pytest.mark.parametrize("ctor,expected", [
    (lambda: MnistNet(), 21_840),
    (lambda: ResNet101(10), 42_512_970),
    (lambda: DenseNet121(10), 6_956_298),
    (lambda: RegNetY_400MF(10), 5_714_362),
    (lambda: TransformerModel(33278, 200, 2, 200, 2, 0.2), 13_828_478),
])
def test_param_parity(ctor, expected):
    assert nparams(ctor()) == expected


@pytest.mark.parametrize("ctor", [
    lambda: ResNet18(10), lambda: ResNet50(10), lambda: DenseNet121(10),
    lambda: GoogLeNet(10), lambda: RegNetX_200MF(10),
    lambda: RegNetY_400MF(10),
])
def test_cv_forward_backward(ctor):
    m = ctor()
    x = torch.randn(2, 3, 32, 32)
    out = m(x)
    assert out.shape == (2, 10)
    out.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_cifar100_heads():
    assert build_model("densenet", 100)(torch.randn(2, 3, 32, 32)).shape == (2, 100)


def test_mnistnet_forward():
    m = MnistNet()
    out = m(torch.randn(3, 1, 28, 28))
    assert out.shape == (3, 10)
    # log_softmax output sums to ~1 in prob space
    assert torch.allclose(out.exp().sum(1), torch.ones(3), atol=1e-5)


def test_transformer_forward_and_causality():
    c = LM_CONFIG
    m = TransformerModel(c["ntokens"], c["emsize"], c["nhead"], c["nhid"],
                         c["nlayers"], 0.0)
    m.eval()
    src = torch.randint(0, c["ntokens"], (10, 3))
    out = m(src)
    assert out.shape == (10, 3, c["ntokens"])
    # causality: token t's output must not depend on tokens > t
    src2 = src.clone()
    src2[7:] = (src2[7:] + 1) % c["ntokens"]
    out2 = m(src2)
    assert torch.allclose(out[:7], out2[:7], atol=1e-5)
    assert not torch.allclose(out[7:], out2[7:], atol=1e-5)


def test_googlenet_b3_order_fixed():
    """The reference's 5x5-reduce branch crashes (GN before conv,
    Net/GoogleNet.py:29-30); ours must run."""
    m = GoogLeNet(10)
    out = m(torch.randn(1, 3, 32, 32))
    assert out.shape == (1, 10)
