"""Engine integration on GPU: full epochs (train + validate + DBS
bookkeeping) for a CV model and the LM on one device."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


def _args(extra):
    from dynamic_load_balance_distributeddnn_amd.cli import get_parser

    return get_parser().parse_args(extra)


@needs_gpu
def test_cv_epoch_on_gpu(tmp_path, monkeypatch):
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.01")
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = _args(["-d", "false", "-ws", "1", "-b", "64", "-e", "2",
                  "-ds", "cifar10", "-m", "densenet", "-dbs", "true"])
    tr = Trainer(args, 0, 1, torch.device("cuda:0"), logger=None)
    compute, sync, loss0 = tr.train_epoch(0)
    assert compute > 0 and np.isfinite(loss0)
    tr.nodes_time = np.array([compute])
    _, _, loss1 = tr.train_epoch(1)
    val_loss, acc = tr.validate_epoch(1)
    assert np.isfinite(val_loss) and 0 <= acc <= 100
    assert np.isfinite(loss1)


@needs_gpu
def test_lm_epoch_on_gpu(monkeypatch):
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.02")
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = _args(["-d", "false", "-ws", "1", "-b", "32", "-e", "1",
                  "-ds", "wikitext2", "-m", "transformer", "-dbs", "true"])
    tr = Trainer(args, 0, 1, torch.device("cuda:0"), logger=None)
    compute, sync, loss = tr.train_epoch(0)
    assert compute > 0 and np.isfinite(loss)
    val_loss, acc = tr.validate_epoch(0)
    assert np.isfinite(val_loss)


@needs_gpu
def test_training_reduces_loss_gpu():
    """A few dozen steps on one synthetic batch must reduce the loss —
    guards against silently-broken kernel gradients end-to-end."""
    import torch.nn.functional as F

    from dynamic_load_balance_distributeddnn_amd.models import DenseNet121
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.parallel.optim import FlatSGD

    torch.manual_seed(0)
    model = DenseNet121(10).cuda().to(memory_format=torch.channels_last)
    sync = GradientSynchronizer(model)
    opt = FlatSGD(sync, lr=0.01)
    x = torch.randn(64, 3, 32, 32, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (64,), device="cuda")
    losses = []
    for _ in range(60):
        sync.zero()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        sync.finish()
        opt.step()
        losses.append(loss.item())
    # atomics make exact trajectories run-dependent; this asserts clear
    # optimization progress — strict gradient correctness is covered by
    # test_densenet_grads_match_cpu_fp32_reference
    assert min(losses) < losses[0] * 0.85, losses[::10]
    assert all(torch.isfinite(torch.tensor(losses)))


@needs_gpu
def test_densenet_grads_match_cpu_fp32_reference():
    """Model-level numerics: whole DenseNet-121 fwd+bwd on the gfx950
    kernel path vs the CPU fp32 torch path, same weights and data."""
    import torch.nn.functional as F

    from dynamic_load_balance_distributeddnn_amd.models import DenseNet121

    torch.manual_seed(3)
    model_cpu = DenseNet121(10)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))

    loss_cpu = F.cross_entropy(model_cpu(x), y)
    loss_cpu.backward()

    model_gpu = DenseNet121(10)
    model_gpu.load_state_dict(model_cpu.state_dict())
    model_gpu = model_gpu.cuda().to(memory_format=torch.channels_last)
    xg = x.cuda().to(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss_gpu = F.cross_entropy(model_gpu(xg), y.cuda())
    loss_gpu.backward()

    assert abs(loss_gpu.item() - loss_cpu.item()) < 0.05, \
        (loss_gpu.item(), loss_cpu.item())
    # gradient direction must agree; the earliest layers sit under ~120
    # bf16 layers of accumulated rounding, so the bar is depth-dependent
    coses = []
    for (n1, p1), (n2, p2) in zip(model_cpu.named_parameters(),
                                  model_gpu.named_parameters()):
        g1 = p1.grad.flatten().float()
        g2 = p2.grad.cpu().flatten().float()
        cos = torch.nn.functional.cosine_similarity(g1, g2, dim=0).item()
        coses.append((cos, n1))
        assert cos > 0.85, (n1, cos)
    mean_cos = sum(c for c, _ in coses) / len(coses)
    assert mean_cos > 0.97, sorted(coses)[:5]


@needs_gpu
def test_engine_hipgraph_step_matches_eager(tmp_path, monkeypatch):
    """world==1 CV epochs run as hipGraph replays (capture once, replay
    per step) and must produce the same training result as eager: the
    capture snapshots/restores optimizer state, so trajectories align."""
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.01")
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    def run(no_graphs):
        if no_graphs:
            monkeypatch.setenv("DLB_NO_GRAPHS", "1")
        else:
            monkeypatch.delenv("DLB_NO_GRAPHS", raising=False)
        args = _args(["-d", "false", "-ws", "1", "-b", "64", "-e", "1",
                      "-ds", "cifar10", "-m", "densenet"])
        tr = Trainer(args, 0, 1, torch.device("cuda:0"), logger=None)
        _, _, loss = tr.train_epoch(0)
        return tr, loss

    tr_g, loss_g = run(False)
    assert tr_g._graph not in (None, False), "graph capture did not engage"
    tr_e, loss_e = run(True)
    assert abs(loss_g - loss_e) < 2e-2 * max(1.0, abs(loss_e))


@needs_gpu
def test_dbs_interval_mode_on_gpu(monkeypatch):
    """`-dbsi` on GPU: the mid-epoch hipEvent drain + EMA + solver path
    must run (world==1 makes the split trivial but exercises the timer
    mechanics end-to-end on the device timeline)."""
    monkeypatch.setenv("DLB_SYNTH_SCALE", "0.01")
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = _args(["-d", "false", "-ws", "1", "-b", "64", "-e", "1",
                  "-ds", "cifar10", "-m", "densenet", "-dbsi", "3"])
    tr = Trainer(args, 0, 1, torch.device("cuda:0"), logger=None)
    compute, sync, loss = tr.train_epoch(0)
    assert compute > 0 and np.isfinite(loss)
    assert tr._ema_iter_s is not None and tr._ema_iter_s > 0
