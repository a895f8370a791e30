"""GPU (MI355X) tests: kernel numerics vs plain fp32 PyTorch references,
and the flagship training step on device."""

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


@needs_gpu
def test_extension_loaded_and_required():
    from dynamic_load_balance_distributeddnn_amd import ops

    assert ops.available(), "HIP extension must be built in-tree"


@needs_gpu
def test_sgd_kernel_matches_torch_reference():
    from dynamic_load_balance_distributeddnn_amd.ops import ext

    torch.manual_seed(0)
    n = 1_000_003  # odd size exercises the scalar tail
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.randn(n, device="cuda")
    p_ref, m_ref = p.clone(), m.clone()

    ext().sgd_momentum(p, g, m, 0.05, 0.9)

    # plain fp32 torch reference of the same update
    m_ref.mul_(0.9).add_(g)
    p_ref.add_(m_ref, alpha=-0.05)
    # kernel uses FMA contraction (one rounding) where torch's mul_+add_
    # rounds twice -> up to 1 ulp difference, kernel being the more exact
    torch.testing.assert_close(m, m_ref, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(p, p_ref, rtol=1e-6, atol=1e-6)


@needs_gpu
def test_flat_sgd_trains_on_gpu():
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.parallel.optim import FlatSGD

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 1)).cuda()
    sync = GradientSynchronizer(model)
    opt = FlatSGD(sync, lr=0.05)
    x = torch.randn(128, 16, device="cuda")
    y = torch.randn(128, 1, device="cuda")
    losses = []
    for _ in range(50):
        sync.zero()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        sync.finish()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.5


@needs_gpu
@pytest.mark.parametrize("name", ["densenet", "resnet50", "regnet",
                                  "googlenet", "transformer"])
def test_zoo_step_bf16(name):
    import bench as B

    torch.manual_seed(0)
    model = B.build(name).cuda()
    if name == "transformer":
        x = torch.randint(0, 33278, (35, 4), device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
            loss = torch.nn.functional.nll_loss(
                out.reshape(-1, 33278),
                torch.randint(0, 33278, (35 * 4,), device="cuda"))
    else:
        x = torch.randn(8, 3, 32, 32, device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(
                model(x), torch.randint(0, 10, (8,), device="cuda"))
    loss.backward()
    assert torch.isfinite(loss)


@needs_gpu
def test_step_timer_events():
    from dynamic_load_balance_distributeddnn_amd.parallel import StepTimer

    dev = torch.device("cuda")
    t = StepTimer(dev)
    x = torch.randn(4096, 4096, device=dev)
    t.iter_start()
    y = x @ x
    t.backward_done()
    t.comm_done()
    z = y @ x
    t.step_done()
    compute, sync = t.epoch_totals()
    assert compute > 0
    assert sync >= 0
    assert z.isfinite().all()
