"""Multi-process CPU/gloo tests: weighted grad sync, time exchange,
and the 2-worker MnistNet DBS integration run (BASELINE config #1)."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dynamic_load_balance_distributeddnn_amd.cli import get_parser


def _dist_worker(rank, world, port, fn, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = fn(rank, world)
        torch.save(result, os.path.join(outdir, f"rank{rank}.pt"))
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world, port):
    import tempfile

    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as outdir:
        procs = [ctx.Process(target=_dist_worker,
                             args=(r, world, port, fn, outdir))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
        return {r: torch.load(os.path.join(outdir, f"rank{r}.pt"),
                              weights_only=False)
                for r in range(world)}


# ---------------------------------------------------------------- workers
def _grad_sync_worker(rank, world):
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    torch.manual_seed(1234)  # same init on both ranks
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    sync = GradientSynchronizer(model, bucket_bytes=128)  # force many buckets
    weights = [0.75, 0.25]
    sync.set_weight(weights[rank])

    torch.manual_seed(rank)  # different data per rank
    x = torch.randn(6, 8)
    y = torch.randn(6, 4)
    sync.zero()
    loss = ((model(x) - y) ** 2).mean()
    loss.backward()
    sync.finish()
    return [p.grad.clone() for p in model.parameters()], x, y


def test_weighted_allreduce_matches_reference_math(free_port):
    """grad <- sum_r w_r * g_r (reference SSGD, dbs.py:291-301)."""
    res = run_distributed(_grad_sync_worker, 2, free_port)
    grads0, x0, y0 = res[0]
    grads1, x1, y1 = res[1]
    # both ranks end with identical reduced gradients
    for g0, g1 in zip(grads0, grads1):
        assert torch.allclose(g0, g1, atol=1e-6)

    # recompute expected: w0*g(rank0 data) + w1*g(rank1 data)
    torch.manual_seed(1234)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    expected = None
    for w, (x, y) in zip([0.75, 0.25], [(x0, y0), (x1, y1)]):
        model.zero_grad()
        ((model(x) - y) ** 2).mean().backward()
        gs = [w * p.grad for p in model.parameters()]
        expected = gs if expected is None else [a + b for a, b in zip(expected, gs)]
    for got, want in zip(grads0, expected):
        assert torch.allclose(got, want, atol=1e-5)


def _time_exchange_worker(rank, world):
    from dynamic_load_balance_distributeddnn_amd.scheduler import \
        exchange_times

    return exchange_times(10.0 + rank)


def test_time_exchange_rank_ordered(free_port):
    res = run_distributed(_time_exchange_worker, 3, free_port)
    for rank in range(3):
        np.testing.assert_allclose(res[rank], [10.0, 11.0, 12.0])


def _engine_worker(rank, world):
    from dynamic_load_balance_distributeddnn_amd.engine import Trainer

    args = get_parser().parse_args(
        ["-d", "true", "-ws", "2", "-b", "32", "-e", "2", "-ds", "mnist",
         "-m", "mnistnet", "-dbs", "true"])
    trainer = Trainer(args, rank, world, torch.device("cpu"), logger=None)
    compute0, sync0, loss0 = trainer.train_epoch(0)
    from dynamic_load_balance_distributeddnn_amd.scheduler import \
        exchange_times
    trainer.nodes_time = exchange_times(compute0)
    compute1, sync1, loss1 = trainer.train_epoch(1)
    val_loss, acc = trainer.validate_epoch(1)
    return dict(losses=(loss0, loss1), batches=trainer.sched.batches.tolist(),
                val=(val_loss, acc), times=trainer.nodes_time.tolist())


def test_mnistnet_two_worker_dbs_cpu(free_port):
    """BASELINE config #1: MnistNet 2-worker DBS on CPU/gloo."""
    res = run_distributed(_engine_worker, 2, free_port)
    assert sum(res[0]["batches"]) == 32
    assert res[0]["batches"] == res[1]["batches"]  # replicated decision
    assert res[0]["times"] == res[1]["times"]
    for r in range(2):
        l0, l1 = res[r]["losses"]
        assert np.isfinite(l0) and np.isfinite(l1)


def _straggler_worker(rank, world):
    import time

    from dynamic_load_balance_distributeddnn_amd.scheduler import (
        DBSScheduler, exchange_times)

    sched = DBSScheduler(world, global_batch=64)
    for _ in range(4):
        compute = 0.2 if rank == 1 else 0.1  # rank 1 is 2x slower
        times = exchange_times(compute)
        sched.step(times)
    return sched.batches.tolist()


def test_straggler_shifts_partition(free_port):
    """DBS core promise: slow rank ends with a smaller batch share."""
    res = run_distributed(_straggler_worker, 2, free_port)
    batches = res[0]
    assert batches == res[1]
    assert sum(batches) == 64
    assert batches[1] < batches[0]
    # converged near the 2:1 speed ratio -> ~(43, 21)
    assert batches[1] <= 24


def _deferred_sync_worker(rank, world):
    """LM-style deferred bucket launch: clip-then-reduce ordering."""
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    torch.manual_seed(7)
    model = torch.nn.Linear(16, 16)
    sync = GradientSynchronizer(model, defer=True)
    sync.set_weight(0.5)
    torch.manual_seed(rank + 50)
    x = torch.randn(4, 16)
    sync.zero()
    (model(x) ** 2).mean().backward()
    # something between backward and reduce (the LM grad clip) must see
    # LOCAL grads — no bucket may have launched yet
    assert all(not b.launched for b in sync.buckets)
    local = [p.grad.clone() for p in model.parameters()]
    sync.finish()
    return local, [p.grad.clone() for p in model.parameters()]


def test_deferred_sync_launches_at_finish(free_port):
    res = run_distributed(_deferred_sync_worker, 2, free_port)
    local0, reduced0 = res[0]
    local1, reduced1 = res[1]
    for l0, l1, r0, r1 in zip(local0, local1, reduced0, reduced1):
        assert torch.allclose(r0, r1, atol=1e-6)       # ranks agree
        assert torch.allclose(r0, 0.5 * l0 + 0.5 * l1, atol=1e-6)


def _direct_write_worker(rank, world):
    """The manual-backward grad path: grads written straight into the
    arena views + mark_ready, never returned to autograd (the GPU
    dense-block backward uses this; semantics pinned here on gloo)."""
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    torch.manual_seed(21)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
    sync = GradientSynchronizer(model, bucket_bytes=128)
    weights = [0.6, 0.4]
    sync.set_weight(weights[rank])
    assert all(getattr(p, "_dlb_sink", None) is sync
               for p in model.parameters())

    torch.manual_seed(100 + rank)
    fake = {n: torch.randn_like(p) for n, p in model.named_parameters()}
    sync.zero()
    with torch.no_grad():
        for n, p in model.named_parameters():
            p.grad.copy_(fake[n])          # direct write into arena view
            sync.mark_ready(p)             # bucket notification
    sync.finish()
    return fake, {n: p.grad.clone() for n, p in model.named_parameters()}


def test_direct_grad_write_and_mark_ready(free_port):
    res = run_distributed(_direct_write_worker, 2, free_port)
    fake0, red0 = res[0]
    fake1, red1 = res[1]
    for n in red0:
        assert torch.allclose(red0[n], red1[n], atol=1e-6)
        want = 0.6 * fake0[n] + 0.4 * fake1[n]
        assert torch.allclose(red0[n], want, atol=1e-5), n


def test_arena_views_are_16B_aligned():
    """The fused kernels vector-load gamma/beta from arena views; the
    8-element offset alignment guarantees 16-byte pointers."""
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer

    model = torch.nn.Sequential(torch.nn.Linear(5, 7), torch.nn.Linear(7, 3),
                                torch.nn.Linear(3, 11))
    sync = GradientSynchronizer(model)
    for p in model.parameters():
        off, _ = sync.offsets[id(p)]
        assert off % 8 == 0
        assert p.grad.data_ptr() % 16 == 0


def _four_rank_worker(rank, world):
    """4-way: time exchange ordering + weighted reduce + partition step
    (the mechanics the driver's 4/8-GPU scaling run exercises)."""
    from dynamic_load_balance_distributeddnn_amd.parallel import \
        GradientSynchronizer
    from dynamic_load_balance_distributeddnn_amd.scheduler import (
        DBSScheduler, exchange_times)

    times = exchange_times(1.0 + rank * 0.5)
    sched = DBSScheduler(world, global_batch=512)
    sched.step(times)

    torch.manual_seed(3)
    model = torch.nn.Linear(16, 8)
    sync = GradientSynchronizer(model, bucket_bytes=64)
    sync.set_weight(float(sched.weights[rank]))
    torch.manual_seed(200 + rank)
    x = torch.randn(4, 16)
    sync.zero()
    (model(x) ** 2).mean().backward()
    sync.finish()
    return dict(times=times.tolist(), batches=sched.batches.tolist(),
                weights=float(sched.weights[rank]),
                grad=model.weight.grad.clone())


def test_four_rank_mechanics(free_port):
    res = run_distributed(_four_rank_worker, 4, free_port)
    t0 = res[0]["times"]
    assert t0 == [1.0, 1.5, 2.0, 2.5]
    for r in range(4):
        assert res[r]["times"] == t0
        assert res[r]["batches"] == res[0]["batches"]
        assert torch.allclose(res[r]["grad"], res[0]["grad"], atol=1e-6)
    assert sum(res[0]["batches"]) == 512
    b = res[0]["batches"]
    assert b[0] >= b[1] >= b[2] >= b[3]     # faster ranks get more
    assert abs(sum(res[r]["weights"] for r in range(4)) - 1.0) < 1e-6
