"""Multi-rank engine epochs ON the GPU: 2 ranks sharing one MI355X over
gloo — the reference's oversubscription shape (`-gpu 0,0`, README.md:
23-28) and the BASELINE config-#4 straggler mechanism.  RCCL cannot host
two ranks on one GPU, so this is exactly the committed config-#4 route
(launcher.py) exercised end-to-end on hardware: kernels on cuda:0 in
both processes, gradient exchange over gloo, DBS feedback loop live."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs ROCm GPU")


def _worker(rank, world, port, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DLB_SYNTH_SCALE"] = "0.01"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dynamic_load_balance_distributeddnn_amd.cli import get_parser
        from dynamic_load_balance_distributeddnn_amd.engine import Trainer

        args = get_parser().parse_args(
            ["-d", "false", "-ws", "2", "-b", "64", "-e", "2",
             "-ds", "cifar10", "-m", "densenet", "-dbs", "true"])
        tr = Trainer(args, rank, world, torch.device("cuda:0"), logger=None)
        from dynamic_load_balance_distributeddnn_amd.scheduler import \
            exchange_times

        losses = []
        for epoch in range(2):
            compute, sync, loss = tr.train_epoch(epoch)
            tr.nodes_time = exchange_times(compute, tr.device)
            losses.append(loss)
        torch.save(dict(losses=losses,
                        batches=tr.sched.batches.tolist(),
                        nodes_time=tr.nodes_time.tolist()),
                   os.path.join(outdir, f"rank{rank}.pt"))
    finally:
        dist.destroy_process_group()


@needs_gpu
def test_two_rank_engine_epochs_share_one_gpu(tmp_path):
    port = 29617
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(600)
        assert p.exitcode == 0, f"worker exit {p.exitcode}"
    res = {r: torch.load(tmp_path / f"rank{r}.pt", weights_only=False)
           for r in range(2)}
    # replicated-deterministic DBS state and a live gradient exchange
    assert res[0]["batches"] == res[1]["batches"]
    assert sum(res[0]["batches"]) == 64
    assert res[0]["nodes_time"] == res[1]["nodes_time"]
    for r in range(2):
        assert all(np.isfinite(v) for v in res[r]["losses"])
