"""The driver depends on bench.py's exact contract — pin it."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run(args, env=None):
    out = subprocess.run([sys.executable, os.path.join(REPO, "bench.py")]
                         + args, capture_output=True, text=True, timeout=600,
                         cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-1500:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    return json.loads(line)


def test_bench_json_contract_cpu():
    rec = _run(["--steps", "2", "--warmup", "1", "--device", "cpu",
                "--global-batch", "8", "--model", "mnistnet"])
    assert REQUIRED.issubset(rec.keys())
    assert rec["n_gpus"] == 1
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True
    assert rec["data"] == "synthetic"
    assert rec["config"]["global_batch"] == 8
    assert rec["value"] > 0 and rec["ms_per_step"] > 0


def test_bench_torchrun_two_ranks(free_port):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               MASTER_PORT=str(free_port))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--device", "cpu",
         "--global-batch", "8", "--model", "mnistnet"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-1500:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dbs-dp2"


def test_bench_flagship_two_ranks_cpu(free_port):
    """The driver's SCALE run shape, on gloo: flagship model, 2 ranks,
    split global batch, MAX-over-ranks aggregation."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               MASTER_PORT=str(free_port))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--device", "cpu",
         "--global-batch", "16", "--model", "densenet"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-1500:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["model"] == "DenseNet-121"
    assert rec["config"]["global_batch"] == 16
    assert rec["value"] > 0
