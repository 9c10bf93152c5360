"""bench.py contract test: full distributed flow (torchrun rendezvous,
BucketedDDP, max-over-ranks timing, JSON line) on CPU/gloo with a tiny model.
The driver runs the same file with cuda/RCCL on MI355X."""

import json
import os
import subprocess
import sys


def test_bench_distributed_cpu(tmp_path):
    env = dict(os.environ)
    env.update(AL_BENCH_DEVICE="cpu", AL_BENCH_BACKEND="gloo",
               MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch", "4", "--model", "resnet18", "--img", "32",
         "--classes", "10"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))), env=env)
    assert out.returncode == 0, out.stdout + out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["metric"] == "images/sec (train)"
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "weak"
    assert rec["config"]["global_batch"] == 8
    assert rec["value"] > 0
    assert rec["dtype"] == "bf16"


def test_bench_distributed_cpu_world4(tmp_path):
    """World=4 on gloo: exercises the bucket-order broadcast and uniform
    collectives across more ranks than the standard world=2 tests (the
    driver's 8-GPU scaling run is the first hardware execution)."""
    env = dict(os.environ)
    env.update(AL_BENCH_DEVICE="cpu", AL_BENCH_BACKEND="gloo",
               MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29519", "bench.py", "--gpus", "4", "--steps", "3",
         "--warmup", "1", "--batch", "4", "--model", "resnet18", "--img", "32",
         "--classes", "10"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))), env=env)
    assert out.returncode == 0, out.stdout + out.stderr
    rec = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert rec["n_gpus"] == 4
    assert rec["config"]["global_batch"] == 16
    assert rec["value"] > 0


def test_bench_single_cpu():
    env = dict(os.environ)
    env.update(AL_BENCH_DEVICE="cpu")
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1", "--batch", "4",
         "--model", "resnet18", "--img", "32", "--classes", "10"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))), env=env)
    assert out.returncode == 0, out.stdout + out.stderr
    rec = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert rec["n_gpus"] == 1 and rec["value"] > 0
