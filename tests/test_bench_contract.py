"""bench.py driver contract: single-process and torchrun multi-rank (the
exact launch shape the benchmark driver uses), JSON output schema."""

import json
import os
import subprocess
import sys

from mp_util import free_port

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _check_json(line: str, n_gpus: int):
    out = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["n_gpus"] == n_gpus
    assert out["value"] > 0
    assert out["data"] == "synthetic"
    assert out["scaling"] == "weak"
    assert "model" in out["config"] and "global_batch" in out["config"]
    return out


def test_bench_single_process():
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "mlp", "--steps", "3",
         "--warmup", "1", "--device", "cpu"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    _check_json(line, 1)


def test_bench_torchrun_two_ranks():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "2",
         "--model", "mlp", "--steps", "3", "--warmup", "1",
         "--device", "cpu"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line from rank 0"
    out = _check_json(lines[0], 2)
    assert out["config"]["parallelism"] == "dp2"
    # whole-job aggregate: 2 ranks → global batch = 2 × per-rank
    assert out["config"]["global_batch"] == 128


def test_bench_torchrun_four_ranks():
    """The driver's SCALE run launches N=4 the same way — 4-rank gloo
    must work end-to-end before it meets RCCL."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "4",
         "--model", "mlp", "--steps", "3", "--warmup", "1",
         "--device", "cpu"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1
    out = _check_json(lines[0], 4)
    assert out["config"]["parallelism"] == "dp4"


def test_bench_torchrun_resnet50_two_ranks():
    """The flagship model itself (not just the MLP) through the torchrun
    shape: channels_last buckets + DDP broadcast + engine collectives."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "2",
         "--model", "resnet50", "--batch-size", "1", "--steps", "1",
         "--warmup", "0", "--device", "cpu"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    out = _check_json(lines[0], 2)
    assert out["config"]["model"] == "resnet50"


def test_bench_torchrun_eight_ranks():
    """The full driver SCALE shape at N=8 (gloo, MLP) — rendezvous,
    8-way bucket math, aggregate math."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "8",
         "--model", "mlp", "--steps", "2", "--warmup", "1",
         "--device", "cpu"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    out = _check_json(lines[0], 8)
    assert out["config"]["parallelism"] == "dp8"
    assert out["config"]["global_batch"] == 8 * 64
