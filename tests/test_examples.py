"""Keep the example scripts runnable (CPU, small) — the reference's
examples doubled as its smoke tests."""

import os
import subprocess
import sys

import torch

from mp_util import free_port, run_in_processes

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=300, env_extra=None):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env.update(env_extra or {})
    return subprocess.run([sys.executable] + args, cwd=ROOT, env=env,
                          capture_output=True, text=True, timeout=timeout)


def test_mnist_example_two_ranks():
    r = _run(["-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", str(free_port()),
              "examples/train_mnist_mlp.py"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "step 40 loss" in r.stdout


def test_elastic_example():
    r = _run(["examples/elastic_benchmark.py"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "elastic cycle complete" in r.stdout


def test_benchmark_example_cpu():
    r = _run(["examples/benchmark_byteps_amd.py", "--no-cuda",
              "--model", "resnet50", "--batch-size", "2",
              "--num-warmup-batches", "1", "--num-batches-per-iter", "1",
              "--num-iters", "2"], timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "Total img/sec" in r.stdout


def _metric(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    avg = bps.metric_average(float(rank + 1), "acc")
    bps.shutdown()
    return avg


def test_metric_average():
    world = 2
    results = run_in_processes(_metric, world)
    for r in results:
        assert abs(r - 1.5) < 1e-6
