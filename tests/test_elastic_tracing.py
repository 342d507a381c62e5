"""Elastic suspend/resume, tracing output, telemetry, and launcher units."""

import json
import os
import subprocess
import sys

import torch

from mp_util import run_in_processes


def _elastic(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    m = torch.nn.Linear(8, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    from byteps_amd.common import _state
    keys_before = dict(_state.registry._keys)

    def step():
        opt.zero_grad()
        ((m(torch.randn(4, 8))) ** 2).mean().backward()
        opt.step()

    step()
    bps.suspend()
    bps.resume(num_workers=1, num_servers=0)
    keys_after = dict(_state.registry._keys)
    step()     # must still work (hooks re-armed)
    ok = keys_before == keys_after and bps.initialized()
    bps.shutdown()
    return bool(ok)


def test_elastic_suspend_resume():
    assert all(run_in_processes(_elastic, 2))


def _traced(rank, world, trace_dir):
    import byteps_amd.torch as bps
    bps.init()
    m = torch.nn.Linear(8, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    for _ in range(5):
        opt.zero_grad()
        ((m(torch.randn(4, 8))) ** 2).mean().backward()
        opt.step()
    bps.shutdown()
    return True


def test_chrome_trace_written(tmp_path):
    trace_dir = str(tmp_path / "traces")
    run_in_processes(_traced, 2, trace_dir, extra_env={
        "BPS_TRACE_ON": "1", "BPS_TRACE_DIR": trace_dir,
        "BPS_TRACE_START_STEP": "1", "BPS_TRACE_END_STEP": "3"})
    for rank in range(2):
        path = os.path.join(trace_dir, str(rank), "comm.json")
        assert os.path.exists(path), path
        data = json.load(open(path))
        evs = data["traceEvents"]
        assert evs, "no trace events"
        assert all(e["ph"] == "X" and "dur" in e for e in evs)
        assert any(e["pid"].startswith("Comm.") for e in evs)


def test_telemetry_speed():
    from byteps_amd.common import telemetry
    telemetry.record(10_000_000)
    ts, mbps = telemetry.get_pushpull_speed()
    assert ts > 0
    assert mbps >= 0.0


def test_launcher_spawns_workers(tmp_path):
    script = tmp_path / "probe.py"
    script.write_text(
        "import os, sys\n"
        "print('R', os.environ['RANK'], os.environ['WORLD_SIZE'],"
        " os.environ['LOCAL_RANK'])\n")
    env = dict(os.environ)
    env.update({"BPS_ROLE": "worker", "BPS_LOCAL_SIZE": "2",
                "BPS_NUM_WORKER": "1"})
    out = subprocess.run(
        [sys.executable, "-m", "byteps_amd.launcher.launch",
         sys.executable, str(script)],
        env=env, capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    lines = sorted(l for l in out.stdout.splitlines() if l.startswith("R "))
    assert lines == ["R 0 2 0", "R 1 2 1"], out.stdout


def test_launcher_cpu_ranges():
    from byteps_amd.launcher.launch import cpu_ranges
    r = cpu_ranges(2)
    assert len(r) == 2
    assert set(r[0]).isdisjoint(r[1]) or len(os.sched_getaffinity(0)) < 2
