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


# -- elastic re-bucketing: world 2 → 3 ---------------------------------------

def _make_grow_model():
    torch.manual_seed(7)
    return torch.nn.Sequential(
        torch.nn.Linear(23, 130), torch.nn.ReLU(), torch.nn.Linear(130, 5))


def _grow_worker(rank, world, tmpdir):
    import os
    import time
    import byteps_amd.torch as bps

    base_port = int(os.environ["MASTER_PORT"])
    m = _make_grow_model()
    torch.manual_seed(42)
    xs = [torch.randn(4, 23) for _ in range(3)]
    ys = [torch.randn(4, 5) for _ in range(3)]

    def mark(tag):
        open(os.path.join(tmpdir, "%s.%d" % (tag, rank)), "w").close()

    def wait(tag, ranks, deadline=180.0):
        t0 = time.time()
        while not all(os.path.exists(os.path.join(tmpdir, "%s.%d" % (tag, r)))
                      for r in ranks):
            if time.time() - t0 > deadline:
                raise RuntimeError(
                    "rank %d: timed out waiting for %s markers (a peer "
                    "died before marking — see its traceback)" % (rank, tag))
            time.sleep(0.05)

    if rank < 2:
        os.environ["WORLD_SIZE"] = "2"
        bps.init()
        opt = bps.DistributedOptimizer(
            torch.optim.SGD(m.parameters(), lr=0.05),
            named_parameters=m.named_parameters())
        for _ in range(2):
            opt.zero_grad()
            ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
            opt.step()
        bps.suspend()
        mark("suspended")
        wait("suspended", (0, 1))
        os.environ["WORLD_SIZE"] = "3"
        bps.resume(num_workers=3, num_servers=0)
    else:
        wait("suspended", (0, 1))
        os.environ["WORLD_SIZE"] = "3"
        os.environ["MASTER_PORT"] = str(base_port + 1)  # resume's rotation
        bps.init()
        opt = bps.DistributedOptimizer(
            torch.optim.SGD(m.parameters(), lr=0.05),
            named_parameters=m.named_parameters())

    # elastic join protocol: sync params from rank 0 (reference
    # broadcast_parameters at (re)start, torch/__init__.py:268-299)
    bps.broadcast_parameters(m.state_dict(), root_rank=0)
    # one synced step at world 3: every bucket must divide by 3 and the
    # averaged grads must match the 3-way big batch
    opt.zero_grad()
    ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
    opt.synchronize()
    eng = opt._engine
    assert eng.world == 3
    for b in eng.buckets:
        assert b.buffer.numel() % 3 == 0
    grads = [p.grad.detach().clone() for p in m.parameters()]
    params = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return grads, params


def test_elastic_rebucket_grow_2_to_3(tmp_path):
    results = run_in_processes(_grow_worker, 3, str(tmp_path),
                               extra_env={"BPS_PARTITION_BYTES": "16384"})
    # all ranks agree bit-for-bit after the world-3 step
    for got, _ in results[1:]:
        for a, b in zip(results[0][0], got):
            assert torch.equal(a, b), "ranks disagree after re-bucketing"
    # and the averaged grad equals the big-batch grad on the broadcast
    # (rank-0) params
    m = _make_grow_model()
    with torch.no_grad():
        for p, v in zip(m.parameters(), results[0][1]):
            p.copy_(v)
    torch.manual_seed(42)
    xs = [torch.randn(4, 23) for _ in range(3)]
    ys = [torch.randn(4, 5) for _ in range(3)]
    loss = sum(((m(x) - y) ** 2).mean() for x, y in zip(xs, ys)) / 3
    loss.backward()
    for got, exp in zip(results[0][0], [p.grad for p in m.parameters()]):
        assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6), \
            "averaged grads diverge from the 3-way big batch"


# -- dist_launch units (SSH fan-out command construction) --------------------

def test_dist_launch_hostfile_and_env(tmp_path, monkeypatch):
    from byteps_amd.launcher.dist_launch import read_hostfile, forwarded_env
    hf = tmp_path / "hosts"
    hf.write_text("10.0.0.1:8\n# comment\n10.0.0.2\n\n10.0.0.3:4 # gpu4\n")
    assert read_hostfile(str(hf)) == ["10.0.0.1", "10.0.0.2", "10.0.0.3"]
    monkeypatch.setenv("BPS_NUM_SERVER", "2")
    monkeypatch.setenv("DMLC_PS_ROOT_URI", "10.0.0.9")
    monkeypatch.setenv("HOME", "/root")          # must NOT forward
    pairs = forwarded_env(["EXTRA=1"])
    assert "BPS_NUM_SERVER=2" in pairs
    assert "DMLC_PS_ROOT_URI=10.0.0.9" in pairs
    assert "EXTRA=1" in pairs
    assert not any(p.startswith("HOME=") for p in pairs)
