"""Hierarchical PS path tests: workers (gloo, CPU) + the native C++ KV
server colocated on localhost — the reference's single-machine
forced-distributed harness (tests/meta_test.py:27-85)."""

import numpy as np
import pytest
import torch

from mp_util import run_in_processes


@pytest.fixture()
def server():
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    yield "127.0.0.1:%d" % srv.port
    srv.stop()


def _ps_env(uri, extra=None):
    env = {"BPS_FORCE_DISTRIBUTED": "1", "BPS_SERVER_URIS": uri,
           "BPS_NUM_SERVER": "1"}
    env.update(extra or {})
    return env


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def _baseline(world, steps, lr=0.05):
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = torch.cat(xs), torch.cat(ys)
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


def _ps_worker(rank, world, steps, partition_bytes):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    if partition_bytes:
        for e in __import__("byteps_amd.torch.engine",
                            fromlist=["_engines"])._engines:
            pass
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = xs[rank], ys[rank]
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_single_worker_parity(server):
    """1 worker + 1 colocated server: full KV roundtrip, must equal local
    SGD exactly (BASELINE.json config 1 plumbing)."""
    expected = _baseline(1, 3)
    results = run_in_processes(_ps_worker, 1, 3, None,
                               extra_env=_ps_env(server))
    for got, exp in zip(results[0], expected):
        assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6)


def test_ps_two_nodes_parity(server):
    """2 simulated single-GPU nodes (local_size=1): server sums across
    nodes; must equal big-batch baseline."""
    expected = _baseline(2, 3)
    results = run_in_processes(
        _ps_worker, 2, 3, None,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1"}))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def test_ps_one_node_sharded_parity(server):
    """2 ranks on ONE node: intra-node reduce + per-shard push/pull +
    all-gather; must equal big-batch baseline."""
    expected = _baseline(2, 3)
    results = run_in_processes(_ps_worker, 2, 3, None,
                               extra_env=_ps_env(server))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _pp_tensor(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    t = torch.arange(100, dtype=torch.float32) + rank * 100
    out = bps.push_pull(t, average=False, name="pp.ps")
    expect = sum(torch.arange(100, dtype=torch.float32) + r * 100
                 for r in range(world))
    ok = torch.allclose(out, expect)
    bps.shutdown()
    return bool(ok)


def test_ps_functional_push_pull(server):
    assert all(run_in_processes(
        _pp_tensor, 2, extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                                  "LOCAL_WORLD_SIZE": "1"})))


def _ps_topk_full(rank, world, steps):
    """topk with k == full size is lossless → exact parity through the
    compressed wire format."""
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    opt = bps.DistributedOptimizer(
        opt, named_parameters=m.named_parameters(),
        compression_params={"compressor_type": "topk",
                            "compressor_k": 1 << 20})
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = xs[rank], ys[rank]
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_topk_lossless_parity(server):
    expected = _baseline(2, 2)
    results = run_in_processes(
        _ps_topk_full, 2, 2,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1",
                                   "BPS_MIN_COMPRESS_BYTES": "0"}))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-4, atol=1e-5)


def _ps_onebit(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.01)
    opt = bps.DistributedOptimizer(
        opt, named_parameters=m.named_parameters(),
        compression_params={"compressor_type": "onebit",
                            "ef_type": "vanilla"})
    torch.manual_seed(42)
    x = torch.randn(16, 16)
    y = torch.randn(16, 4)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = ((m(x) - y) ** 2).mean()
        loss.backward()
        losses.append(float(loss))
        opt.step()
    bps.shutdown()
    return losses


def test_ps_onebit_trains(server):
    """onebit + EF through worker-GPU(CPU here)-server roundtrip: loss
    must decrease (numerics are lossy by design)."""
    results = run_in_processes(
        _ps_onebit, 2, 30,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1",
                                   "BPS_MIN_COMPRESS_BYTES": "0"}))
    for losses in results:
        assert losses[-1] < losses[0] * 0.9, losses[:3] + losses[-3:]


def _async_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    torch.manual_seed(0)
    m = torch.nn.Linear(16, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.02)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    torch.manual_seed(100 + rank)
    x = torch.randn(32, 16)
    w = torch.randn(16, 4)
    y = x @ w
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = ((m(x) - y) ** 2).mean()
        loss.backward()
        losses.append(float(loss))
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return losses, out


def test_ps_async_mode(server):
    """BYTEPS_ENABLE_ASYNC: workers push weight deltas without a round
    barrier; loss must decrease and replicas converge to the shared
    server weights."""
    results = run_in_processes(
        _async_worker, 2, 40,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1",
                                   "BPS_ENABLE_ASYNC": "1"}))
    for losses, _params in results:
        assert losses[-1] < losses[0] * 0.8, (losses[0], losses[-1])


def test_ps_reduce_roots_parity(server):
    """BPS_REDUCE_ROOTS alternative strategy: whole-bucket reduce to a
    hashed root + root-only push/pull + broadcast."""
    expected = _baseline(2, 3)
    results = run_in_processes(
        _ps_worker, 2, 3, None,
        extra_env=_ps_env(server, {"BPS_REDUCE_ROOTS": "1"}))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def test_ps_three_ranks_one_node_parity(server):
    """Non-power-of-two world: bucket alignment must make shards divide
    exactly (a fixed align=64 silently dropped the tail at world=3)."""
    expected = _baseline(3, 3)
    results = run_in_processes(_ps_worker, 3, 3, None,
                               extra_env=_ps_env(server))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _ps_fp8(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    opt = bps.DistributedOptimizer(
        opt, named_parameters=m.named_parameters(),
        compression_params={"compressor_type": "fp8", "ef_type": "vanilla"})
    torch.manual_seed(42)
    # same regressable data on every rank → loss floor near 0
    x = torch.randn(16, 16)
    w = torch.randn(16, 4)
    y = x @ w * 0.3
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = ((m(x) - y) ** 2).mean()
        loss.backward()
        losses.append(float(loss))
        opt.step()
    bps.shutdown()
    return losses


def test_ps_fp8_trains(server):
    """fp8 e4m3 wire through the full worker↔server loop: near-lossless
    (2^-3 relative), loss must track the uncompressed trajectory."""
    results = run_in_processes(
        _ps_fp8, 2, 50,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1",
                                   "BPS_MIN_COMPRESS_BYTES": "0"}))
    for losses in results:
        assert losses[-1] < losses[0] * 0.5, (losses[0], losses[-1])


def test_ps_two_nodes_two_ranks_each(server):
    """4 ranks as 2 simulated nodes × 2 GPUs: intra-node reduce within
    node subgroups, per-shard push with expected pushers = 2 nodes,
    all-gather within each node — the full hierarchical topology."""
    expected = _baseline(4, 3)
    results = run_in_processes(
        _ps_worker, 4, 3, None,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "2",
                                   "LOCAL_WORLD_SIZE": "2"}))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _ps_overrides(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    # lossless topk for layer "0.", raw for everything else — exact parity
    opt = bps.DistributedOptimizer(
        opt, named_parameters=m.named_parameters(),
        compression_params={
            "param_overrides": {"0.": {"compressor_type": "topk",
                                       "compressor_k": 1 << 20}}})
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    for _ in range(steps):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_per_param_compression_overrides(server):
    """param_overrides route different params through different wire
    codecs (reference per-param byteps_* attrs) — lossless settings must
    preserve exact parity."""
    expected = _baseline(2, 2)
    results = run_in_processes(
        _ps_overrides, 2, 2,
        extra_env=_ps_env(server, {"BPS_LOCAL_SIZE": "1",
                                   "LOCAL_WORLD_SIZE": "1",
                                   "BPS_MIN_COMPRESS_BYTES": "0"}))
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-4, atol=1e-5)


def _sched_worker(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    t = torch.ones(64) * (rank + 1)
    out = bps.push_pull(t, average=False, name="sched.pp")
    ok = torch.allclose(out, torch.full((64,), float(sum(range(1, world + 1)))))
    bps.shutdown()
    return bool(ok)


def test_ps_scheduler_discovery():
    """Workers with no BPS_SERVER_URIS discover the server through the
    rendezvous scheduler (reference DMLC_PS_ROOT_URI flow)."""
    from byteps_amd.launcher.scheduler import Scheduler, register_server
    from byteps_amd.ops import _core
    sched = Scheduler(port=0, num_servers=1).start()
    srv = _core.Server(0, 2, False)
    srv.start()
    try:
        register_server("127.0.0.1", srv.port, "127.0.0.1", sched.port)
        results = run_in_processes(
            _sched_worker, 2,
            extra_env={"BPS_FORCE_DISTRIBUTED": "1",
                       "BPS_NUM_SERVER": "1",
                       "BPS_ROOT_URI": "127.0.0.1",
                       "BPS_ROOT_PORT": str(sched.port),
                       "BPS_LOCAL_SIZE": "1",
                       "LOCAL_WORLD_SIZE": "1"})
        assert all(results)
    finally:
        srv.stop()
        sched.stop()


# -- multiple servers (BASELINE config 3 names 2 CPU PS) ---------------------

def _make_wide_model(seed=0):
    torch.manual_seed(seed)
    # ~70k elems → ~16 buckets at the 4096-elem floor, so BOTH servers
    # get keys
    return torch.nn.Sequential(
        torch.nn.Linear(64, 512), torch.nn.ReLU(),
        torch.nn.Linear(512, 64))


def _wide_baseline(world, steps, lr=0.05):
    m = _make_wide_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr)
    torch.manual_seed(42)
    xs = [torch.randn(8, 64) for _ in range(world)]
    ys = [torch.randn(8, 64) for _ in range(world)]
    for _ in range(steps):
        opt.zero_grad()
        loss = sum(((m(x) - y) ** 2).mean() for x, y in zip(xs, ys)) / world
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


def _ps_multi_server_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_wide_model()
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters())
    torch.manual_seed(42)
    xs = [torch.randn(8, 64) for _ in range(world)]
    ys = [torch.randn(8, 64) for _ in range(world)]
    for _ in range(steps):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    # both servers must actually own keys (djb2 spread)
    import byteps_amd.common as C
    loads = list(C._state.assigner.load)
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return loads, out


def test_ps_two_servers_parity():
    """2 workers + 2 PS shards: key→server hash sharding end-to-end
    (reference mixed/colocated placement, common/global.cc:628-677)."""
    from byteps_amd.ops import _core
    s1 = _core.Server(0, 2, False)
    s2 = _core.Server(0, 2, False)
    s1.start()
    s2.start()
    try:
        env = {"BPS_FORCE_DISTRIBUTED": "1",
               "BPS_NUM_SERVER": "2",
               "BPS_SERVER_URIS": "127.0.0.1:%d,127.0.0.1:%d"
                                  % (s1.port, s2.port),
               "BPS_PARTITION_BYTES": "4096"}
        expected = _wide_baseline(2, 3)
        results = run_in_processes(_ps_multi_server_worker, 2, 3,
                                   extra_env=env)
        for loads, params in results:
            assert len(loads) == 2
            assert min(loads) > 0, "one server owns no keys: %s" % loads
            for got, exp in zip(params, expected):
                assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5)
    finally:
        s1.stop()
        s2.stop()


# -- gradient accumulation through the PS path -------------------------------

def _ps_accum_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters(),
        backward_passes_per_step=2)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    for _ in range(steps):
        for _ in range(2):
            ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
            opt.step()
        opt.zero_grad()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_with_grad_accumulation(server):
    """backward_passes_per_step=2 through the full KV round trip: the
    accumulated (doubled) gradient must reach the server once per
    window."""
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(2)]
    ys = [torch.randn(8, 4) for _ in range(2)]
    for _ in range(3):
        opt.zero_grad()
        loss = 2 * sum(((m(x) - y) ** 2).mean()
                       for x, y in zip(xs, ys)) / 2
        loss.backward()
        opt.step()
    expected = [p.detach().clone() for p in m.parameters()]
    results = run_in_processes(_ps_accum_worker, 2, 3,
                               extra_env=_ps_env(server))
    for got, exp in zip(results[0], expected):
        assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5)


# -- elastic suspend/resume THROUGH the PS path ------------------------------

def _ps_resume_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters())
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    for _ in range(steps):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    bps.suspend()
    bps.resume(num_workers=world, num_servers=1)
    for _ in range(steps):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_suspend_resume_continuity(server):
    """Suspend + resume with the SAME world keeps PS keys stable (server
    re-init validation accepts the matching config) and training
    continues exactly."""
    expected = _baseline(1, 6)
    results = run_in_processes(_ps_resume_worker, 1, 3,
                               extra_env=_ps_env(server))
    for got, exp in zip(results[0], expected):
        assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6)


def test_server_process_entry():
    """`python -m byteps_amd.server` — the standalone server process
    (reference byteps/server/__init__.py) — serves a KV round trip."""
    import os
    import re
    import signal
    import struct
    import subprocess
    import sys
    import time
    env = dict(os.environ)
    env["BPS_SERVER_PORT"] = "0"
    env["BPS_LOG_LEVEL"] = "INFO"
    proc = subprocess.Popen(
        [sys.executable, "-m", "byteps_amd.server"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True, cwd=os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
    try:
        port = None
        t0 = time.time()
        while time.time() - t0 < 30:
            line = proc.stdout.readline()
            m = re.search(r"listening on :(\d+)", line or "")
            if m:
                port = int(m.group(1))
                break
        assert port, "server did not announce its port"
        from byteps_amd.ops import _core
        kv = _core.KVClient(0, ["127.0.0.1:%d" % port])
        payload = struct.pack("<QIIII", 128, 1, 0, 0, 0)
        pb = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
        t = kv.submit(0, 5, 3, pb.data_ptr(), len(payload), 0, 0, 0, 0)
        kv.wait(t)
        x = torch.randn(128)
        send = x.view(torch.uint8).reshape(-1).clone()
        t = kv.submit(0, 1, 3, send.data_ptr(), 512, 0, 0, 0, 1)
        kv.wait(t)
        recv = torch.empty(512, dtype=torch.uint8)
        t = kv.submit(0, 2, 3, 0, 0, recv.data_ptr(), 512, 0, 1)
        rl, _ = kv.wait(t)
        assert rl == 512
        assert torch.equal(recv.view(torch.float32), x)
        kv.close()
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
