"""Colocated IPC/shm fast path (kv.cc regions + server locator frames):
payloads must not cross the socket when worker and server share a host,
and everything must fall back to inline TCP when IPC is off.
Reference rationale: ps-lite's IPC van, docs/best-practice.md:32."""

import struct

import pytest
import torch

from mp_util import run_in_processes


def _server():
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    return srv


def _roundtrip(kv, send, recv, key, n, expected=1):
    payload = struct.pack("<QIIII", n, expected, 0, 0, 0)
    pb = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
    t = kv.submit(0, 5, key, pb.data_ptr(), len(payload), 0, 0, 0, 0)
    _l, aux = kv.wait(t)
    assert aux != 0xFFFFFFFFFFFFFFFF
    x = torch.randn(n)
    send.view(torch.float32)[:n].copy_(x)
    t = kv.submit(0, 1, key, send.data_ptr(), n * 4, 0, 0, 0, 1)
    kv.wait(t)
    t = kv.submit(0, 2, key, 0, 0, recv.data_ptr(), recv.numel(), 0, 1)
    rl, _ver = kv.wait(t)
    assert rl == n * 4
    assert torch.equal(recv.view(torch.float32)[:n], x)


def test_ipc_colocated_roundtrip():
    from byteps_amd.ops import core
    c = core()
    srv = _server()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        n = 4096
        import ctypes
        bufs = []
        for _ in range(2):
            addr = kv.ipc_alloc(0, n * 4)
            assert addr != 0, "colocated ipc_alloc must succeed"
            raw = (ctypes.c_uint8 * (n * 4)).from_address(addr)
            bufs.append((raw, torch.frombuffer(raw, dtype=torch.uint8)))
        assert kv.ipc_active(0)
        _roundtrip(kv, bufs[0][1], bufs[1][1], key=1, n=n)
        kv.close()
    finally:
        srv.stop()


def test_ipc_disabled_falls_back_inline(monkeypatch):
    monkeypatch.setenv("BPS_ENABLE_IPC", "0")
    from byteps_amd.ops import core
    c = core()
    srv = _server()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        assert kv.ipc_alloc(0, 4096) == 0
        assert not kv.ipc_active(0)
        n = 1024
        send = torch.empty(n * 4, dtype=torch.uint8)
        recv = torch.empty(n * 4, dtype=torch.uint8)
        _roundtrip(kv, send, recv, key=2, n=n)
        kv.close()
    finally:
        srv.stop()


def test_ipc_mixed_inline_and_region():
    """A pull whose recv buffer is PRIVATE while the push staging is in
    the region (and vice versa) must still work — locate() decides per
    request."""
    from byteps_amd.ops import core
    c = core()
    srv = _server()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        n = 2048
        import ctypes
        addr = kv.ipc_alloc(0, n * 4)
        assert addr
        raw = (ctypes.c_uint8 * (n * 4)).from_address(addr)
        send = torch.frombuffer(raw, dtype=torch.uint8)
        recv = torch.empty(n * 4, dtype=torch.uint8)  # private
        _roundtrip(kv, send, recv, key=3, n=n)
        kv.close()
    finally:
        srv.stop()


def _ps_ipc_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                            torch.nn.Linear(32, 4))
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
    import byteps_amd.common as C
    ipc = C._state.kv.ipc_active(0) if C._state.kv is not None else False
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return ipc, out


@pytest.mark.parametrize("ipc_on", [True, False])
def test_ps_training_parity_ipc_vs_inline(ipc_on):
    """End-to-end PS training must produce identical results whether the
    wire is shm or inline TCP."""
    from byteps_amd.ops import _core
    srv = _server()
    try:
        env = {"BPS_FORCE_DISTRIBUTED": "1",
               "BPS_SERVER_URIS": "127.0.0.1:%d" % srv.port,
               "BPS_NUM_SERVER": "1",
               "BPS_ENABLE_IPC": "1" if ipc_on else "0"}
        results = run_in_processes(_ps_ipc_worker, 2, 3, extra_env=env)
        for rank, (ipc, _params) in enumerate(results):
            assert ipc == ipc_on, "rank %d ipc=%s" % (rank, ipc)
        # both ranks converge to identical params
        for a, b in zip(results[0][1], results[1][1]):
            assert torch.equal(a, b)
    finally:
        srv.stop()


def test_ipc_region_growth(monkeypatch):
    """Staging demand beyond one region must spawn more regions (each
    with its own hello) and keep locators valid across all of them."""
    monkeypatch.setenv("BPS_IPC_REGION_MB", "1")
    from byteps_amd.ops import core
    c = core()
    srv = _server()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        import ctypes
        bufs = []
        # 6 x 512 KiB spans 3+ one-MiB regions
        for i in range(6):
            nb = 512 * 1024
            addr = kv.ipc_alloc(0, nb)
            assert addr != 0, "region growth failed at %d" % i
            raw = (ctypes.c_uint8 * nb).from_address(addr)
            bufs.append(torch.frombuffer(raw, dtype=torch.uint8))
        # round-trip through buffers in the FIRST and LAST region
        _roundtrip(kv, bufs[0], bufs[1], key=10, n=1024)
        _roundtrip(kv, bufs[4], bufs[5], key=11, n=1024)
        kv.close()
    finally:
        srv.stop()
