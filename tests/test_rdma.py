"""RDMA lane fallback behavior (docs/rdma.md): with no libibverbs in
this environment the lane must never activate, never crash, and the
colocated shm lane keeps priority."""

import torch

from mp_util import run_in_processes


def test_rdma_env_without_lib_falls_back(monkeypatch):
    monkeypatch.setenv("BPS_ENABLE_RDMA", "1")
    from byteps_amd.ops import core
    c = core()
    srv = c.Server(0, 1, False)
    srv.start()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        # colocated: shm lane wins regardless of the RDMA env
        addr = kv.ipc_alloc(0, 8192)
        assert addr != 0
        assert kv.ipc_active(0)
        kv.close()
    finally:
        srv.stop()


def test_rdma_env_with_ipc_off_inlines(monkeypatch):
    """shm disabled + no verbs library → ipc_alloc reports unavailable
    and the inline TCP path carries the traffic."""
    monkeypatch.setenv("BPS_ENABLE_RDMA", "1")
    monkeypatch.setenv("BPS_ENABLE_IPC", "0")
    import struct
    from byteps_amd.ops import core
    c = core()
    srv = c.Server(0, 1, False)
    srv.start()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % srv.port])
        assert kv.ipc_alloc(0, 4096) == 0
        n = 512
        payload = struct.pack("<QIIII", n, 1, 0, 0, 0)
        pb = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
        t = kv.submit(0, 5, 9, pb.data_ptr(), len(payload), 0, 0, 0, 0)
        kv.wait(t)
        x = torch.randn(n)
        send = x.clone().view(torch.uint8).reshape(-1).contiguous()
        t = kv.submit(0, 1, 9, send.data_ptr(), n * 4, 0, 0, 0, 1)
        kv.wait(t)
        recv = torch.empty(n * 4, dtype=torch.uint8)
        t = kv.submit(0, 2, 9, 0, 0, recv.data_ptr(), n * 4, 0, 1)
        rl, _ = kv.wait(t)
        assert rl == n * 4
        assert torch.equal(recv.view(torch.float32), x)
        kv.close()
    finally:
        srv.stop()


def _ps_rdma_env_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                            torch.nn.Linear(32, 4))
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters())
    torch.manual_seed(42)
    x, y = torch.randn(8, 16), torch.randn(8, 4)
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_training_with_rdma_env_set():
    """Full PS training with BPS_ENABLE_RDMA=1 on a verbs-less host:
    must complete over the fallback lanes with correct numerics."""
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    try:
        env = {"BPS_FORCE_DISTRIBUTED": "1",
               "BPS_SERVER_URIS": "127.0.0.1:%d" % srv.port,
               "BPS_NUM_SERVER": "1",
               "BPS_ENABLE_RDMA": "1",
               "BPS_ENABLE_IPC": "0"}
        results = run_in_processes(_ps_rdma_env_worker, 1, 2, extra_env=env)
        assert len(results[0]) == 4
    finally:
        srv.stop()
