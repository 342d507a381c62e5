"""World 3/4/6/8 coverage (gloo, CPU) — the bucket alignment
(lcm(64, world), engine.py) and PS shard math (nelem // node_world,
ps_pipeline.py) exist FOR these worlds; the driver's 8-GPU scaling run
must not be the first time they execute."""

import pytest
import torch

from mp_util import run_in_processes


def _make_model(seed=0):
    torch.manual_seed(seed)
    # mixed sizes incl. a prime-ish dim so alignment padding is exercised
    return torch.nn.Sequential(
        torch.nn.Linear(17, 129), torch.nn.ReLU(),
        torch.nn.Linear(129, 37), torch.nn.ReLU(),
        torch.nn.Linear(37, 5))


def _data(world, batch=4):
    torch.manual_seed(42)
    xs = [torch.randn(batch, 17) for _ in range(world)]
    ys = [torch.randn(batch, 5) for _ in range(world)]
    return xs, ys


def _baseline_grads(world):
    m = _make_model()
    xs, ys = _data(world)
    loss = sum(((m(x) - y) ** 2).mean() for x, y in zip(xs, ys)) / world
    loss.backward()
    return [p.grad.detach().clone() for p in m.parameters()]


def _allreduce_worker(rank, world, partition_bytes):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    from byteps_amd.torch.engine import GradEngine
    eng = GradEngine(list(m.named_parameters()),
                     partition_bytes=partition_bytes)
    # alignment invariant: every bucket divides evenly for this world
    for b in eng.buckets:
        assert b.buffer.numel() % world == 0, \
            "bucket %d (%d elems) not divisible by world %d" % (
                b.plan.index, b.buffer.numel(), world)
    xs, ys = _data(world)
    ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
    eng.synchronize()
    grads = [p.grad.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return grads


@pytest.mark.parametrize("world", [3, 4, 6, 8])
def test_allreduce_grads_world(world):
    expected = _baseline_grads(world)
    results = run_in_processes(_allreduce_worker, world, 4096)
    for r in range(world):
        for got, exp in zip(results[r], expected):
            assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6), \
                "world=%d rank=%d grad mismatch" % (world, r)


# -- PS path at awkward worlds ----------------------------------------------

@pytest.fixture()
def server():
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    yield "127.0.0.1:%d" % srv.port
    srv.stop()


def _ps_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters())
    xs, ys = _data(world)
    for _ in range(steps):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def _ps_baseline(world, steps, lr=0.05):
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr)
    xs, ys = _data(world)
    for _ in range(steps):
        opt.zero_grad()
        loss = sum(((m(x) - y) ** 2).mean() for x, y in zip(xs, ys)) / world
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


@pytest.mark.parametrize("world", [3, 6])
def test_ps_sharded_world(server, world):
    """PS reduce-scatter sharding at worlds where 64 ∤ world: the
    lcm(64, world) bucket alignment must give exact shard division and
    bit-consistent results across ranks."""
    expected = _ps_baseline(world, 2)
    env = {"BPS_FORCE_DISTRIBUTED": "1", "BPS_SERVER_URIS": server,
           "BPS_NUM_SERVER": "1", "BPS_PARTITION_BYTES": "4096"}
    results = run_in_processes(_ps_worker, world, 2, extra_env=env)
    for r in range(world):
        for got, exp in zip(results[r], expected):
            assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5), \
                "PS world=%d rank=%d param mismatch" % (world, r)


def _rings_worker(rank, world, nrings):
    import os
    os.environ["BPS_NUM_RINGS"] = str(nrings)
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    from byteps_amd.torch.engine import GradEngine
    eng = GradEngine(list(m.named_parameters()), partition_bytes=4096)
    assert len(eng._rings) == nrings
    xs, ys = _data(world)
    ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
    eng.synchronize()
    grads = [p.grad.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return grads


def test_multi_ring_collectives_world4():
    """BPS_NUM_RINGS=2: buckets round-robin across two communicators;
    results must match the single-ring big-batch baseline (reference
    NcclManagerExpr multi-ring, nccl_manager.cc:216-318)."""
    expected = _baseline_grads(4)
    results = run_in_processes(_rings_worker, 4, 2)
    for r in range(4):
        for got, exp in zip(results[r], expected):
            assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6)


def _rings_wire_worker(rank, world):
    import os
    os.environ["BPS_NUM_RINGS"] = "2"
    os.environ["BPS_COMM_DTYPE"] = "bf16"
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    from byteps_amd.torch.engine import GradEngine
    eng = GradEngine(list(m.named_parameters()), partition_bytes=4096)
    xs, ys = _data(world)
    ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
    eng.synchronize()
    grads = [p.grad.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return grads


def test_rings_with_bf16_wire_world2():
    """Multi-ring + reduced-precision wire: scratch all-reduces spread
    across rings, fused cast-back epilogue on the CPU fallback path."""
    expected = _baseline_grads(2)
    results = run_in_processes(_rings_wire_worker, 2)
    for r in range(2):
        for got, exp in zip(results[r], expected):
            assert torch.allclose(got, exp, rtol=3e-2, atol=3e-3)
        for a, b in zip(results[0], results[1]):
            assert torch.equal(a, b)
