"""DistributedOptimizer / DistributedDataParallel parity tests over gloo
world_size 2: the distributed run must match a single-process run on the
concatenated batch (gradient averaging semantics)."""

import pytest
import torch

from mp_util import run_in_processes


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 4))


def _data(world):
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    return xs, ys


def _baseline_params(world, steps, lr=0.05, momentum=0.9):
    """Single-process reference: same model, batch = concat of all ranks'."""
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr, momentum=momentum)
    xs, ys = _data(world)
    x, y = torch.cat(xs), torch.cat(ys)
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


def _dist_opt(rank, world, steps, use_ddp, partition_bytes):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    xs, ys = _data(world)
    x, y = xs[rank], ys[rank]
    if use_ddp:
        from byteps_amd.torch.parallel import DistributedDataParallel as DDP
        net = DDP(m, partition_bytes=partition_bytes)
        opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
        for _ in range(steps):
            net.zero_grad_buckets()
            ((net(x) - y) ** 2).mean().backward()
            opt.step()
    else:
        opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
        opt = bps.DistributedOptimizer(
            opt, named_parameters=m.named_parameters())
        for _ in range(steps):
            opt.zero_grad()
            ((m(x) - y) ** 2).mean().backward()
            opt.step()
    result = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return result


@pytest.mark.parametrize("use_ddp", [False, True])
def test_distributed_matches_large_batch(use_ddp):
    world, steps = 2, 4
    expected = _baseline_params(world, steps)
    results = run_in_processes(_dist_opt, world, steps, use_ddp, None)
    for got in results:
        for p_got, p_exp in zip(got, expected):
            # MSE over concat batch == mean of per-rank MSEs (equal sizes),
            # so averaged grads match the big-batch grads exactly
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def test_ddp_small_buckets():
    """Force many buckets (tiny partition size) — exercises priority
    scheduling + multi-bucket pipeline."""
    world, steps = 2, 3
    expected = _baseline_params(world, steps)
    results = run_in_processes(_dist_opt, world, steps, True, 4096 * 2)
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _no_sync(rank, world):
    import byteps_amd.torch as bps
    from byteps_amd.torch.parallel import DistributedDataParallel as DDP
    bps.init()
    m = _make_model()
    net = DDP(m)
    xs, ys = _data(world)
    x, y = xs[rank], ys[rank]
    net.zero_grad_buckets()
    with net.no_sync():
        ((net(x) - y) ** 2).mean().backward()   # accumulate only
    ((net(x) - y) ** 2).mean().backward()       # second pass syncs
    g0 = m[0].weight.grad.detach().clone()
    bps.shutdown()
    return g0


def test_ddp_no_sync_accumulates():
    world = 2
    results = run_in_processes(_no_sync, world)
    # ranks must agree after the synced pass
    assert torch.allclose(results[0], results[1], rtol=1e-5, atol=1e-6)
    # and equal 2 * average of single-pass grads
    m = _make_model()
    xs, ys = _data(world)
    grads = []
    for r in range(world):
        m.zero_grad()
        ((m(xs[r]) - ys[r]) ** 2).mean().backward()
        grads.append(m[0].weight.grad.detach().clone())
    expect = 2 * sum(grads) / world
    assert torch.allclose(results[0], expect, rtol=1e-5, atol=1e-6)


def _bcast_opt_state(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model(seed=rank)   # deliberately diverged
    opt = torch.optim.Adam(m.parameters(), lr=0.01)
    if rank == 0:
        # build real state on root
        ((m(torch.randn(4, 16)) - torch.randn(4, 4)) ** 2).mean().backward()
        opt.step()
    bps.broadcast_parameters(m.state_dict(), root_rank=0)
    bps.broadcast_optimizer_state(opt, root_rank=0)
    sd = opt.state_dict()
    steps = [v["step"] for v in sd["state"].values()] if sd["state"] else []
    w = m[0].weight.detach().clone()
    bps.shutdown()
    return (w, steps)


def test_broadcast_optimizer_state():
    results = run_in_processes(_bcast_opt_state, 2)
    (w0, s0), (w1, s1) = results
    assert torch.allclose(w0, w1)
    assert len(s0) == len(s1)
    for a, b in zip(s0, s1):
        assert float(a) == float(b)


# -- buffer broadcast (BN running stats sync each forward) -------------------

def _buffers_worker(rank, world):
    import byteps_amd.torch as bps
    from byteps_amd.torch.parallel import DistributedDataParallel as DDP
    bps.init()
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(8, 8),
                            torch.nn.BatchNorm1d(8))
    net = DDP(m, broadcast_buffers=True)
    # desync buffers on non-root, then one forward must re-sync them
    if rank != 0:
        with torch.no_grad():
            m[1].running_mean.fill_(42.0)
    torch.manual_seed(100 + rank)
    net(torch.randn(4, 8))      # train forward: desync is re-broadcast,
                                # then BN re-updates stats per-rank
    m.eval()                    # eval: no stat update after the sync
    net(torch.randn(4, 8))
    out = [b.detach().clone() for b in m.buffers()]
    bps.shutdown()
    return out


def test_ddp_broadcast_buffers_each_forward():
    """DDP re-broadcasts buffers from rank 0 before each forward
    (reference parallel/distributed.py:209-220) — a desynced running
    stat on rank 1 must be overwritten."""
    results = run_in_processes(_buffers_worker, 2)
    for b0, b1 in zip(results[0], results[1]):
        assert torch.equal(b0, b1), "buffers diverged across ranks"
    assert not torch.any(results[1][0] == 42.0), "rank-1 buffer not resynced"
