"""CrossBarrier: barrier-free pipelined updates must match plain
synchronous SGD (reference cross_barrier.py semantics)."""

import torch

from mp_util import run_in_processes


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def _baseline(world, steps, lr=0.05, momentum=0.9):
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr, momentum=momentum)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = torch.cat(xs), torch.cat(ys)
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


def _cb_worker(rank, world, steps):
    import byteps_amd.torch as bps
    from byteps_amd.torch.cross_barrier import CrossBarrier
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    cb = CrossBarrier(m, opt)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = xs[rank], ys[rank]
    for _ in range(steps):
        cb.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        cb.step()
    cb.synchronize()
    out = [p.detach().clone() for p in m.parameters()]
    cb.stop()
    bps.shutdown()
    return out


def test_cross_barrier_single():
    expected = _baseline(1, 5)
    results = run_in_processes(_cb_worker, 1, 5)
    for got, exp in zip(results[0], expected):
        assert torch.allclose(got, exp, rtol=1e-5, atol=1e-6)


def test_cross_barrier_world2():
    expected = _baseline(2, 5)
    results = run_in_processes(_cb_worker, 2, 5)
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _cb_adam(rank, world, steps):
    import byteps_amd.torch as bps
    from byteps_amd.torch.cross_barrier import CrossBarrier
    bps.init()
    m = _make_model()
    opt = torch.optim.Adam(m.parameters(), lr=0.01)
    cb = CrossBarrier(m, opt)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    for _ in range(steps):
        cb.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        cb.step()
    cb.synchronize()
    out = [p.detach().clone() for p in m.parameters()]
    cb.stop()
    bps.shutdown()
    return out


def test_cross_barrier_adam_world2():
    """Any-optimizer support: per-bucket Adam instances must match the
    synchronous big-batch Adam run."""
    world, steps = 2, 5
    m = _make_model()
    opt = torch.optim.Adam(m.parameters(), lr=0.01)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = torch.cat(xs), torch.cat(ys)
    for _ in range(steps):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    expected = [p.detach().clone() for p in m.parameters()]
    results = run_in_processes(_cb_adam, world, steps)
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


# -- CrossBarrier + PS mode (ordered all-gather issuer, ADVICE A4) -----------

def _cb_ps_worker(rank, world, steps):
    import byteps_amd.torch as bps
    from byteps_amd.torch.cross_barrier import CrossBarrier
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    cb = CrossBarrier(m, opt)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    x, y = xs[rank], ys[rank]
    for _ in range(steps):
        cb.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        cb.step()
    cb.synchronize()
    out = [p.detach().clone() for p in m.parameters()]
    cb.stop()
    bps.shutdown()
    return out


def test_cross_barrier_ps_world2():
    """CrossBarrier over the PS pipeline: the poller consumes buckets in
    completion order while the issuer serializes the trailing
    all-gathers — must match synchronous SGD (the unordered version
    could deadlock/corrupt; ADVICE.md finding 4)."""
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    try:
        env = {"BPS_FORCE_DISTRIBUTED": "1",
               "BPS_SERVER_URIS": "127.0.0.1:%d" % srv.port,
               "BPS_NUM_SERVER": "1"}
        expected = _baseline(2, 4)
        results = run_in_processes(_cb_ps_worker, 2, 4, extra_env=env)
        for r in range(2):
            for got, exp in zip(results[r], expected):
                assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5)
    finally:
        srv.stop()


def test_cross_barrier_bf16_wire_world2():
    """CrossBarrier with the reduced-precision wire (BPS_COMM_DTYPE=bf16):
    per-bucket cast-back + averaging fused in _finish_bucket."""
    expected = _baseline(2, 4)
    results = run_in_processes(_cb_worker, 2, 4,
                               extra_env={"BPS_COMM_DTYPE": "bf16"})
    for r in range(2):
        for got, exp in zip(results[r], expected):
            # bf16 wire: ~3 decimal digits per step, 4 steps
            assert torch.allclose(got, exp, rtol=5e-2, atol=5e-3), \
                "bf16-wire cross-barrier diverged"
