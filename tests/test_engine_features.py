"""Engine feature coverage: gradient accumulation, wire compression,
credits, autograd push_pull, split params (tensor > partition)."""

import torch

from mp_util import run_in_processes


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def _accum(rank, world, bpps):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    opt = bps.DistributedOptimizer(
        opt, named_parameters=m.named_parameters(),
        backward_passes_per_step=bpps)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world * bpps)]
    ys = [torch.randn(8, 4) for _ in range(world * bpps)]
    for i in range(bpps):
        loss = ((m(xs[rank * bpps + i]) - ys[rank * bpps + i]) ** 2).mean()
        loss.backward()
        opt.step()        # only the bpps-th call syncs + steps
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_backward_passes_per_step():
    world, bpps = 2, 2
    # baseline: one step on the sum of all 4 micro-batch grads / world
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world * bpps)]
    ys = [torch.randn(8, 4) for _ in range(world * bpps)]
    opt.zero_grad()
    for i in range(world * bpps):
        (((m(xs[i]) - ys[i]) ** 2).mean() / world).backward()
    opt.step()
    expected = [p.detach().clone() for p in m.parameters()]

    results = run_in_processes(_accum, world, bpps)
    for got in results:
        for p_got, p_exp in zip(got, expected):
            assert torch.allclose(p_got, p_exp, rtol=1e-5, atol=1e-6)


def _wire_fp16(rank, world):
    import byteps_amd.torch as bps
    from byteps_amd.torch.compression import Compression
    bps.init()
    t = torch.full((64,), 1.5) * (rank + 1)
    out = bps.push_pull(t, average=False, name="w16",
                        compression=Compression.fp16)
    expect = 1.5 * sum(range(1, world + 1))
    ok = torch.allclose(out, torch.full((64,), expect)) \
        and out.dtype == torch.float32
    bps.shutdown()
    return bool(ok)


def test_wire_compression_fp16():
    assert all(run_in_processes(_wire_fp16, 2))


def _autograd_pp(rank, world):
    import byteps_amd.torch as bps
    from byteps_amd.torch import BytePSPushPull
    bps.init()
    x = torch.ones(8, requires_grad=True)
    y = BytePSPushPull.apply(x * (rank + 1), True, "ag")
    y.sum().backward()
    # d(avg over ranks of (rank+1)x)/dx then averaged again in backward
    ok = x.grad is not None and torch.isfinite(x.grad).all()
    bps.shutdown()
    return bool(ok)


def test_autograd_push_pull():
    assert all(run_in_processes(_autograd_pp, 2))


def _credit(rank, world):
    import byteps_amd.torch as bps
    from byteps_amd.torch.parallel import DistributedDataParallel as DDP
    bps.init()
    m = _make_model()
    net = DDP(m, partition_bytes=8192)
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    torch.manual_seed(42)
    x, y = torch.randn(8, 16), torch.randn(8, 4)
    for _ in range(3):
        net.zero_grad_buckets()
        ((net(x) - y) ** 2).mean().backward()
        opt.step()
    ok = all(torch.isfinite(p).all() for p in m.parameters())
    bps.shutdown()
    return bool(ok)


def test_scheduling_credit():
    """Byte-credit limits in-flight buckets; training must still complete
    (credits only defer, never drop)."""
    assert all(run_in_processes(_credit, 2,
                                extra_env={"BPS_SCHEDULING_CREDIT": "8192"}))


def _split_param(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    torch.manual_seed(0)
    # one param far larger than the partition size → spans buckets
    m = torch.nn.Linear(512, 512)   # 262k params > 8k-elem partitions
    opt = torch.optim.SGD(m.parameters(), lr=0.01)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    torch.manual_seed(7)
    x, y = torch.randn(16, 512), torch.randn(16, 512)
    for _ in range(2):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_param_split_across_partitions():
    results = run_in_processes(
        _split_param, 2, extra_env={"BPS_PARTITION_BYTES": "32768"})
    for p0, p1 in zip(*results):
        assert torch.allclose(p0, p1, rtol=1e-5, atol=1e-6)


def _comm_bf16(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    m = _make_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    opt = bps.DistributedOptimizer(opt, named_parameters=m.named_parameters())
    torch.manual_seed(42)
    xs = [torch.randn(8, 16) for _ in range(world)]
    ys = [torch.randn(8, 4) for _ in range(world)]
    for _ in range(3):
        opt.zero_grad()
        ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_comm_dtype_bf16_wire():
    """BPS_COMM_DTYPE=bf16: collectives run on a bf16 scratch; results
    must match fp32 within bf16 tolerance and agree across ranks."""
    world = 2
    results = run_in_processes(_comm_bf16, world,
                               extra_env={"BPS_COMM_DTYPE": "bf16"})
    ref = run_in_processes(_comm_bf16, world)
    for p_a, p_b in zip(*results):
        assert torch.allclose(p_a, p_b)     # ranks agree exactly
    for p_got, p_ref in zip(results[0], ref[0]):
        assert torch.allclose(p_got, p_ref, rtol=3e-2, atol=3e-2)
