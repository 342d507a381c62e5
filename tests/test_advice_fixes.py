"""Regression tests for the round-1 advisor findings:

- A1: params split across buckets train correctly under CrossBarrier
  (averaged shards reach p.grad; stepped exactly once).
- A2: Compression.fp16/bf16 is WIRE-only — fp32 models construct fine and
  accumulate at full precision (reference torch/compression.py:34-76).
- A3: backward_passes_per_step>1 with a split param keeps the accumulated
  gradient (the flush must see the private grad, not stale zeros).
- A5: server rejects a re-init whose nelem/codec differ.
"""

import torch

from mp_util import run_in_processes


def _split_model(seed=0):
    # Linear(2048,10): weight = 20480 elems > the 4096-elem bucket floor,
    # so it SPLITS across buckets at partition_bytes=16384
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(32, 2048), torch.nn.ReLU(),
        torch.nn.Linear(2048, 10))


PART_BYTES = 16384  # → 4096-elem buckets


def _data(world, batch=6):
    torch.manual_seed(42)
    xs = [torch.randn(batch, 32) for _ in range(world)]
    ys = [torch.randn(batch, 10) for _ in range(world)]
    return xs, ys


def _baseline(world, steps, passes=1, lr=0.05, momentum=0.9):
    m = _split_model()
    opt = torch.optim.SGD(m.parameters(), lr=lr, momentum=momentum)
    xs, ys = _data(world)
    for _ in range(steps):
        opt.zero_grad()
        loss = 0.0
        for _ in range(passes):
            loss = loss + sum(((m(x) - y) ** 2).mean()
                              for x, y in zip(xs, ys)) / world
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in m.parameters()]


# -- A1: CrossBarrier + split params ----------------------------------------

def _cb_split_worker(rank, world, steps):
    import byteps_amd.torch as bps
    from byteps_amd.torch.cross_barrier import CrossBarrier
    bps.init()
    m = _split_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    cb = CrossBarrier(m, opt, partition_bytes=PART_BYTES)
    assert cb._engine._split_params, "test needs a split param"
    xs, ys = _data(world)
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


def test_cross_barrier_split_param_world2():
    expected = _baseline(2, 4)
    results = run_in_processes(_cb_split_worker, 2, 4)
    for r in range(2):
        for got, exp in zip(results[r], expected):
            assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5), \
                "split-param CrossBarrier diverged from synchronous SGD"


# -- A2: wire-only Compression.fp16/bf16 ------------------------------------

def _wire_comp_worker(rank, world, kind):
    import byteps_amd.torch as bps
    bps.init()
    m = _split_model()
    comp = bps.Compression.bf16 if kind == "bf16" else bps.Compression.fp16
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters(), compression=comp)
    xs, ys = _data(world)
    ((m(xs[rank]) - ys[rank]) ** 2).mean().backward()
    opt.synchronize()
    # gradients must still be fp32 (wire-only narrowing)
    dtypes = {p.grad.dtype for p in m.parameters()}
    grads = [p.grad.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return dtypes, grads


def test_compression_bf16_is_wire_only_world2():
    results = run_in_processes(_wire_comp_worker, 2, "bf16")
    dtypes, grads = results[0]
    assert dtypes == {torch.float32}
    # the averaged grads must agree across ranks and be close to the
    # full-precision average (bf16 wire: ~3 decimal digits)
    m = _split_model()
    xs, ys = _data(2)
    loss = sum(((m(x) - y) ** 2).mean() for x, y in zip(xs, ys)) / 2
    loss.backward()
    for got, p in zip(grads, m.parameters()):
        assert torch.allclose(got, p.grad, rtol=2e-2, atol=2e-3)
    for a, b in zip(*[r[1] for r in results]):
        assert torch.equal(a, b)


def test_compression_fp16_constructs_on_fp32_model():
    # single-process: the advertised API must not crash at construction
    results = run_in_processes(_wire_comp_worker, 1, "fp16")
    dtypes, _ = results[0]
    assert dtypes == {torch.float32}


# -- A3: accumulation + split params ----------------------------------------

def _accum_split_worker(rank, world, steps):
    import byteps_amd.torch as bps
    bps.init()
    m = _split_model()
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9),
        named_parameters=m.named_parameters(),
        backward_passes_per_step=2)
    assert opt._engine._split_params, "test needs a split param"
    xs, ys = _data(world)
    x, y = xs[rank], ys[rank]
    for _ in range(steps):
        for _ in range(2):
            ((m(x) - y) ** 2).mean().backward()
            opt.step()          # first call accumulates, second syncs
        opt.zero_grad()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_accumulation_split_param_world2():
    expected = _baseline(2, 3, passes=2)
    results = run_in_processes(
        _accum_split_worker, 2, 3,
        extra_env={"BPS_PARTITION_BYTES": str(PART_BYTES)})
    for got, exp in zip(results[0], expected):
        assert torch.allclose(got, exp, rtol=1e-4, atol=1e-5), \
            "accumulated split-param grads were lost"


# -- A5: server re-init validation ------------------------------------------

def test_server_rejects_mismatched_reinit():
    import struct
    from byteps_amd.ops import core
    c = core()
    server = c.Server(0, 1, False)
    server.start()
    try:
        kv = c.KVClient(0, ["127.0.0.1:%d" % server.port])
        payload = struct.pack("<QIIII", 1024, 1, 0, 0, 0)
        buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
        t = kv.submit(0, 5, 7, buf.data_ptr(), len(payload), 0, 0, 0, 0)
        _len, aux = kv.wait(t)
        assert aux != 0xFFFFFFFFFFFFFFFF
        # same key, different nelem → error marker
        payload2 = struct.pack("<QIIII", 2048, 1, 0, 0, 0)
        buf2 = torch.frombuffer(bytearray(payload2), dtype=torch.uint8)
        t = kv.submit(0, 5, 7, buf2.data_ptr(), len(payload2), 0, 0, 0, 0)
        _len, aux = kv.wait(t)
        assert aux == 0xFFFFFFFFFFFFFFFF, "mismatched re-init not rejected"
        # matching re-init with a new world is adopted
        payload3 = struct.pack("<QIIII", 1024, 3, 0, 0, 0)
        buf3 = torch.frombuffer(bytearray(payload3), dtype=torch.uint8)
        t = kv.submit(0, 5, 7, buf3.data_ptr(), len(payload3), 0, 0, 0, 0)
        _len, aux = kv.wait(t)
        assert aux != 0xFFFFFFFFFFFFFFFF
        kv.close()
    finally:
        server.stop()
