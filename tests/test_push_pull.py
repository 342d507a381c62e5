"""push_pull correctness over gloo, world_size 2 (CPU) — the reference's
sum tests across shapes/dtypes (tests/test_mxnet.py:60-122)."""

import pytest
import torch

from mp_util import run_in_processes


def _pp_sum(rank, world, shape, dtype_name):
    import byteps_amd.torch as bps
    bps.init()
    dtype = getattr(torch, dtype_name)
    t = (torch.arange(int(torch.tensor(shape).prod()))
         .reshape(shape).to(dtype) + rank)
    out = bps.push_pull(t, average=False, name="t.%s.%s" % (shape, dtype_name))
    expect = sum((torch.arange(int(torch.tensor(shape).prod()))
                  .reshape(shape).to(dtype) + r) for r in range(world))
    ok = torch.allclose(out.float(), expect.float(), rtol=1e-5, atol=1e-5)
    bps.shutdown()
    return bool(ok)


@pytest.mark.parametrize("shape", [(17,), (4, 5), (2, 3, 4)])
@pytest.mark.parametrize("dtype_name", ["float32", "float64", "int64"])
def test_push_pull_sum(shape, dtype_name):
    assert all(run_in_processes(_pp_sum, 2, shape, dtype_name))


def _pp_avg(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    t = torch.full((100,), float(rank + 1))
    out = bps.push_pull(t, average=True, name="avg")
    expect = sum(range(1, world + 1)) / world
    ok = torch.allclose(out, torch.full((100,), expect))
    bps.shutdown()
    return bool(ok)


def test_push_pull_average():
    assert all(run_in_processes(_pp_avg, 2))


def _pp_inplace(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    t = torch.ones(33) * (rank + 1)
    handle = bps.push_pull_async_inplace(t, average=False, name="ip")
    done = bps.poll(handle)  # may or may not be done; must not raise
    assert done in (True, False)
    bps.synchronize(handle)
    ok = torch.allclose(t, torch.full((33,), float(sum(range(1, world + 1)))))
    bps.shutdown()
    return bool(ok)


def test_push_pull_inplace_async():
    assert all(run_in_processes(_pp_inplace, 2))


def _bcast(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    params = {"a": torch.full((8,), float(rank)),
              "b": torch.full((3, 3), float(rank * 10))}
    bps.broadcast_parameters(params, root_rank=0)
    ok = (torch.allclose(params["a"], torch.zeros(8)) and
          torch.allclose(params["b"], torch.zeros(3, 3)))
    obj = {"lr": 0.1, "step": 7} if rank == 0 else None
    obj = bps.broadcast_object(obj, root_rank=0)
    ok = ok and obj == {"lr": 0.1, "step": 7}
    bps.shutdown()
    return bool(ok)


def test_broadcast_parameters_and_object():
    assert all(run_in_processes(_bcast, 2))


def _group_sync(rank, world):
    import byteps_amd.torch as bps
    bps.init()
    bps.set_num_grads(3)
    handles = []
    counts = []
    for i in range(3):
        t = torch.ones(10) * (rank + 1)
        h, c = bps.push_pull_group_sync_inplace(t, average=False, name="g%d" % i)
        handles.append((h, t))
        counts.append(c)
    assert counts == [1, 2, 3]
    for h, t in handles:
        bps.synchronize(h)
        assert torch.allclose(t, torch.full((10,), float(sum(range(1, world + 1)))))
    bps.shutdown()
    return True


def test_group_sync():
    assert all(run_in_processes(_group_sync, 2))
