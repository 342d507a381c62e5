"""PyTorch-ROCm plugin — the canonical byteps_amd front end.

API parity with reference byteps/torch/__init__.py: ``init``, ``shutdown``,
``suspend``, ``resume``, ``rank``, ``size``, ``local_rank``, ``local_size``,
``push_pull*``, ``poll``, ``synchronize``, ``declare``,
``DistributedOptimizer``, ``broadcast_parameters``,
``broadcast_optimizer_state``, ``broadcast_object``, ``Compression``, and
``byteps_amd.torch.parallel.DistributedDataParallel``.
"""

from __future__ import annotations

import collections
import io
import pickle
from typing import Optional

import torch
import torch.distributed as dist

from .. import common as _C
from ..common import (init, shutdown, suspend, resume, rank, size,
                      local_rank, local_size, initialized)
from ..common.telemetry import get_pushpull_speed
from .compression import Compression
from .engine import GradEngine, register_engine
from .ops import (push_pull, push_pull_async, push_pull_inplace,
                  push_pull_async_inplace, push_pull_group_sync_inplace,
                  poll, synchronize, declare, set_num_grads, BytePSPushPull)

__all__ = [
    "init", "shutdown", "suspend", "resume", "rank", "size", "local_rank",
    "local_size", "initialized", "push_pull", "push_pull_async",
    "push_pull_inplace", "push_pull_async_inplace",
    "push_pull_group_sync_inplace", "poll", "synchronize", "declare",
    "set_num_grads", "DistributedOptimizer", "broadcast_parameters",
    "broadcast_optimizer_state", "broadcast_object", "Compression",
    "BytePSPushPull", "metric_average", "get_pushpull_speed",
]


# --------------------------------------------------------------------------
# DistributedOptimizer
# --------------------------------------------------------------------------

class _DistributedOptimizer(torch.optim.Optimizer):
    """Optimizer wrapper: overlapped gradient push_pull during backward,
    synchronized before ``step()`` (reference torch/__init__.py:85-265).

    Unlike the reference — which registered one push_pull per parameter and
    copied each gradient through pinned shared memory — gradients here live
    directly in persistent flat buckets (``param.grad`` is a view) and the
    engine issues one bucketed RCCL collective (or PS push/pull) per
    partition, in priority order.
    """

    def __init__(self, params, named_parameters=None,
                 compression=Compression.none,
                 backward_passes_per_step: int = 1,
                 process_group=None,
                 compression_params=None):
        super(self.__class__, self).__init__(params)
        self._compression = compression
        self.backward_passes_per_step = backward_passes_per_step
        self._pass_count = 0

        if named_parameters is not None:
            named = [(n, p) for n, p in named_parameters if p.requires_grad]
        else:
            named = [("param.%d" % i, p)
                     for i, p in enumerate(
                         q for g in self.param_groups for q in g["params"])
                     if p.requires_grad]
        if len(named) != len({n for n, _ in named}):
            raise ValueError("named_parameters contains duplicate names")

        self._async_ps = _C.get_config().enable_async and _C._state.ps_enabled
        if self._async_ps:
            # Asynchronous PS training (reference torch/__init__.py:195-223):
            # no gradient sync — workers push weight deltas and pull back
            # the server's running weights after every local step.
            from .async_ps import AsyncPSWorker
            self._engine = None
            self._async_worker = AsyncPSWorker(named)
            return

        # Compression.fp16/bf16 narrows only the WIRE (the all-reduce
        # scratch), never the gradient accumulation dtype — reference
        # torch/compression.py:34-76 compressed at enqueue and
        # decompressed at synchronize, leaving autograd untouched.
        comm_dtype = (torch.bfloat16 if compression is Compression.bf16
                      else torch.float16 if compression is Compression.fp16
                      else None)
        self._engine = register_engine(GradEngine(
            named, process_group=process_group, comm_dtype=comm_dtype,
            compression_params=compression_params))
        if backward_passes_per_step > 1:
            self._engine.set_sync_enabled(False)

    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002
        if self._engine is None:
            return super(self.__class__, self).zero_grad(set_to_none=False)
        # grads are views into persistent buckets — zero the buckets instead
        # of detaching (set_to_none would break the aliasing)
        self._engine.zero_grad()

    def synchronize(self):
        if self._engine is not None:
            self._engine.synchronize()

    def step(self, closure=None):
        if self._engine is None:      # async PS mode
            loss = super(self.__class__, self).step(closure)
            self._async_worker.exchange()
            return loss
        self._pass_count += 1
        if self._pass_count < self.backward_passes_per_step:
            # accumulate only — no sync, no step
            return None
        self._pass_count = 0
        if self.backward_passes_per_step > 1:
            # one full accumulation window done: sync the accumulated grads.
            # Split params accumulated into their PRIVATE grad while sync
            # was off — refresh the bucket views first, or the flush would
            # all-reduce the stale spans.
            self._engine.set_sync_enabled(True)
            self._engine.sync_split_grads()
            self._engine.flush()
        self._engine.synchronize()
        loss = super(self.__class__, self).step(closure)
        if self.backward_passes_per_step > 1:
            self._engine.set_sync_enabled(False)
            self._engine.zero_grad()
        return loss


def DistributedOptimizer(optimizer, named_parameters=None,
                         compression=Compression.none,
                         backward_passes_per_step: int = 1,
                         process_group=None,
                         compression_params=None):
    """Wrap ``optimizer`` for distributed training
    (reference torch/__init__.py:226-265 dynamic subclassing pattern).
    ``compression_params`` selects a codec for the PS path, e.g.
    ``{"compressor_type": "onebit", "ef_type": "vanilla"}`` (reference
    mxnet compression_params, mxnet/__init__.py:250-317)."""
    cls = type(optimizer.__class__.__name__, (optimizer.__class__,),
               dict(_DistributedOptimizer.__dict__))
    obj = cls.__new__(cls)
    obj.__dict__.update(optimizer.__dict__)
    _DistributedOptimizer.__init__(
        obj, obj.param_groups, named_parameters, compression,
        backward_passes_per_step, process_group, compression_params)
    return obj


# --------------------------------------------------------------------------
# Broadcasts
# --------------------------------------------------------------------------

def metric_average(value, name: str):
    """Average a scalar metric across all ranks (reference keras
    MetricAverageCallback, _keras/callbacks.py:62-86)."""
    _C._require_init()
    t = value.detach().clone().float() if torch.is_tensor(value) \
        else torch.tensor(float(value))
    avg = push_pull(t, average=True, name="byteps.Metric." + name)
    return avg.item() if avg.numel() == 1 else avg


def broadcast_parameters(params, root_rank: int = 0) -> None:
    """Broadcast a ``state_dict()`` or list of (name, tensor) pairs from
    ``root_rank`` (reference torch/__init__.py:268-299 — which emulated
    broadcast as zero+push_pull because ps-lite had no broadcast; RCCL has
    a native one)."""
    _C._require_init()
    if isinstance(params, dict):
        items = sorted(params.items())
    else:
        items = list(params)
    if _C.size() <= 1 or not dist.is_initialized():
        return
    handles = []
    for _name, t in items:
        if t is None or not torch.is_tensor(t):
            continue
        handles.append(dist.broadcast(t.data, src=root_rank, async_op=True))
    for h in handles:
        h.wait()


def broadcast_object(obj, root_rank: int = 0, name: Optional[str] = None):
    """Pickle-broadcast an arbitrary object (reference
    torch/__init__.py:426-466: cloudpickle through a ByteTensor push_pull)."""
    _C._require_init()
    if _C.size() <= 1 or not dist.is_initialized():
        return obj
    dev = _C.device() if torch.cuda.is_available() else torch.device("cpu")
    if _C.rank() == root_rank:
        buf = io.BytesIO()
        pickle.dump(obj, buf, protocol=pickle.HIGHEST_PROTOCOL)
        data = torch.frombuffer(
            bytearray(buf.getvalue()), dtype=torch.uint8).to(dev)
        length = torch.tensor([data.numel()], dtype=torch.long, device=dev)
    else:
        length = torch.zeros(1, dtype=torch.long, device=dev)
    dist.broadcast(length, src=root_rank)
    if _C.rank() != root_rank:
        data = torch.empty(int(length.item()), dtype=torch.uint8, device=dev)
    dist.broadcast(data, src=root_rank)
    if _C.rank() == root_rank:
        return obj
    return pickle.loads(data.cpu().numpy().tobytes())


def broadcast_optimizer_state(optimizer, root_rank: int = 0) -> None:
    """Broadcast optimizer state from root (reference
    torch/__init__.py:302-424): tensor state entries are broadcast
    in-place; scalar entries (e.g. ``step``) travel pickled."""
    _C._require_init()
    if _C.size() <= 1 or not dist.is_initialized():
        return

    # LBFGS keeps state in lists — unsupported, as in the reference
    if optimizer.__class__.__name__ == "LBFGS":
        raise ValueError("cannot broadcast torch.optim.LBFGS state")

    state_dict = optimizer.state_dict()
    # Lazy optimizer state must exist CONSISTENTLY on every rank before
    # the tensor broadcast (mismatched per-rank tensor lists deadlock).
    # The decision is driven by the ROOT's state: if the root has none
    # (fresh optimizer), every rank materializes it with a dummy step
    # (reference's lazy-creation idea, torch/__init__.py:318-335, made
    # symmetric).  The dummy step bypasses our synchronizing wrapper and
    # restores params afterwards (weight decay would perturb them).
    root_has_state = broadcast_object(
        bool(state_dict["state"]), root_rank, name="byteps.opt_has_state")
    if not root_has_state or not state_dict["state"]:
        for group in optimizer.param_groups:
            for p in group["params"]:
                if p.requires_grad and p.grad is None:
                    p.grad = torch.zeros_like(p)
        if any(g["params"] for g in optimizer.param_groups):
            snap = [p.detach().clone() for g in optimizer.param_groups
                    for p in g["params"]]
            if hasattr(optimizer, "_engine"):
                type(optimizer).__mro__[1].step(optimizer)
            else:
                optimizer.step()
            with torch.no_grad():
                for p, s in zip((p for g in optimizer.param_groups
                                 for p in g["params"]), snap):
                    p.copy_(s)
            state_dict = optimizer.state_dict()

    scalars = collections.OrderedDict()
    tensors = []
    for sid, pstate in sorted(state_dict["state"].items()):
        for k, v in sorted(pstate.items()):
            if torch.is_tensor(v) and v.numel() > 0:
                tensors.append(("%s.%s" % (sid, k), v))
            else:
                scalars["%s.%s" % (sid, k)] = v
    scalars["__param_groups__"] = [
        {k: v for k, v in g.items() if k != "params"}
        for g in state_dict["param_groups"]]

    scalars = broadcast_object(scalars, root_rank)
    broadcast_parameters(tensors, root_rank)

    if _C.rank() != root_rank:
        groups = scalars.pop("__param_groups__")
        for g, meta in zip(state_dict["param_groups"], groups):
            g.update(meta)
        for key, val in scalars.items():
            sid, k = key.split(".", 1)
            sid = int(sid) if sid.isdigit() else sid
            if sid in state_dict["state"]:
                state_dict["state"][sid][k] = val
        optimizer.load_state_dict(state_dict)
