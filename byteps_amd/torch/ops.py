"""Functional ``push_pull`` API with integer handles.

API parity with the reference (byteps/torch/ops.py:51-237 and the pybind
handle manager, torch/handle_manager.cc:22-52):  ``push_pull``,
``push_pull_async``, ``push_pull_inplace``, ``push_pull_async_inplace``,
``push_pull_group_sync_inplace``, ``poll``, ``synchronize``, ``declare``.

``push_pull`` of a tensor is semantically an all-reduce SUM (optionally
averaged).  Single node → one RCCL all-reduce over xGMI.  PS mode →
reduce-scatter + KV push/pull + all-gather (the hierarchical route of the
reference, common/operations.cc:429-485).
"""

from __future__ import annotations

import threading
from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

from .. import common as C
from ..common.naming import gradient_name

_lock = threading.Lock()
_handles: Dict[int, Tuple] = {}
_next_handle = [0]
_grad_count = [0]          # for push_pull_group_sync_inplace
_num_grads = [0]
_name_serial: Dict[str, int] = {}


def _alloc_handle(entry: Tuple) -> int:
    with _lock:
        h = _next_handle[0]
        _next_handle[0] += 1
        _handles[h] = entry
    return h


def _auto_name(tensor: torch.Tensor, name: Optional[str]) -> str:
    if name is not None:
        return name
    # reference auto-names unnamed tensors by serial number
    # (torch/ops.cc:104: "byteps.noname.<n>")
    key = "byteps.noname"
    with _lock:
        n = _name_serial.get(key, 0)
        _name_serial[key] = n + 1
    return "%s.%d" % (key, n)


def declare(name: str) -> int:
    """Pre-declare a tensor name (reference byteps_torch_declare_tensor)."""
    C._require_init()
    return C._state.registry.declare(name)


def _start(tensor: torch.Tensor, average: bool, name: str,
           version: int, priority: int) -> int:
    C._require_init()
    world = C.size()
    if world <= 1 and not C._state.ps_enabled:
        work = None
    elif C._state.ps_enabled:
        from . import ps_pipeline
        work = ps_pipeline.get_tensor_pipeline().submit_tensor(
            tensor, name, priority)
    else:
        work = dist.all_reduce(tensor, op=dist.ReduceOp.SUM, async_op=True)
    return _alloc_handle((work, tensor, average, world))


def push_pull_async_inplace(tensor: torch.Tensor, average: bool = True,
                            name: Optional[str] = None, version: int = 0,
                            priority: int = 0) -> int:
    name = _auto_name(tensor, name)
    declare(name)
    return _start(tensor, average, name, version, priority)


def push_pull_async(tensor: torch.Tensor, average: bool = True,
                    name: Optional[str] = None, version: int = 0,
                    priority: int = 0) -> int:
    out = tensor.detach().clone()
    return push_pull_async_inplace(out, average, name, version, priority)


def push_pull(tensor: torch.Tensor, average: bool = True,
              name: Optional[str] = None, version: int = 0,
              priority: int = 0,
              compression=None) -> torch.Tensor:
    """Synchronous out-of-place push_pull (reference torch/ops.py:80-107).
    ``compression`` is an intra-node wire compressor from
    :mod:`byteps_amd.torch.compression` (none/fp16)."""
    from .compression import Compression
    comp = compression or Compression.none
    compressed, ctx = comp.compress(tensor)
    handle = push_pull_async(compressed, average, name, version, priority)
    out = synchronize(handle)
    return comp.decompress(out, ctx)


def push_pull_inplace(tensor: torch.Tensor, average: bool = True,
                      name: Optional[str] = None, version: int = 0,
                      priority: int = 0) -> torch.Tensor:
    handle = push_pull_async_inplace(tensor, average, name, version, priority)
    return synchronize(handle)


def set_num_grads(n: int) -> None:
    """Reference byteps_torch_set_num_grads (torch/ops.cc:137-166)."""
    _num_grads[0] = n
    _grad_count[0] = 0


def push_pull_group_sync_inplace(tensor: torch.Tensor, average: bool = True,
                                 name: Optional[str] = None, version: int = 0,
                                 priority: int = 0) -> Tuple[int, int]:
    """DDP variant: returns (handle, current grad count); the caller
    self-synchronizes when count reaches ``set_num_grads`` value."""
    h = push_pull_async_inplace(tensor, average, name, version, priority)
    with _lock:
        _grad_count[0] += 1
        cnt = _grad_count[0]
        if cnt == _num_grads[0]:
            _grad_count[0] = 0
    return h, cnt


def poll(handle: int) -> bool:
    with _lock:
        entry = _handles.get(handle)
    if entry is None:
        return True
    work = entry[0]
    if work is None:
        return True
    if hasattr(work, "is_completed"):
        return work.is_completed()
    return work.done()


def synchronize(handle: int) -> torch.Tensor:
    with _lock:
        entry = _handles.pop(handle, None)
    if entry is None:
        raise ValueError("invalid byteps_amd handle %r" % handle)
    work, tensor, average, world = entry
    if work is not None:
        if hasattr(work, "wait"):
            work.wait()
        else:
            work.result()
    if average and world > 1:
        tensor.div_(world)
    return tensor


# reference torch/ops.py names it `synchronize`; `wait_and_clear` is the
# C-side primitive (torch/ops.cc:129-135)
wait_and_clear = synchronize


class BytePSPushPull(torch.autograd.Function):
    """Differentiable push_pull (reference torch/ops.py:109-124)."""

    @staticmethod
    def forward(ctx, tensor, average, name):
        ctx.average = average
        ctx.name = name
        return push_pull(tensor, average, name)

    @staticmethod
    def backward(ctx, grad_output):
        return push_pull(grad_output, ctx.average,
                         gradient_name(ctx.name)), None, None
