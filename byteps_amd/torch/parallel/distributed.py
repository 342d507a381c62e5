"""DistributedDataParallel over the byteps_amd gradient engine.

Parity with reference byteps/torch/parallel/distributed.py: module state
broadcast at construction (:179-182), per-forward buffer sync (:209-220),
gradient group-sync during backward with self-synchronization when the
last gradient arrives (:261-287), and a ``no_sync()`` context for gradient
accumulation (:184-207).

MI355X-native differences: gradients are zero-copy views into persistent
flat buckets, synchronized by bucketed RCCL all-reduce over xGMI issued in
backward order (highest priority first) — not per-parameter push_pull
through CPU shared memory.
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.distributed as dist

from ... import common as _C
from ..engine import GradEngine, register_engine


class DistributedDataParallel(torch.nn.Module):
    def __init__(self, module: torch.nn.Module, device_ids=None,
                 broadcast_buffers: bool = True,
                 process_group=None,
                 partition_bytes: Optional[int] = None,
                 compression_params: Optional[dict] = None):
        super().__init__()
        _C._require_init()
        self.module = module
        self.broadcast_buffers = broadcast_buffers
        self.require_backward_grad_sync = True
        self._group = process_group

        if device_ids is not None and len(device_ids) > 1:
            raise ValueError(
                "byteps_amd DDP is one process per GPU; pass a single "
                "device id (reference parallel/distributed.py:122-141)")

        named = list(module.named_parameters())
        self._engine = register_engine(GradEngine(
            named, process_group=process_group,
            partition_bytes=partition_bytes,
            compression_params=compression_params))
        self._engine.on_all_ready = self._self_synchronize

        # broadcast initial model state from rank 0 so replicas agree
        if _C.size() > 1 and dist.is_initialized():
            from .. import broadcast_parameters
            broadcast_parameters(
                [(n, p) for n, p in named], root_rank=0)
            buffers = list(module.named_buffers())
            if buffers:
                broadcast_parameters(buffers, root_rank=0)

    # -- forward -----------------------------------------------------------

    def forward(self, *args, **kwargs):
        if (self.broadcast_buffers and _C.size() > 1
                and dist.is_initialized()):
            bufs = [(n, b) for n, b in self.module.named_buffers()]
            if bufs:
                from .. import broadcast_parameters
                broadcast_parameters(bufs, root_rank=0)
        return self.module(*args, **kwargs)

    # -- backward sync ------------------------------------------------------

    def _self_synchronize(self) -> None:
        if self.require_backward_grad_sync:
            self._engine.synchronize()

    @contextlib.contextmanager
    def no_sync(self):
        """Skip gradient synchronization inside the context (gradient
        accumulation).  Grads still accumulate into the flat buckets."""
        old = self.require_backward_grad_sync
        self.require_backward_grad_sync = False
        self._engine.set_sync_enabled(False)
        try:
            yield
        finally:
            self.require_backward_grad_sync = old
            self._engine.set_sync_enabled(old)

    def zero_grad_buckets(self) -> None:
        self._engine.zero_grad()

    # passthroughs ----------------------------------------------------------

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)
