"""Asynchronous PS training (reference BYTEPS_ENABLE_ASYNC,
torch/__init__.py:195-223 + server/server.cc:315-319,434-436):

- every parameter gets a server key ``byteps.AsyncParam.<name>``;
- rank 0 seeds the server store with the initial weights (one push — the
  server's async path sums into the store without a round barrier);
- after each local ``optimizer.step()`` the worker pushes its weight
  *delta* (w_new − w_prev) and pulls the server's current weights back
  into the parameters.  No worker ever waits for another.
"""

from __future__ import annotations

import struct
from typing import Dict, List, Tuple

import torch
import torch.distributed as dist

from .. import common as C
from ..common.naming import async_param_name, partition_key

_OP_INIT, _OP_PUSH, _OP_PULL = 5, 1, 2
_ASYNC_CMD = 1 << 16


class AsyncPSWorker:
    def __init__(self, named_params: List[Tuple[str, torch.nn.Parameter]]):
        from .ps_pipeline import _kv_client
        C._require_init()
        self.kv = _kv_client()
        self.params = [(n, p) for n, p in named_params]
        self.keys: Dict[str, Tuple[int, int]] = {}
        self.prev: Dict[str, torch.Tensor] = {}
        self.host: Dict[str, torch.Tensor] = {}
        st = C._state
        for name, p in self.params:
            full = async_param_name(name)
            dkey = st.registry.declare(full)
            pkey = partition_key(dkey, 0)
            server = st.assigner.assign(pkey, p.numel() * 4)
            self.keys[name] = (pkey, server)
            self.prev[name] = p.detach().float().clone()
            pin = p.is_cuda
            self.host[name] = torch.empty(p.numel(), dtype=torch.float32,
                                          pin_memory=pin)
        self._bootstrap()

    def _bootstrap(self) -> None:
        """Declare keys on the server; rank 0 seeds the store with w0, then
        everyone pulls so all ranks start from identical weights."""
        st = C._state
        for name, p in self.params:
            pkey, server = self.keys[name]
            payload = struct.pack("<QII", p.numel(), 1, 0)
            buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
            t = self.kv.submit(server, _OP_INIT, pkey, buf.data_ptr(),
                               len(payload), 0, 0, _ASYNC_CMD, 0)
            self.kv.wait(t)
        if dist.is_initialized() and st.size > 1:
            dist.barrier()
        if st.rank == 0:
            for name, p in self.params:
                pkey, server = self.keys[name]
                h = self.host[name]
                h.copy_(p.detach().float().reshape(-1))
                t = self.kv.submit(server, _OP_PUSH, pkey, h.data_ptr(),
                                   h.numel() * 4, 0, 0, _ASYNC_CMD, 0)
                self.kv.wait(t)
        if dist.is_initialized() and st.size > 1:
            dist.barrier()
        for name, p in self.params:
            self._pull_into(name, p)
            self.prev[name].copy_(p.detach().float())

    def _pull_into(self, name: str, p: torch.nn.Parameter) -> None:
        pkey, server = self.keys[name]
        h = self.host[name]
        t = self.kv.submit(server, _OP_PULL, pkey, 0, 0, h.data_ptr(),
                           h.numel() * 4, _ASYNC_CMD, 0)
        self.kv.wait(t)
        with torch.no_grad():
            p.reshape(-1).copy_(h.to(p.device, p.dtype, non_blocking=False))

    def exchange(self) -> None:
        """Push w_new − w_prev, pull fresh global weights."""
        tickets = []
        for name, p in self.params:
            pkey, server = self.keys[name]
            h = self.host[name]
            with torch.no_grad():
                delta = p.detach().float().reshape(-1) - \
                    self.prev[name].reshape(-1)
            h.copy_(delta)
            tickets.append(self.kv.submit(
                server, _OP_PUSH, pkey, h.data_ptr(), h.numel() * 4,
                0, 0, _ASYNC_CMD, 0))
        for t in tickets:
            self.kv.wait(t)
        for name, p in self.params:
            self._pull_into(name, p)
            self.prev[name].copy_(p.detach().float())
