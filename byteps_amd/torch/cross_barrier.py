"""Cross-barrier training: remove the per-iteration global barrier.

Re-creation of the reference's CrossBarrier (byteps/torch/
cross_barrier.py): instead of `synchronize-all → step-all`, each bucket's
optimizer update is applied by a poller thread as soon as that bucket's
communication completes, and the *next* iteration's forward blocks
per-submodule until the parameters it touches are updated
(reference :188-222 forward pre-hook locks, :159-186 poller).

MI355X-native differences: the unit of pipelining is a bucket (one RCCL
collective / PS round-trip), not a single parameter; and instead of
re-implementing SGD/Adam/RMSProp by hand (reference :236-381), each
bucket gets its own instance of the *user's* optimizer class over just
its parameters — any torch optimizer works, and per-bucket stepping from
the poller thread shares no state with autograd.
"""

from __future__ import annotations

import queue
import threading
from typing import Dict, List, Optional

import torch

from ..common.logging_util import get_logger
from .engine import Bucket, GradEngine, register_engine

log = get_logger()


class CrossBarrier:
    """Wrap (model, optimizer) for barrier-free training::

        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        cb = bps.CrossBarrier(model, opt, model.named_parameters())
        for x, y in data:
            cb.zero_grad()
            loss_fn(model(x), y).backward()
            cb.step()          # returns immediately; updates pipeline
    """

    def __init__(self, model: torch.nn.Module, optimizer,
                 named_parameters=None, process_group=None,
                 partition_bytes: Optional[int] = None,
                 auto_zero: bool = True):
        self.model = model
        self.optimizer = optimizer
        # auto_zero: the poller zeroes each bucket right after its
        # update lands, so zero_grad() need not drain the pipeline and
        # step N+1's backward overlaps step N's tail (the point of the
        # reference's cross-barrier).  Buckets holding SPLIT params are
        # excluded (their views feed the split copy-back) and zeroed at
        # split-completion instead.
        self.auto_zero = auto_zero
        if named_parameters is None:
            named_parameters = model.named_parameters()
        named = [(n, p) for n, p in named_parameters if p.requires_grad]
        self._engine = register_engine(GradEngine(
            named, process_group=process_group,
            partition_bytes=partition_bytes))
        self._engine.on_bucket_issued = self._enqueue

        # map param → hyperparameter group of the user optimizer
        self._group_of: Dict[int, dict] = {}
        for group in optimizer.param_groups:
            for p in group["params"]:
                self._group_of[id(p)] = group

        # Params split across buckets (numel > partition size) need special
        # handling: their p.grad is PRIVATE (not a bucket view), refreshed
        # from the averaged shard views only once ALL containing buckets
        # completed — and stepped exactly once, by a dedicated optimizer.
        eng = self._engine
        self._split_pidx_of = {id(eng.params[pidx]): pidx
                               for pidx in eng._split_params}

        def _mk_opt(opt_cls, params):
            groups = []
            for p in params:
                g = self._group_of.get(id(p))
                hyper = {k: v for k, v in (g or {}).items() if k != "params"}
                groups.append({"params": [p], **hyper})
            return opt_cls(groups)

        # one optimizer instance per bucket over exactly its NON-split
        # params; one per split param (stepped by its last bucket)
        opt_cls = type(optimizer)
        self._bucket_opts: Dict[int, torch.optim.Optimizer] = {}
        for b in self._engine.buckets:
            whole = [p for p in dict.fromkeys(b.params)
                     if id(p) not in self._split_pidx_of]
            if whole:
                self._bucket_opts[b.plan.index] = _mk_opt(opt_cls, whole)
        self._split_opts: Dict[int, torch.optim.Optimizer] = {}
        self._split_total: Dict[int, int] = {}
        self._split_remaining: Dict[int, int] = {}
        for pidx, bks in eng._split_params.items():
            self._split_opts[pidx] = _mk_opt(opt_cls, [eng.params[pidx]])
            self._split_total[pidx] = len(bks)
            self._split_remaining[pidx] = len(bks)

        # per-parameter "updated" events; forward pre-hooks block on them
        self._events: Dict[int, threading.Event] = {}
        for p in self._engine.params:
            ev = threading.Event()
            ev.set()
            self._events[id(p)] = ev
        self._install_prehooks()

        self._q: "queue.Queue[Optional[Bucket]]" = queue.Queue()
        self._pending = 0
        self._pending_lock = threading.Lock()
        self._idle = threading.Event()
        self._idle.set()
        self._poller = threading.Thread(target=self._poll_loop, daemon=True,
                                        name="bps-crossbarrier")
        self._poller.start()

    # -- pipeline ----------------------------------------------------------

    def _enqueue(self, bucket: Bucket) -> None:
        for p in bucket.params:
            self._events[id(p)].clear()
        with self._pending_lock:
            self._pending += 1
            self._idle.clear()
        self._q.put(bucket)

    def _poll_loop(self) -> None:
        while True:
            bucket = self._q.get()
            if bucket is None:
                return
            try:
                self._finish_bucket(bucket)
            except Exception:
                log.exception("cross-barrier poller failed")
            for p in bucket.params:
                # split params unlock only when their LAST bucket lands
                # (_finish_bucket sets those events itself)
                if id(p) not in self._split_pidx_of:
                    self._events[id(p)].set()
            with self._pending_lock:
                self._pending -= 1
                if self._pending == 0:
                    self._idle.set()

    def _finish_bucket(self, bucket: Bucket) -> None:
        eng = self._engine
        wire_src = None
        if bucket.ps_ticket is not None:
            eng._ps.wait(bucket.ps_ticket)
        elif bucket.work is not None:
            bucket.work.wait()
            if eng.comm_dtype is not None:
                wire_src = eng._wire_scratch[bucket.plan.index]
        divide = eng.average and eng.world > 1 and not eng.prescale
        if wire_src is not None or divide:
            alpha = (1.0 / eng.world) if divide else 1.0
            from .. import ops as _ops
            if bucket.buffer.is_cuda and _ops.have_core():
                # fused cast-back + averaging divide, one CDNA4 kernel
                _ops.cast_scale_(
                    bucket.buffer,
                    wire_src if wire_src is not None else bucket.buffer,
                    alpha)
            else:
                if wire_src is not None:
                    bucket.buffer.copy_(wire_src)
                if divide:
                    bucket.buffer.div_(eng.world)
        opt = self._bucket_opts.get(bucket.plan.index)
        if opt is not None:
            self._refresh_hyper(opt)
            opt.step()
        if self.auto_zero:
            # zero per-span: split params' views are WRITTEN (not
            # accumulated) at hook time and still feed the split
            # copy-back, so only non-split spans reset here
            for q, g in zip(bucket.params, bucket.grads):
                if id(q) not in self._split_pidx_of:
                    g.zero_()
        # split params: when the last containing bucket lands, refresh
        # p.grad from the averaged shard views and step exactly once
        for p in bucket.params:
            pidx = self._split_pidx_of.get(id(p))
            if pidx is None:
                continue
            self._split_remaining[pidx] -= 1
            if self._split_remaining[pidx] > 0:
                continue
            self._split_remaining[pidx] = self._split_total[pidx]
            flat = p.grad.reshape(-1)
            off = 0
            for b in eng._split_params[pidx]:
                for q, g in zip(b.params, b.grads):
                    if q is p:
                        flat.narrow(0, off, g.numel()).copy_(g)
                        off += g.numel()
            sopt = self._split_opts[pidx]
            self._refresh_hyper(sopt)
            sopt.step()
            if self.auto_zero:
                p.grad.zero_()      # autograd accumulates into the
                                    # private grad; views are copy-target
            self._events[id(p)].set()
        bucket.reset()

    def _refresh_hyper(self, opt: torch.optim.Optimizer) -> None:
        # refresh live hyperparams (lr schedules) from the user optimizer
        for g in opt.param_groups:
            src = self._group_of.get(id(g["params"][0]))
            if src:
                for k, v in src.items():
                    if k != "params":
                        g[k] = v

    # -- forward gating ------------------------------------------------------

    def _install_prehooks(self) -> None:
        def make_hook(params: List[torch.nn.Parameter]):
            events = [self._events[id(p)] for p in params
                      if id(p) in self._events]

            def hook(_module, _inp):
                for ev in events:
                    ev.wait()
            return hook

        for m in self.model.modules():
            direct = list(m.parameters(recurse=False))
            if direct:
                m.register_forward_pre_hook(make_hook(direct))

    # -- user API -----------------------------------------------------------

    def zero_grad(self) -> None:
        if self.auto_zero:
            return                  # poller zeroes per bucket at update
        self.synchronize()          # grads must not be zeroed mid-update
        self._engine.zero_grad()

    def step(self) -> None:
        """End of backward: issue any not-yet-ready buckets and return —
        updates complete asynchronously."""
        self._engine.flush()
        # reset per-step engine counters without waiting
        with self._engine._lock:
            self._engine._ready_params = 0
            self._engine._inflight_bytes = 0
            self._engine._step += 1

    def synchronize(self) -> None:
        """Block until every in-flight bucket is communicated + applied."""
        self._idle.wait()

    def stop(self) -> None:
        self.synchronize()
        self._q.put(None)
        self._poller.join(timeout=10)
