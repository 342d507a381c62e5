"""The gradient-synchronization engine.

MI355X-native re-design of the reference's 12-stage queue pipeline
(reference common/core_loops.cc:755-855, common/scheduled_queue.cc).  One
process per GPU; the intra-node data plane is RCCL over xGMI driven through
``torch.distributed`` (gloo on CPU for tests).  Design:

- **Persistent flat buckets**: parameters are packed (in reverse
  registration order, so the gradients backward produces first complete
  first) into flat buffers; ``param.grad`` is a zero-copy *view* into its
  bucket (the reference instead copied every gradient into pinned shared
  memory, common/core_loops.cc:378-443 — on MI355X gradients stay resident
  in HBM3E and PCIe is only crossed in PS mode, with compressed bytes).
- **Priority scheduling**: ready buckets are drained through a max-heap
  keyed by (priority, -index) (reference scheduled_queue.cc:86-95); byte
  credits (``BPS_SCHEDULING_CREDIT``) bound in-flight communication
  (reference scheduled_queue.cc:33-45).
- **Overlap**: collectives are issued with ``async_op=True`` → RCCL runs on
  its own HIP stream, overlapping the remaining backward; ``synchronize()``
  waits on the Work events (and in PS mode the KV completion flags).
- **PS mode** (``BPS_NUM_SERVER>0`` or ``BPS_FORCE_DISTRIBUTED``): per
  bucket reduce-scatter → (compress →) D2H into pinned staging → push/pull
  to the C++ KV server → H2D (→ decompress) → all-gather, with only this
  rank's shard crossing PCIe (reference route construction,
  common/operations.cc:429-485).
"""

from __future__ import annotations

import heapq
import threading
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Sequence

import torch
import torch.distributed as dist

from .. import common as C
from ..common.config import Config
from ..common.logging_util import get_logger
from ..common.partition import PartitionPlan, plan_partitions

log = get_logger()


def _is_dense_permutation(t: torch.Tensor) -> bool:
    """True iff t's layout is a stride-permutation covering exactly
    [0, numel) — e.g. channels_last — so a flat range can alias it."""
    if t.numel() == 0:
        return True
    expect = 1
    for s, sz in sorted(zip(t.stride(), t.shape)):
        if sz == 1:
            continue
        if s != expect:
            return False
        expect *= sz
    return True


# --------------------------------------------------------------------------
# Bucket: one schedulable flat buffer
# --------------------------------------------------------------------------

@dataclass
class Bucket:
    plan: PartitionPlan
    buffer: torch.Tensor                      # flat grad buffer (persistent)
    params: List[torch.nn.Parameter]          # params whose grads live here
    grads: List[torch.Tensor]                 # the views (same order)
    priority: int = 0
    declared_key: int = 0
    compression_params: Optional[dict] = None   # resolved codec config
    # per-step state
    ready_count: int = 0
    issued: bool = False
    # engine-step number of the last issue.  Unlike ``issued`` (cleared by
    # reset(), which the cross-barrier poller calls as soon as a bucket's
    # update lands), the stamp survives until the next step — flush()
    # keys on it so a bucket can never be double-issued within one step
    # when the poller wins the race against step()'s flush.
    stamp: int = -1
    work: Optional[object] = None             # dist Work handle
    ps_ticket: Optional[object] = None        # PS pipeline ticket

    @property
    def nbytes(self) -> int:
        return self.buffer.numel() * self.buffer.element_size()

    def reset(self) -> None:
        self.ready_count = 0
        self.issued = False
        self.work = None
        self.ps_ticket = None


# --------------------------------------------------------------------------
# Engine
# --------------------------------------------------------------------------

class GradEngine:
    """Owns the buckets for one optimizer/DDP instance.

    ``params`` must be passed in registration (model) order on every rank —
    bucket layout is a pure function of that order, so all ranks agree
    without communication (mirrors the reference's sorted declare,
    common/global.cc:412-429).
    """

    def __init__(
        self,
        named_params: Sequence,               # iterable of (name, Parameter)
        process_group: Optional[dist.ProcessGroup] = None,
        partition_bytes: Optional[int] = None,
        average: bool = True,
        grad_dtype: Optional[torch.dtype] = None,
        prescale: bool = False,
        compression_params: Optional[dict] = None,
        comm_dtype: Optional[torch.dtype] = None,
    ) -> None:
        self.compression_params = dict(compression_params or {})
        C._require_init()
        self.cfg: Config = C.get_config()
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.average = average
        self.prescale = prescale and self.world > 1
        self._lock = threading.Lock()
        self._pending: List = []              # heap of (-priority, idx)
        self._inflight_bytes = 0
        self._credit = self.cfg.scheduling_credit  # 0 → unlimited
        self._hook_handles: List = []
        self._sync_enabled = True
        self._step = 0
        self._ready_params = 0
        # invoked (from the last autograd hook) once every param's grad is
        # ready — DDP uses it to self-synchronize (reference
        # parallel/distributed.py:261-270)
        self.on_all_ready: Optional[Callable] = None
        # invoked right after a bucket's collective is issued — the
        # cross-barrier poller consumes this to apply per-bucket optimizer
        # updates as results land (reference cross_barrier.py:159-186)
        self.on_bucket_issued: Optional[Callable] = None

        named = [(n, p) for n, p in named_params if p.requires_grad]
        if not named:
            raise ValueError("GradEngine: no parameters require grad")
        self.param_names = [n for n, _ in named]
        self.params = [p for _, p in named]
        self._name_of = {id(p): n for n, p in named}
        for name in self.param_names:
            C._state.registry.declare("byteps.Gradient." + name)

        self._grad_dt = grad_dtype or self.params[0].dtype
        self._device = self.params[0].device
        part_bytes = partition_bytes or self.cfg.partition_bytes
        self._part_elems = max(
            part_bytes // torch.tensor([], dtype=self._grad_dt).element_size(),
            4096)
        self._generation = 0
        self._build_buckets()
        dt = self._grad_dt

        # optional reduced-precision wire for the RCCL collectives
        # (``comm_dtype`` argument — set by Compression.fp16/bf16 — or
        # BPS_COMM_DTYPE=bf16): cast bucket → persistent 16-bit scratch,
        # all-reduce the scratch, cast back at synchronize.  Halves xGMI
        # bytes; gradients still ACCUMULATE at full precision — only the
        # wire is narrowed, matching the reference's Compression semantics
        # (torch/compression.py:34-76).
        if comm_dtype is None:
            from ..common.config import env_str
            wire = env_str("BPS_COMM_DTYPE", default="").lower()
            comm_dtype = torch.bfloat16 if wire in ("bf16", "bfloat16") \
                else torch.float16 if wire in ("fp16", "half") else None
        self.comm_dtype = comm_dtype
        if self.comm_dtype is not None and dt == self.comm_dtype:
            self.comm_dtype = None
        self._wire_scratch: Dict[int, torch.Tensor] = {}
        if self.comm_dtype is not None:
            # eager: the fused cast+scale descriptor needs stable pointers
            for b in self.buckets:
                self._wire_scratch[b.plan.index] = torch.empty_like(
                    b.buffer, dtype=self.comm_dtype)
        self._fused_desc = None     # (desc dev tensor, total_vec, vec)

        # multi-ring collectives (reference BYTEPS_NCCL_NUM_RINGS /
        # NcclManagerExpr, nccl_manager.cc:216-318): buckets round-robin
        # across N communicators so RCCL can overlap their ring
        # reductions instead of serializing on one comm's stream.
        # Default 1 (RCCL's own pipelining is usually enough); the knob
        # exists for the 8-GPU operator (docs/preflight-8gpu.md).
        from ..common.config import env_int
        nrings = max(1, env_int("BPS_NUM_RINGS", "BYTEPS_NCCL_NUM_RINGS",
                                default=1))
        self._rings: List = []
        if (nrings > 1 and self.group is None and self.world > 1
                and dist.is_initialized()):
            ranks = list(range(self.world))
            self._rings = [dist.new_group(ranks) for _ in range(nrings)]

        self._ps = None
        if C._state.ps_enabled:
            from . import ps_pipeline
            self._ps = ps_pipeline.get_pipeline(self)

        self._attach_hooks()
        log.debug("GradEngine: %d params → %d buckets (%.1f MiB each max)",
                  len(self.params), len(self.buckets),
                  self._part_elems * self.buckets[0].buffer.element_size()
                  / 2**20)

    # -- bucket plan (built at init; rebuilt on elastic re-bucketing) ------

    def _build_buckets(self) -> None:
        """Pack params into flat aligned buckets and alias ``p.grad``.
        A pure function of (param order, world, partition size,
        compression config) — identical on every rank without
        communication."""
        dt, dev = self._grad_dt, self._device
        part_elems = self._part_elems

        # Per-parameter compression overrides (reference: per-param
        # byteps_* attrs, mxnet/__init__.py:250-317): params whose
        # resolved codec config differs go into separate bucket groups so
        # each bucket has ONE wire format.  "param_overrides" maps a name
        # substring → kwargs merged over the base config.
        overrides = self.compression_params.get("param_overrides") or {}
        base_cfg = {k: v for k, v in self.compression_params.items()
                    if k != "param_overrides"}

        def resolved_cfg(name: str) -> dict:
            cfg = dict(base_cfg)
            for pat, extra in overrides.items():
                if pat in name:
                    cfg.update(extra)
            return cfg

        # Bucket in REVERSE registration order: backward runs output → input,
        # so the last-registered params produce gradients first and land in
        # bucket 0 (highest priority) — the priority semantics of the
        # reference (mxnet/__init__.py:58-60, scheduled_queue.cc:86-95).
        rev = list(range(len(self.params)))[::-1]
        groups: Dict[str, List[int]] = {}
        group_cfg: Dict[str, dict] = {}
        for i in rev:
            cfg = resolved_cfg(self.param_names[i])
            gkey = repr(sorted(cfg.items()))
            groups.setdefault(gkey, []).append(i)
            group_cfg[gkey] = cfg
        # align so reduce-scatter shards divide evenly for THIS world:
        # lcm(64, world) keeps 16-byte vector alignment for the HIP
        # kernels and exact shard division for any rank count (a plain 64
        # silently truncates the bucket tail at e.g. world=3 or 6)
        import math
        align = 64 * self.world // math.gcd(64, self.world)

        # key namespace: generation-suffixed after a re-bucket so PS
        # servers allocate fresh state (old nelem would be rejected by
        # the server's re-init validation)
        key_fmt = "byteps.Partition.%d" if self._generation == 0 \
            else "byteps.Partition.g%d.%%d" % self._generation

        self.buckets: List[Bucket] = []
        self.param_bucket: Dict[int, List[Bucket]] = {}   # param idx → buckets
        for gkey, idxs in groups.items():
            sizes = [self.params[i].numel() for i in idxs]
            plans = plan_partitions(sizes, part_elems, align=align)
            for plan in plans:
                buf = torch.zeros(plan.numel, dtype=dt, device=dev)
                index = len(self.buckets)
                bucket = Bucket(plan=plan, buffer=buf, params=[], grads=[])
                bucket.plan.index = index
                key = C._state.registry.declare(key_fmt % index)
                bucket.declared_key = key
                bucket.compression_params = group_cfg[gkey]
                for span in plan.spans:
                    pidx = idxs[span.param_index]
                    p = self.params[pidx]
                    view = buf.narrow(0, span.offset, span.numel)
                    bucket.params.append(p)
                    bucket.grads.append(view)
                    self.param_bucket.setdefault(pidx, []).append(bucket)
                self.buckets.append(bucket)
        # priorities follow creation order (reverse-registration within
        # each group, groups in first-seen order)
        n = len(self.buckets)
        for i, b in enumerate(self.buckets):
            b.priority = n - i

        # Attach p.grad views.  A param split across buckets cannot be a
        # single view — those (rare: only params > partition size) get a
        # private grad and a copy at ready time.
        self._split_params: Dict[int, List] = {}
        for pidx, p in enumerate(self.params):
            bks = self.param_bucket[pidx]
            if len(bks) == 1 and self._single_span(bks[0], p) is not None:
                view = self._single_span(bks[0], p)
                # Alias with the PARAM's stride order (e.g. channels_last
                # convs) so autograd accumulates without a layout
                # conversion — the "gradient layout contract".  Any dense
                # permuted layout is a bijection of the same flat range,
                # so collectives on the flat buffer see the same bytes.
                if not p.is_contiguous() and _is_dense_permutation(p):
                    p.grad = view.as_strided(p.shape, p.stride())
                else:
                    p.grad = view.view_as(p)
            else:
                # split across partitions: keep torch-allocated grad, copy
                # into the views when the grad is produced
                p.grad = torch.zeros_like(p)
                self._split_params[pidx] = bks

        # per-bucket comm scratch invalidated by any rebuild
        self._wire_scratch = {}
        if getattr(self, "comm_dtype", None) is not None:
            for b in self.buckets:
                self._wire_scratch[b.plan.index] = torch.empty_like(
                    b.buffer, dtype=self.comm_dtype)
        self._fused_desc = None

    def rebucket(self) -> None:
        """Elastic resume with a changed world: rebuild the bucket plan
        under the new alignment and re-alias every ``p.grad`` (the
        reference re-declared keys in original order,
        common/operations.cc:96-119; here the plan itself depends on
        world, so it is rebuilt deterministically on every rank)."""
        self._generation += 1
        with self._lock:
            self._pending.clear()
            self._inflight_bytes = 0
            self._ready_params = 0
        self._build_buckets()
        log.info("rebucket: generation %d, %d buckets for world %d",
                 self._generation, len(self.buckets), self.world)

    # -- helpers -----------------------------------------------------------

    def _single_span(self, bucket: Bucket, p: torch.nn.Parameter):
        spans = [g for q, g in zip(bucket.params, bucket.grads) if q is p]
        if len(spans) == 1 and spans[0].numel() == p.numel():
            return spans[0]
        return None

    def _attach_hooks(self) -> None:
        for pidx, p in enumerate(self.params):
            h = p.register_post_accumulate_grad_hook(
                self._make_hook(pidx))
            self._hook_handles.append(h)

    def _make_hook(self, pidx: int) -> Callable:
        def hook(param: torch.nn.Parameter) -> None:
            self._on_grad_ready(pidx)
        return hook

    # -- per-step flow ------------------------------------------------------

    def _on_grad_ready(self, pidx: int) -> None:
        if not self._sync_enabled:
            return
        p = self.params[pidx]
        if pidx in self._split_params:
            # copy the private grad into its bucket views
            flat = p.grad.reshape(-1)
            off = 0
            for b in self._split_params[pidx]:
                for q, g in zip(b.params, b.grads):
                    if q is p:
                        g.copy_(flat.narrow(0, off, g.numel()))
                        off += g.numel()
        with self._lock:
            self._ready_params += 1
            all_ready = self._ready_params == len(self.params)
            for b in self.param_bucket[pidx]:
                b.ready_count += 1
                if b.ready_count == len(b.params) and not b.issued:
                    heapq.heappush(self._pending, (-b.priority, b.plan.index))
            self._drain_locked()
        if all_ready and self.on_all_ready is not None:
            self.on_all_ready()

    def _drain_locked(self) -> None:
        """Issue every currently-ready bucket in priority order, subject to
        the byte credit."""
        while self._pending:
            negp, idx = self._pending[0]
            b = self.buckets[idx]
            if self._credit and self._inflight_bytes + b.nbytes > self._credit \
                    and self._inflight_bytes > 0:
                break
            heapq.heappop(self._pending)
            self._issue(b)

    def _issue(self, b: Bucket) -> None:
        if b.issued or b.stamp == self._step:
            return
        b.issued = True
        b.stamp = self._step
        self._inflight_bytes += b.nbytes
        if C._state.tracer is not None:
            C._state.tracer.begin(b.declared_key, "comm", self._step)
        self._debug_sample(b, "pre-comm")
        if self.world > 1 or self._ps is not None:
            if self.prescale:
                b.buffer.div_(self.world)
            if self._ps is not None:
                b.ps_ticket = self._ps.submit(b)
            elif self.comm_dtype is not None:
                scratch = self._wire_scratch.get(b.plan.index)
                if scratch is None:
                    scratch = torch.empty_like(b.buffer,
                                               dtype=self.comm_dtype)
                    self._wire_scratch[b.plan.index] = scratch
                scratch.copy_(b.buffer)
                b.work = dist.all_reduce(
                    scratch, op=dist.ReduceOp.SUM,
                    group=self._ring_for(b), async_op=True)
            else:
                b.work = dist.all_reduce(
                    b.buffer, op=dist.ReduceOp.SUM,
                    group=self._ring_for(b), async_op=True)
        if self.on_bucket_issued is not None:
            self.on_bucket_issued(b)

    def _ring_for(self, b: Bucket):
        if not self._rings:
            return self.group
        return self._rings[b.plan.index % len(self._rings)]

    def _debug_sample(self, b: Bucket, stage: str) -> None:
        """BPS_DEBUG_SAMPLE_TENSOR: print first/last values of a watched
        tensor's bucket after each stage (reference
        common/core_loops.cc:37-67)."""
        watch = self.cfg.debug_sample_tensor
        if not watch:
            return
        for p, g in zip(b.params, b.grads):
            name = self._name_of.get(id(p), "?")
            if watch in name:
                log.info("[sample %s] %s: first=%g last=%g (bucket %d)",
                         stage, name, float(g.reshape(-1)[0]),
                         float(g.reshape(-1)[-1]), b.plan.index)

    # -- public API ---------------------------------------------------------

    def zero_grad(self) -> None:
        """Zero all bucket buffers (and private split grads)."""
        torch._foreach_zero_([b.buffer for b in self.buckets])
        for pidx in self._split_params:
            self.params[pidx].grad.zero_()

    def set_sync_enabled(self, enabled: bool) -> None:
        """Gradient accumulation: when disabled, hooks accumulate into the
        flat buffers without communicating (reference no_sync,
        parallel/distributed.py:184-207)."""
        self._sync_enabled = enabled

    def sync_split_grads(self) -> None:
        """Copy every split param's private (accumulated) grad into its
        bucket views.  Needed before flush() whenever the hooks ran with
        sync disabled (gradient accumulation): the hooks early-return in
        that state, so the views still hold the previous round's bytes."""
        for pidx, bks in self._split_params.items():
            p = self.params[pidx]
            flat = p.grad.reshape(-1)
            off = 0
            for b in bks:
                for q, g in zip(b.params, b.grads):
                    if q is p:
                        g.copy_(flat.narrow(0, off, g.numel()))
                        off += g.numel()

    def flush(self) -> None:
        """Force-issue buckets whose params produced no grad this step
        (their spans hold zeros — contributing zero to the sum is correct)."""
        with self._lock:
            for b in self.buckets:
                if not b.issued and b.stamp != self._step:
                    heapq.heappush(self._pending, (-b.priority, b.plan.index))
            saved_credit = self._credit
            self._credit = 0          # credits never hold back a flush
            try:
                self._drain_locked()
            finally:
                self._credit = saved_credit

    def synchronize(self) -> None:
        """Wait for all issued buckets; apply deferred averaging; reset."""
        if not self._sync_enabled:
            return
        self.flush()
        needs_wire = False
        for b in self.buckets:
            if b.ps_ticket is not None:
                self._ps.wait(b.ps_ticket)
            elif b.work is not None:
                b.work.wait()
                if self.comm_dtype is not None:
                    needs_wire = True
        divide = self.average and self.world > 1 and not self.prescale
        if needs_wire or divide:
            self._apply_wire_and_average(needs_wire, divide)
        if self._split_params:
            # split params read back their reduced (averaged) grads —
            # unconditionally: with average=False the SUM must still
            # reach p.grad
            for pidx, bks in self._split_params.items():
                p = self.params[pidx]
                flat = p.grad.reshape(-1)
                off = 0
                for b in bks:
                    for q, g in zip(b.params, b.grads):
                        if q is p:
                            flat.narrow(0, off, g.numel()).copy_(g)
                            off += g.numel()
        if C._state.tracer is not None:
            for b in self.buckets:
                C._state.tracer.end(b.declared_key, "comm", self._step)
        if self.cfg.debug_sample_tensor:
            for b in self.buckets:
                self._debug_sample(b, "post-comm")
        for b in self.buckets:
            b.reset()
        self._ready_params = 0
        self._inflight_bytes = 0
        self._step += 1

    def _apply_wire_and_average(self, needs_wire: bool,
                                divide: bool) -> None:
        """Post-collective epilogue: the averaging divide fused with the
        reduced-precision wire cast-back — ONE hand-written CDNA4 kernel
        launch over all buckets on GPU (bps_cast_scale_many, kernels.hip;
        replaces per-bucket copy_ + torch._foreach_div_; the reference
        divided per-tensor on the framework side, torch/ops.cc:78-91)."""
        alpha = (1.0 / self.world) if divide else 1.0
        bufs = [b.buffer for b in self.buckets]
        from .. import ops as _ops
        if bufs[0].is_cuda and _ops.have_core():
            if self._fused_desc is None:
                srcs = [self._wire_scratch[b.plan.index]
                        for b in self.buckets] if needs_wire else bufs
                self._fused_desc = _ops.build_cast_scale_desc(bufs, srcs)
            desc, total_vec, _vec = self._fused_desc
            src_dt = self.comm_dtype if needs_wire else bufs[0].dtype
            _ops.cast_scale_many_(desc, len(bufs), total_vec, alpha,
                                  src_dt, bufs[0].dtype, bufs[0].device)
        else:
            if needs_wire:
                for b in self.buckets:
                    b.buffer.copy_(self._wire_scratch[b.plan.index])
            if divide:
                torch._foreach_div_(bufs, float(self.world))

    def detach(self) -> None:
        for h in self._hook_handles:
            h.remove()
        self._hook_handles.clear()


_engines: List[GradEngine] = []


def register_engine(e: GradEngine) -> GradEngine:
    _engines.append(e)
    return e


def _shutdown_engine() -> None:
    for e in _engines:
        try:
            e.detach()
            if e._ps is not None:
                e._ps.close()
        except Exception:
            pass
    _engines.clear()


def _suspend_engines() -> None:
    """Elastic suspend: detach hooks but keep engines registered so
    resume() can re-arm them (reference byteps_suspend,
    common/operations.cc:96-107)."""
    for e in _engines:
        e.detach()
        if e._ps is not None:
            try:
                e._ps.close()
            except Exception:
                pass
            e._ps = None


def _resume_engines() -> None:
    for e in _engines:
        e._attach_hooks()
        # topology may have changed: rebuild the bucket plan for the new
        # world (alignment and PS shard math are world-dependent).  Keys
        # move to a generation-suffixed namespace so PS servers allocate
        # fresh state (reference re-declared keys in original order,
        # operations.cc:96-119).
        new_world = dist.get_world_size() if dist.is_initialized() else 1
        if new_world != e.world:
            e.world = new_world
            if e.on_bucket_issued is not None:
                # a CrossBarrier owns per-bucket optimizers over the OLD
                # plan — rebuilding under it would silently orphan them
                log.warning(
                    "resume: world changed under a CrossBarrier engine — "
                    "recreate the CrossBarrier; skipping re-bucket")
            else:
                e.rebucket()
        e._ps = None
        if C._state.ps_enabled:
            from . import ps_pipeline
            e._ps = ps_pipeline.get_pipeline(e)
