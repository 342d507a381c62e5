"""Hierarchical PS pipeline — the inter-node push/pull path.

Route per bucket (reference queue-list construction,
common/operations.cc:429-485, re-designed for MI355X):

  intra-node RCCL reduce-scatter (xGMI)          [issued in backward-hook
    → [compress shard — HIP codec kernel]         order, deterministic
    → D2H of (compressed) shard into pinned       across ranks]
      staging on a side HIP stream
    → KV push → KV pull (C++ TCP client, GIL released)
    → H2D → [decompress — HIP kernel]
    → intra-node all-gather                      [issued at synchronize()
                                                  in fixed bucket order]

Only this rank's shard — compressed — crosses PCIe and the NIC; full
gradients never leave HBM3E (the reference staged every byte through CPU
shm even on one node, common/core_loops.cc:378-443).

Collective-ordering rule: RCCL collectives on one communicator must be
issued in the same order on every rank.  Reduce-scatters are issued from
autograd hooks (same backward graph ⇒ same order — the same contract
torch DDP relies on) on the node group; trailing all-gathers are issued
by ONE dedicated issuer thread, in submit (hook) order, on a SEPARATE
communicator (``ag_group``) — so the two phases can never interleave
differently across ranks, whichever thread (synchronize() or the
cross-barrier poller) consumes the results.  The middle (KV) section
runs in a thread pool with per-task side streams and completes in any
order; push and pull are in flight together (full duplex — reference
docs/faq.md:23-25), the server defers the pull reply until the round's
merge is complete.
"""

from __future__ import annotations

import struct
import threading
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Dict, Optional

import torch
import torch.distributed as dist

from .. import common as C
from ..common import telemetry
from ..common.logging_util import get_logger
from ..common.naming import partition_key
from ..compression import BaseCompressor

log = get_logger()

_OP_INIT, _OP_PUSH, _OP_PULL = 5, 1, 2


def _make_cmd(codec: int, dtype: int, async_mode: bool) -> int:
    return (codec & 0xFF) | ((dtype & 0xFF) << 8) | (
        (1 << 16) if async_mode else 0)


class _KeyInfo:
    __slots__ = ("key", "server", "nelem", "round", "initialized",
                 "compressor", "server_ef")

    def __init__(self, key: int, server: int, nelem: int,
                 compressor: Optional[BaseCompressor],
                 server_ef: bool = False):
        self.key = key
        self.server = server
        self.nelem = nelem
        self.round = 0
        self.initialized = False
        self.compressor = compressor
        self.server_ef = server_ef


class _Staging:
    """Host staging + side stream for one in-flight bucket.

    Colocated server: the buffers are allocated INSIDE a shm region
    shared with the server (KVClient.ipc_alloc → hipHostRegister'd), so
    the D2H target IS the server's input and the server's reply lands
    directly in the H2D source — zero socket copies for gradient bytes.
    Remote server (or IPC off): private pinned buffers, payload travels
    inline over TCP."""

    def __init__(self, nbytes_send: int, nbytes_recv: int, device,
                 kv=None, server: Optional[int] = None,
                 wire_bytes: int = 0):
        self.on_gpu = device.type == "cuda"
        self._keep = []
        self.ipc = False
        self.send = self._alloc(kv, server, nbytes_send)
        self.recv = self._alloc(kv, server, nbytes_recv)
        # host-side wire transform output (Elias-coded dithering)
        self.wire = self._alloc(kv, server, wire_bytes) if wire_bytes \
            else None
        self.stream = torch.cuda.Stream(device) if self.on_gpu else None

    def _alloc(self, kv, server, nbytes: int) -> torch.Tensor:
        nbytes = max(int(nbytes), 1)
        if kv is not None and server is not None:
            try:
                addr = kv.ipc_alloc(server, nbytes)
            except Exception:
                addr = 0
            if addr:
                import ctypes
                buf = (ctypes.c_uint8 * nbytes).from_address(addr)
                self._keep.append(buf)
                self.ipc = True
                return torch.frombuffer(buf, dtype=torch.uint8)
        return torch.empty(nbytes, dtype=torch.uint8,
                           pin_memory=self.on_gpu)


class Ticket:
    __slots__ = ("future", "bucket", "shard", "done_event", "reply_view",
                 "seq", "ag_done", "ag_work", "error")

    def __init__(self, future: Optional[Future], bucket, shard):
        self.future = future
        self.bucket = bucket
        self.shard = shard
        self.done_event = None          # HIP event: KV H2D complete
        self.reply_view = None
        self.seq = -1                   # submit order (same on all ranks)
        self.ag_done = threading.Event()  # issuer finished this ticket
        self.ag_work = None             # dist Work of the trailing collective
        self.error = None               # exception raised on the issuer


def _kv_client():
    st = C._state
    if st.kv is None:
        from ..ops import core
        from ..common.config import server_addresses
        uris = server_addresses(st.cfg)
        if not uris and st.cfg.num_servers > 0:
            # scheduler rendezvous (reference DMLC_PS_ROOT_URI path)
            from ..launcher.scheduler import discover_servers
            uris = discover_servers(st.cfg.root_uri, st.cfg.root_port)
        if not uris:
            raise RuntimeError(
                "PS mode needs BPS_SERVER_URIS=host:port[,...] or a "
                "scheduler at BPS_ROOT_URI:BPS_ROOT_PORT with "
                "BPS_NUM_SERVER set")
        st.kv = core().KVClient(st.rank, uris)
        if st.assigner is None:
            from ..common.naming import ServerAssigner
            st.assigner = ServerAssigner(len(uris))
    return st.kv


class PSPipeline:
    """Per-GradEngine PS pipeline over the engine's buckets."""

    def __init__(self, engine) -> None:
        self.engine = engine
        st = C._state
        self.cfg = st.cfg
        self.local_size = max(1, st.local_size)
        self.world = engine.world
        self.num_nodes = max(1, self.world // self.local_size)
        self.node_id = st.rank // self.local_size
        self.local_rank = st.rank % self.local_size
        self.kv = _kv_client()
        # alternative strategy (reference BYTEPS_REDUCE_ROOTS,
        # common/global.cc:237-251): whole-bucket reduce to a key-hashed
        # root rank + root-only push/pull + broadcast, instead of
        # reduce-scatter sharding.  On xGMI the sharded path is usually
        # faster (all 7 links busy); root-reduce helps when bucket count
        # ≫ ranks and per-key latency dominates.
        from ..common.config import env_bool
        self.reduce_roots = env_bool("BPS_REDUCE_ROOTS",
                                     "BYTEPS_REDUCE_ROOTS", default=False)
        # pool sized to the in-flight work: more threads than buckets
        # only adds side-stream/GIL contention (measured: 8 threads over
        # 4×32 MiB buckets cost ~10% vs 4, profiles/MEASUREMENTS.md)
        nbuckets = max(1, len(getattr(engine, "buckets", []) or []))
        self.pool = ThreadPoolExecutor(
            max_workers=max(2, min(self.cfg.compressor_threads, nbuckets)),
            thread_name_prefix="bps-ps")
        self.keys: Dict[int, _KeyInfo] = {}
        self._lock = threading.Lock()
        self._staging: Dict[int, _Staging] = {}

        # intra-node subgroups: one for the leading reduce-scatters
        # (node_group, autograd-hook thread) and a SEPARATE communicator
        # for the trailing all-gathers (ag_group, issuer thread) — two
        # threads must never issue onto one communicator (reference kept
        # per-purpose NCCL rings too, nccl_manager.cc:83-84)
        self.node_group = None
        self.ag_group = None
        if self.world > 1 and dist.is_initialized():
            if self.num_nodes > 1:
                for nid in range(self.num_nodes):
                    ranks = list(range(nid * self.local_size,
                                       (nid + 1) * self.local_size))
                    g = dist.new_group(ranks)
                    g2 = dist.new_group(ranks)
                    if nid == self.node_id:
                        self.node_group = g
                        self.ag_group = g2
            else:
                self.node_group = None  # default group == node group
                self.ag_group = dist.new_group(list(range(self.world)))
        self.node_world = dist.get_world_size(self.node_group) \
            if (self.world > 1 and dist.is_initialized()) else 1

        # ordered issuer: tickets enter in submit (hook) order; the issuer
        # waits for each ticket's KV section, then issues its trailing
        # collective — strict FIFO, so issue order is identical on every
        # rank by construction
        import queue as _q
        self._issue_q: "_q.Queue[Optional[Ticket]]" = _q.Queue()
        self._seq = 0
        self._ag_stream = None
        self._closed = False
        self._issuer = threading.Thread(target=self._issue_loop, daemon=True,
                                        name="bps-ps-issuer")
        self._issuer.start()

        # engine-wide defaults; each bucket carries its own resolved
        # config (per-parameter overrides, reference byteps_* attrs)
        self.compression_params: Dict = dict(
            getattr(engine, "compression_params", {}) or {})

    # -- helpers -----------------------------------------------------------

    def _key_info(self, bucket) -> _KeyInfo:
        with self._lock:
            ki = self.keys.get(bucket.plan.index)
            if ki is None:
                from ..compression import create
                if self.reduce_roots:
                    nelem = bucket.buffer.numel()
                    pkey = partition_key(bucket.declared_key, 0)
                else:
                    nelem = bucket.buffer.numel() // self.node_world
                    pkey = partition_key(bucket.declared_key, self.local_rank)
                nbytes = nelem * 4
                server = C._state.assigner.assign(pkey, nbytes)
                compressor = None
                cfg = dict(bucket.compression_params
                           if bucket.compression_params is not None
                           else self.compression_params)
                cfg.pop("param_overrides", None)
                if cfg.get("compressor_type") and \
                        nbytes >= self.cfg.min_compress_bytes:
                    compressor = create(cfg)
                ef = str(cfg.get("ef_type", "")).lower() in \
                    ("vanilla", "1", "true")
                ki = _KeyInfo(pkey, server, nelem, compressor,
                              compressor is not None and ef)
                # elastic resume: continue the round sequence — a fresh
                # round=0 would let version-gated pulls be answered with
                # the PREVIOUS session's merged data
                ki.round = C._state.key_rounds.get(pkey, 0)
                self.keys[bucket.plan.index] = ki
            return ki

    def _staging_for(self, bucket, ki: _KeyInfo) -> _Staging:
        with self._lock:
            s = self._staging.get(bucket.plan.index)
            if s is None:
                raw = ki.nelem * 4
                wire_bytes = 0
                if ki.compressor is not None:
                    codec = ki.compressor.codec
                    if codec in (2, 3):        # topk / randomk: 8 B per pair
                        cap = 8 * min(max(ki.compressor.levels, 1), ki.nelem)
                    elif codec == 1:           # onebit: bits + scale
                        cap = ((ki.nelem + 63) // 64) * 8 + 8
                    elif codec in (4, 5):      # dithering: norm+flag+codes
                        cap = ki.nelem + 5
                        if getattr(ki.compressor, "host_wire", False):
                            wire_bytes = cap
                    else:                      # fp8: amax + codes
                        cap = ki.nelem + 4
                    cap = max(cap, 64)
                else:
                    cap = raw
                s = _Staging(cap, cap, bucket.buffer.device,
                             kv=self.kv, server=ki.server,
                             wire_bytes=wire_bytes)
                self._staging[bucket.plan.index] = s
            return s

    def _ensure_init(self, ki: _KeyInfo, st=None) -> None:
        if ki.initialized:
            return
        comp = ki.compressor
        codec = comp.codec if comp is not None else 0
        levels = comp.levels if comp is not None else 0
        # bit0: ask the server to run error feedback on its merged reply
        flags = 1 if ki.server_ef else 0
        # bit1: colocated shm lane — skip bit-level wire coding on BOTH
        # sides (Elias compression buys nothing when no NIC is crossed)
        if st is not None and st.ipc:
            flags |= 2
        payload = struct.pack("<QIIII", ki.nelem, self.num_nodes, levels,
                              flags, 0)
        buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
        cmd = _make_cmd(codec, 0, self.cfg.enable_async)
        t = self.kv.submit(ki.server, _OP_INIT, ki.key, buf.data_ptr(),
                           len(payload), 0, 0, cmd, 0)
        _len, aux = self.kv.wait(t)
        if aux == 0xFFFFFFFFFFFFFFFF:
            raise RuntimeError(
                "PS server rejected re-init of key %d: nelem/codec differ "
                "from the stored state (elastic resume with a changed "
                "config needs new keys — recreate the optimizer)" % ki.key)
        ki.initialized = True

    # -- main entry ---------------------------------------------------------

    def submit(self, bucket) -> Ticket:
        """Called from the autograd hook (deterministic order).  Issues the
        intra-node reduce-scatter inline, hands the KV section to the
        pool, and registers the ticket with the ordered issuer."""
        ki = self._key_info(bucket)
        buf = bucket.buffer
        dev = buf.device

        skip_kv = False
        if self.node_world > 1 and self.reduce_roots:
            root_local = bucket.declared_key % self.node_world
            root_global = self.node_id * self.node_world + root_local
            dist.reduce(buf, dst=root_global, group=self.node_group)
            shard = buf
            # non-root: nothing to push; the issuer broadcasts the result
            skip_kv = self.local_rank != root_local
        elif self.node_world > 1:
            per = buf.numel() // self.node_world
            shard = buf.narrow(0, self.local_rank * per, per)
            if dist.get_backend(self.node_group) == "nccl":
                dist.reduce_scatter_tensor(shard, buf, group=self.node_group)
            else:
                dist.all_reduce(buf, group=self.node_group)
        else:
            shard = buf

        rs_event = None
        if dev.type == "cuda":
            rs_event = torch.cuda.Event()
            rs_event.record(torch.cuda.current_stream(dev))

        if skip_kv:
            fut = Future()
            fut.set_result((None, None))
            ticket = Ticket(fut, bucket, shard)
        else:
            fut = self.pool.submit(self._kv_section, bucket, ki, shard,
                                   rs_event)
            ticket = Ticket(fut, bucket, shard)
        ticket.seq = self._seq
        self._seq += 1
        self._issue_q.put(ticket)
        return ticket

    def _kv_section(self, bucket, ki: _KeyInfo, shard: torch.Tensor,
                    rs_event) -> tuple:
        """Runs in a pool thread: compress → D2H → push/pull → H2D →
        decompress.  Returns (done_event | None, reply_fp32 | None)."""
        st = self._staging_for(bucket, ki)
        self._ensure_init(ki, st)
        comp = ki.compressor
        ki.round += 1
        C._state.key_rounds[ki.key] = ki.round
        on_gpu = shard.is_cuda
        tracer = C._state.tracer
        step = ki.round - 1

        def _tr(stage, beginning):
            if tracer is not None:
                (tracer.begin if beginning else tracer.end)(
                    ki.key, stage, step)

        _tr("compress+d2h", True)

        stream_ctx = torch.cuda.stream(st.stream) if on_gpu else _null_ctx()
        with stream_ctx:
            if on_gpu:
                st.stream.wait_event(rs_event)
            shard_f = shard if shard.dtype == torch.float32 \
                else shard.float()
            if comp is not None:
                cp = comp.compress(shard_f)
                payload = torch.cat(
                    [p.reshape(-1).view(torch.uint8) for p in cp.parts])
                nbytes = payload.numel()
                push_aux = cp.aux if cp.aux else ki.round
            else:
                payload = shard_f.view(torch.uint8).reshape(-1)
                nbytes = payload.numel()
                push_aux = ki.round
            st.send[:nbytes].copy_(payload, non_blocking=on_gpu)
            if on_gpu:
                st.stream.synchronize()

        _tr("compress+d2h", False)
        # host-side wire transform (Elias-coded dithering): encode the
        # dense staging into the wire buffer; fall back to dense when the
        # stream would be larger
        push_buf = st.send
        if comp is not None and getattr(comp, "host_wire", False) \
                and not st.ipc:
            wlen = comp.encode_wire(st.send[:nbytes], ki.nelem, st.wire)
            if wlen > 0:
                push_buf = st.wire
                nbytes = wlen
        codec = comp.codec if comp is not None else 0
        cmd = _make_cmd(codec, 0, self.cfg.enable_async)
        # full duplex: push and pull are both in flight — the server
        # defers the pull reply until this round's merge is complete
        # (version gate), so the pull can be posted immediately and its
        # latency overlaps other buckets' pushes (reference
        # docs/faq.md:23-25)
        _tr("push", True)
        t_push = self.kv.submit(ki.server, _OP_PUSH, ki.key,
                                push_buf.data_ptr(), nbytes, 0, 0, cmd,
                                push_aux)
        t_pull = self.kv.submit(ki.server, _OP_PULL, ki.key, 0, 0,
                                st.recv.data_ptr(), st.recv.numel(), cmd,
                                ki.round)
        self.kv.wait(t_push)
        _tr("push", False)
        telemetry.record(nbytes)
        _tr("pull", True)
        reply_len, _ver = self.kv.wait(t_pull)
        _tr("pull", False)
        if reply_len > st.recv.numel():
            # the transport drops oversized payloads into scratch — the
            # staging buffer would hold stale bytes; fail loudly
            raise RuntimeError(
                "PS pull reply (%d B) exceeds staging capacity (%d B) for "
                "key %d — codec config mismatch between worker and server"
                % (reply_len, st.recv.numel(), ki.key))
        telemetry.record(reply_len)

        _tr("h2d+decompress", True)
        done_event = None
        with stream_ctx:
            if comp is not None:
                wire = st.recv[:reply_len]
                if getattr(comp, "host_wire", False):
                    # Elias reply → dense host payload before H2D; the
                    # push wire buffer is pinned and free by reply time
                    wire = comp.decode_wire(wire, ki.nelem, out=st.wire)
                if on_gpu:
                    wire = wire.to(shard.device, non_blocking=True)
                # sparse codecs: the server's reply k = min(levels, nelem);
                # recover it from the wire length (8 B per pair)
                reply_aux = reply_len // 8 if comp.codec in (2, 3) else 0
                merged = comp.decompress(wire, ki.nelem, aux=reply_aux)
                shard.copy_(merged.to(shard.dtype).reshape(shard.shape),
                            non_blocking=on_gpu)
            else:
                host_f = st.recv[:reply_len].view(torch.float32)
                if shard.dtype == torch.float32:
                    shard.reshape(-1).copy_(host_f, non_blocking=on_gpu)
                else:
                    shard.reshape(-1).copy_(
                        host_f.to(shard.dtype), non_blocking=on_gpu)
            if on_gpu:
                done_event = torch.cuda.Event()
                done_event.record(st.stream)
        _tr("h2d+decompress", False)
        return done_event, None

    def _issue_loop(self) -> None:
        """Single issuer thread: strict FIFO over tickets (= submit order,
        identical on every rank), so the trailing collectives on ag_group
        can never interleave differently across ranks — whichever thread
        later consumes the results."""
        while True:
            ticket = self._issue_q.get()
            if ticket is None:
                return
            try:
                self._issue_one(ticket)
            except Exception as e:   # surface on the waiter, don't die
                ticket.error = e
            ticket.ag_done.set()

    def _issue_one(self, ticket: Ticket) -> None:
        done_event, _ = ticket.future.result()
        ticket.done_event = done_event
        buf = ticket.bucket.buffer
        on_gpu = buf.is_cuda
        if self.node_world <= 1:
            return                      # nothing collective to issue
        if on_gpu and self._ag_stream is None:
            self._ag_stream = torch.cuda.Stream(buf.device)
        stream_ctx = torch.cuda.stream(self._ag_stream) if on_gpu \
            else _null_ctx()
        with stream_ctx:
            if done_event is not None:
                self._ag_stream.wait_event(done_event)
                ticket.done_event = None    # consumed by the ag chain
            if self.reduce_roots:
                root_local = ticket.bucket.declared_key % self.node_world
                root_global = self.node_id * self.node_world + root_local
                ticket.ag_work = dist.broadcast(
                    buf, src=root_global, group=self.ag_group, async_op=True)
            elif dist.get_backend(self.ag_group) == "nccl":
                ticket.ag_work = dist.all_gather_into_tensor(
                    buf, ticket.shard, group=self.ag_group, async_op=True)
            else:
                # chunks are contiguous views of buf — all_gather fills
                # the bucket in place
                chunks = list(buf.chunk(self.node_world))
                ticket.ag_work = dist.all_gather(
                    chunks, ticket.shard.contiguous(), group=self.ag_group,
                    async_op=True)

    def wait(self, ticket: Ticket) -> None:
        """Called from synchronize() or the cross-barrier poller (any
        order): join the issuer, then chain the result into the caller's
        current stream."""
        ticket.ag_done.wait()
        if ticket.error is not None:
            raise ticket.error
        if ticket.future is not None:
            ticket.future.result()      # re-raise KV-section exceptions
        if ticket.ag_work is not None:
            # Work.wait() blocks (gloo) / makes the current stream wait
            # for the collective (nccl)
            ticket.ag_work.wait()
        elif ticket.done_event is not None:
            torch.cuda.current_stream(
                ticket.bucket.buffer.device).wait_event(ticket.done_event)

    def close(self) -> None:
        if self._closed:
            return
        self._closed = True
        self._issue_q.put(None)
        self._issuer.join(timeout=10)
        self.pool.shutdown(wait=False)


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


# -- standalone tensor pipeline (functional push_pull API in PS mode) -------

class TensorWork:
    def __init__(self, fut: Future):
        self._fut = fut

    def wait(self):
        self._fut.result()

    def is_completed(self):
        return self._fut.done()


class TensorPipeline:
    """push_pull of an arbitrary tensor through the PS: every rank pushes
    the whole tensor (expected pushers = world), the server sums, the pull
    returns the global sum — the reference's non-bucketed push_pull path
    (torch/ops.cc:54-97)."""

    def __init__(self) -> None:
        st = C._state
        self.kv = _kv_client()
        self.pool = ThreadPoolExecutor(max_workers=4,
                                       thread_name_prefix="bps-pp")
        self.world = st.size
        self.rank = st.rank
        self.keys: Dict[str, _KeyInfo] = {}
        self._lock = threading.Lock()

    def submit_tensor(self, tensor: torch.Tensor, name: str,
                      priority: int = 0) -> TensorWork:
        return TensorWork(self.pool.submit(self._run, tensor, name))

    def _run(self, tensor: torch.Tensor, name: str):
        st = C._state
        key = st.registry.declare(name)
        with self._lock:
            ki = self.keys.get(name)
            if ki is None:
                pkey = partition_key(key, 0)
                nelem = tensor.numel()
                server = st.assigner.assign(pkey, nelem * 4)
                ki = _KeyInfo(pkey, server, nelem, None)
                ki.round = C._state.key_rounds.get(pkey, 0)
                self.keys[name] = ki
        if not ki.initialized:
            payload = struct.pack("<QII", ki.nelem, self.world, 0)
            buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
            t = self.kv.submit(ki.server, _OP_INIT, ki.key, buf.data_ptr(),
                               len(payload), 0, 0, 0, 0)
            self.kv.wait(t)
            ki.initialized = True
        ki.round += 1
        C._state.key_rounds[ki.key] = ki.round
        src = tensor.detach()
        host = src.float().cpu().contiguous() if src.is_cuda \
            else src.float().contiguous()
        t = self.kv.submit(ki.server, _OP_PUSH, ki.key, host.data_ptr(),
                           host.numel() * 4, 0, 0, 0, ki.round)
        self.kv.wait(t)
        recv = torch.empty_like(host)
        t = self.kv.submit(ki.server, _OP_PULL, ki.key, 0, 0,
                           recv.data_ptr(), recv.numel() * 4, 0, ki.round)
        self.kv.wait(t)
        with torch.no_grad():
            tensor.reshape(-1).copy_(
                recv.to(tensor.device, tensor.dtype).reshape(-1))


_tensor_pipeline: Optional[TensorPipeline] = None


def get_pipeline(engine) -> PSPipeline:
    return PSPipeline(engine)


def get_tensor_pipeline() -> TensorPipeline:
    global _tensor_pipeline
    if _tensor_pipeline is None:
        _tensor_pipeline = TensorPipeline()
    return _tensor_pipeline
