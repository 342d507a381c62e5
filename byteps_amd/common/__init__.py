"""Process-wide runtime state: init / shutdown / suspend / resume, rank
and size queries (equivalent of the reference's ``BytePSBasics``,
byteps/common/__init__.py:57-100, and ``BytePSGlobal``,
common/global.cc:105-281).

One process per GPU.  The intra-node (and, without PS servers, global)
data plane is a ``torch.distributed`` process group — ``nccl`` (RCCL over
xGMI) when a GPU is present, ``gloo`` otherwise (CPU tests).  The
inter-node plane (PS push/pull over the C++ KV transport) activates when
``BPS_NUM_SERVER > 0`` or ``BPS_FORCE_DISTRIBUTED=1``.
"""

from __future__ import annotations

import atexit
import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist

from .config import Config
from .logging_util import get_logger
from .naming import NameRegistry, ServerAssigner

log = get_logger()


class _State:
    def __init__(self) -> None:
        self.initialized = False
        self.cfg: Optional[Config] = None
        self.rank = 0
        self.size = 1
        self.local_rank = 0
        self.local_size = 1
        self.registry = NameRegistry()
        self.assigner: Optional[ServerAssigner] = None
        self.device: Optional[torch.device] = None
        self.owns_process_group = False
        self.kv = None           # KV client handle (PS mode)
        self.key_rounds = {}     # pkey → last push round (survives the
                                 # pipeline across suspend/resume so the
                                 # server's version gate stays aligned)
        self.tracer = None
        self.base_master_port: Optional[int] = None
        self.resume_count = 0

    @property
    def ps_enabled(self) -> bool:
        assert self.cfg is not None
        return self.cfg.num_servers > 0 or self.cfg.force_distributed


_state = _State()


def _default_backend() -> str:
    # BPS_BACKEND=gloo lets a multi-process run share one GPU for compute
    # while collectives go over gloo (RCCL refuses two ranks on one
    # device) — the single-box dry-run vehicle for the 8-GPU launch shape
    forced = os.environ.get("BPS_BACKEND", "").lower()
    if forced in ("gloo", "nccl"):
        return forced
    return "nccl" if torch.cuda.is_available() else "gloo"


def init(lazy: bool = True, backend: Optional[str] = None) -> None:
    """Initialize byteps_amd.  Safe to call more than once.

    Reads cluster topology from the env (torchrun's RANK/WORLD_SIZE/
    LOCAL_RANK or the bpslaunch BPS_* variables), initializes the
    ``torch.distributed`` process group if needed, and pins this process
    to its GPU.  ``lazy`` is accepted for API parity with the reference
    (byteps/common/__init__.py:61-67); initialization here is always eager
    because process-group setup must happen before the first collective.
    """
    if _state.initialized:
        return
    cfg = Config.from_env()
    _state.cfg = cfg

    rank = int(os.environ.get("RANK", os.environ.get("BPS_RANK", "0")))
    size = int(os.environ.get("WORLD_SIZE", os.environ.get("BPS_SIZE", "1")))
    local_rank = int(os.environ.get("LOCAL_RANK", str(cfg.local_rank)))
    local_size = int(os.environ.get(
        "LOCAL_WORLD_SIZE", str(max(cfg.local_size, 1))))
    # single-node launches: local == global unless told otherwise
    if size > 1 and "LOCAL_WORLD_SIZE" not in os.environ and cfg.local_size <= 1:
        local_size = size
    _state.rank, _state.size = rank, size
    _state.local_rank, _state.local_size = local_rank, local_size
    cfg.local_rank, cfg.local_size = local_rank, local_size

    if torch.cuda.is_available():
        # one process per GPU (identity on a full node); modulo keeps
        # oversubscribed dry-runs alive on smaller boxes (e.g. 2 ranks
        # sharing the single gpurun GPU with gloo collectives)
        dev_index = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(dev_index)
        _state.device = torch.device("cuda", dev_index)
    else:
        _state.device = torch.device("cpu")

    if size > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        if _state.base_master_port is None:
            _state.base_master_port = int(os.environ["MASTER_PORT"])
        dist.init_process_group(
            backend=backend or _default_backend(),
            rank=rank,
            world_size=size,
            timeout=datetime.timedelta(seconds=300),
        )
        _state.owns_process_group = True

    if cfg.num_servers > 0:
        _state.assigner = ServerAssigner(cfg.num_servers)

    if cfg.trace_on:
        from .tracing import Tracer
        _state.tracer = Tracer(cfg, rank)

    _state.initialized = True
    log.debug(
        "byteps_amd init: rank %d/%d local %d/%d device %s ps=%s",
        rank, size, local_rank, local_size, _state.device, _state.ps_enabled)


def shutdown() -> None:
    """Tear down engine threads and the process group
    (reference byteps_shutdown, common/operations.cc:90-94)."""
    if not _state.initialized:
        return
    from ..torch import engine as torch_engine
    torch_engine._shutdown_engine()
    if _state.kv is not None:
        try:
            _state.kv.close()
        except Exception:
            pass
        _state.kv = None
    if _state.tracer is not None:
        _state.tracer.flush()
    if _state.owns_process_group and dist.is_initialized():
        dist.destroy_process_group()
        _state.owns_process_group = False
    _state.initialized = False


def suspend() -> None:
    """Elastic training: release communication state but keep the declared
    tensor table so keys survive (reference byteps_suspend,
    common/operations.cc:96-107)."""
    if not _state.initialized:
        return
    from ..torch import engine as torch_engine
    torch_engine._suspend_engines()
    if _state.kv is not None:
        try:
            _state.kv.close()
        except Exception:
            pass
        _state.kv = None
    if _state.tracer is not None:
        _state.tracer.flush()
    if _state.owns_process_group and dist.is_initialized():
        dist.destroy_process_group()
        _state.owns_process_group = False
    _state.initialized = False


def resume(num_workers: int, num_servers: int,
           global_rank: Optional[int] = None) -> None:
    """Elastic training: re-init with a new cluster shape; re-declare every
    tensor in original order so keys are stable (reference byteps_resume,
    common/operations.cc:109-119)."""
    os.environ["BPS_NUM_WORKER"] = str(num_workers)
    os.environ["BPS_NUM_SERVER"] = str(num_servers)
    if global_rank is not None:
        os.environ["RANK"] = str(global_rank)
    # rotate the rendezvous port deterministically: re-binding the old
    # MASTER_PORT races the previous TCPStore's teardown (observed hangs
    # under load); every rank derives the same next port
    _state.resume_count += 1
    if _state.base_master_port is not None:
        os.environ["MASTER_PORT"] = str(
            _state.base_master_port + _state.resume_count)
    _state.registry.redeclare_all()
    init()
    from ..torch import engine as torch_engine
    torch_engine._resume_engines()


def initialized() -> bool:
    return _state.initialized


def _require_init() -> None:
    if not _state.initialized:
        raise RuntimeError(
            "byteps_amd has not been initialized; call byteps_amd.torch.init() first")


def rank() -> int:
    _require_init()
    return _state.rank


def size() -> int:
    _require_init()
    return _state.size


def local_rank() -> int:
    _require_init()
    return _state.local_rank


def local_size() -> int:
    _require_init()
    return _state.local_size


def device() -> torch.device:
    _require_init()
    assert _state.device is not None
    return _state.device


def get_config() -> Config:
    _require_init()
    assert _state.cfg is not None
    return _state.cfg


atexit.register(shutdown)
