"""Environment-variable configuration catalogue.

Re-creates the reference's pure-env flag system (reference docs/env.md:1-151,
common/global.cc:105-281) with ``BPS_*`` names; the reference's ``BYTEPS_*``
and ``DMLC_*`` names are accepted as aliases so existing launch scripts keep
working.  Every knob is read lazily so tests can monkeypatch ``os.environ``.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List, Optional

_TRUE = {"1", "true", "yes", "on"}


def _env(name: str, *aliases: str, default: Optional[str] = None) -> Optional[str]:
    for key in (name, *aliases):
        val = os.environ.get(key)
        if val is not None and val != "":
            return val
    return default


def env_str(name: str, *aliases: str, default: str = "") -> str:
    v = _env(name, *aliases, default=default)
    return v if v is not None else default


def env_int(name: str, *aliases: str, default: int = 0) -> int:
    v = _env(name, *aliases)
    return int(v) if v is not None else default


def env_bool(name: str, *aliases: str, default: bool = False) -> bool:
    v = _env(name, *aliases)
    if v is None:
        return default
    return v.strip().lower() in _TRUE


# Page size used for partition rounding (reference common/global.cc:134-144
# rounds partition bytes to local_size * 4096).
PAGE_SIZE = 4096

# Default partition/bucket bytes.  The reference used 4 MB sized for PCIe +
# NIC MTUs (common/global.cc:42); xGMI links run ≈153 GB/s per link so ring
# collectives need larger messages — 16 MiB buckets keep the 7-link ring in
# its bandwidth regime while still giving backward/comm overlap granularity.
DEFAULT_PARTITION_BYTES = 16 * 1024 * 1024


@dataclass
class Config:
    """Snapshot of all knobs. Construct via :func:`Config.from_env`."""

    # cluster topology -----------------------------------------------------
    role: str = "worker"                    # worker | server | scheduler
    num_workers: int = 1                    # DMLC_NUM_WORKER
    num_servers: int = 0                    # DMLC_NUM_SERVER (0 → pure RCCL)
    root_uri: str = "127.0.0.1"             # scheduler host
    root_port: int = 9000                   # scheduler port
    # per-node layout
    local_rank: int = 0
    local_size: int = 1
    worker_id: int = 0                      # node index among worker nodes

    # engine knobs ----------------------------------------------------------
    partition_bytes: int = DEFAULT_PARTITION_BYTES
    scheduling_credit: int = 0              # byte credit; 0 → unlimited
    priority_enabled: bool = True
    pin_memory: bool = True
    force_distributed: bool = False         # force PS path even on 1 node
    enable_async: bool = False              # asynchronous PS training
    enable_mixed_mode: bool = False         # colocated + standalone servers
    compressor_threads: int = 8
    min_compress_bytes: int = 65536

    # server knobs ----------------------------------------------------------
    server_engine_threads: int = 4
    server_enable_schedule: bool = False

    # tracing / telemetry ---------------------------------------------------
    trace_on: bool = False
    trace_dir: str = "./traces"
    trace_start_step: int = 10
    trace_end_step: int = 20
    debug_sample_tensor: str = ""
    log_level: str = "INFO"
    telemetry_on: bool = False

    # reduction -------------------------------------------------------------
    reduce_dtype: str = ""                  # "" → same as grad; "fp32" to upcast

    @staticmethod
    def from_env() -> "Config":
        c = Config()
        c.role = env_str("BPS_ROLE", "DMLC_ROLE", default="worker").lower()
        c.num_workers = env_int("BPS_NUM_WORKER", "DMLC_NUM_WORKER", default=1)
        c.num_servers = env_int("BPS_NUM_SERVER", "DMLC_NUM_SERVER", default=0)
        c.root_uri = env_str("BPS_ROOT_URI", "DMLC_PS_ROOT_URI", default="127.0.0.1")
        c.root_port = env_int("BPS_ROOT_PORT", "DMLC_PS_ROOT_PORT", default=9000)
        c.local_rank = env_int("BPS_LOCAL_RANK", "BYTEPS_LOCAL_RANK", "LOCAL_RANK", default=0)
        c.local_size = env_int("BPS_LOCAL_SIZE", "BYTEPS_LOCAL_SIZE", "LOCAL_WORLD_SIZE", default=1)
        c.worker_id = env_int("BPS_WORKER_ID", "DMLC_WORKER_ID", default=0)

        ps_mode = c.num_servers > 0 or env_bool(
            "BPS_FORCE_DISTRIBUTED", "BYTEPS_FORCE_DISTRIBUTED",
            default=False)
        c.partition_bytes = env_int(
            "BPS_PARTITION_BYTES", "BYTEPS_PARTITION_BYTES",
            # PS mode: 32 MiB halves the per-bucket server round-trip
            # count (same-box sweep, profiles/MEASUREMENTS.md); pure-RCCL
            # stays at 16 MiB (sized for xGMI ring overlap granularity)
            default=(32 * 1024 * 1024) if ps_mode
            else DEFAULT_PARTITION_BYTES)
        # round up to a multiple of local_size * PAGE_SIZE so per-rank shards
        # of reduce-scatter stay page aligned (mirrors reference rounding,
        # common/global.cc:134-144)
        align = max(1, c.local_size) * PAGE_SIZE
        c.partition_bytes = ((c.partition_bytes + align - 1) // align) * align

        c.scheduling_credit = env_int(
            "BPS_SCHEDULING_CREDIT", "BYTEPS_SCHEDULING_CREDIT", default=0)
        c.priority_enabled = env_bool(
            "BPS_PRIORITY", "BYTEPS_PRIORITY", default=True)
        c.force_distributed = env_bool(
            "BPS_FORCE_DISTRIBUTED", "BYTEPS_FORCE_DISTRIBUTED", default=False)
        c.enable_async = env_bool(
            "BPS_ENABLE_ASYNC", "BYTEPS_ENABLE_ASYNC", default=False)
        c.enable_mixed_mode = env_bool(
            "BPS_ENABLE_MIXED_MODE", "BYTEPS_ENABLE_MIXED_MODE", default=False)
        c.compressor_threads = env_int(
            "BPS_COMPRESSOR_THREADS", "BYTEPS_THREADPOOL_SIZE", default=8)
        c.min_compress_bytes = env_int(
            "BPS_MIN_COMPRESS_BYTES", "BYTEPS_MIN_COMPRESS_BYTES", default=65536)

        c.server_engine_threads = env_int(
            "BPS_SERVER_ENGINE_THREAD", "BYTEPS_SERVER_ENGINE_THREAD", default=4)
        c.server_enable_schedule = env_bool(
            "BPS_SERVER_ENABLE_SCHEDULE", "BYTEPS_SERVER_ENABLE_SCHEDULE",
            default=False)

        c.trace_on = env_bool("BPS_TRACE_ON", "BYTEPS_TRACE_ON", default=False)
        c.trace_dir = env_str("BPS_TRACE_DIR", "BYTEPS_TRACE_DIR", default="./traces")
        c.trace_start_step = env_int(
            "BPS_TRACE_START_STEP", "BYTEPS_TRACE_START_STEP", default=10)
        c.trace_end_step = env_int(
            "BPS_TRACE_END_STEP", "BYTEPS_TRACE_END_STEP", default=20)
        c.debug_sample_tensor = env_str(
            "BPS_DEBUG_SAMPLE_TENSOR", "BYTEPS_DEBUG_SAMPLE_TENSOR", default="")
        c.log_level = env_str("BPS_LOG_LEVEL", "BYTEPS_LOG_LEVEL", default="INFO")
        c.telemetry_on = env_bool(
            "BPS_TELEMETRY_ON", "BYTEPS_TELEMETRY_ON", default=False)
        c.reduce_dtype = env_str("BPS_REDUCE_DTYPE", default="")
        return c


def server_addresses(cfg: Config) -> List[str]:
    """Static server list: ``BPS_SERVER_URIS=host:port,host:port``.

    When unset, servers register with the scheduler at ``root_uri:root_port``
    and workers learn the list from it (rendezvous implemented in the C++ KV
    layer; reference equivalent: ps-lite Postoffice via DMLC_PS_ROOT_URI,
    reference common/global.cc:283-297).
    """
    raw = env_str("BPS_SERVER_URIS", default="")
    if not raw:
        return []
    return [s.strip() for s in raw.split(",") if s.strip()]
