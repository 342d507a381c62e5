#!/usr/bin/env python3
"""bpslaunch — role-based cluster launcher (reference
launcher/launch.py:234-277).

Roles via ``BPS_ROLE`` (alias ``DMLC_ROLE``):
  worker    spawn one training process per visible GPU with
            RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* set, CPU-affinity
            partitioned NUMA-style across local ranks (reference
            launcher/launch.py:49-199 numactl pinning; no numactl
            dependency here — uses sched_setaffinity)
  server    run the native PS server (python -m byteps_amd.server)
  scheduler run the rendezvous scheduler (byteps_amd.launcher.scheduler)
            at BPS_ROOT_PORT; workers fall back to it when
            BPS_SERVER_URIS is not set

Usage:  bpslaunch python3 train.py [args...]
"""

from __future__ import annotations

import os
import signal
import subprocess
import sys
from typing import List


def visible_gpus() -> int:
    env = os.environ.get("HIP_VISIBLE_DEVICES",
                         os.environ.get("CUDA_VISIBLE_DEVICES"))
    if env is not None:
        return len([x for x in env.split(",") if x.strip() != ""])
    try:
        import torch
        return max(torch.cuda.device_count(), 0)
    except Exception:
        return 0


def cpu_ranges(local_size: int) -> List[List[int]]:
    """Partition visible CPU cores contiguously across local ranks
    (approximates the reference's NUMA-aware allocation,
    launcher/launch.py:49-141; contiguous blocks keep each rank inside
    one NUMA domain on standard topologies)."""
    cpus = sorted(os.sched_getaffinity(0))
    if local_size <= 0:
        return []
    per = max(1, len(cpus) // local_size)
    out = []
    for i in range(local_size):
        beg = i * per
        end = len(cpus) if i == local_size - 1 else (i + 1) * per
        out.append(cpus[beg:end] or cpus)
    return out


def launch_worker(cmd: List[str]) -> int:
    local_size = int(os.environ.get("BPS_LOCAL_SIZE", "0")) or \
        visible_gpus() or 1
    num_nodes = int(os.environ.get(
        "BPS_NUM_WORKER", os.environ.get("DMLC_NUM_WORKER", "1")))
    node_id = int(os.environ.get(
        "BPS_WORKER_ID", os.environ.get("DMLC_WORKER_ID", "0")))
    world = num_nodes * local_size
    master = os.environ.get(
        "MASTER_ADDR",
        os.environ.get("BPS_ROOT_URI",
                       os.environ.get("DMLC_PS_ROOT_URI", "127.0.0.1")))
    port = os.environ.get("MASTER_PORT", "29500")

    ranges = cpu_ranges(local_size)
    procs = []
    for lr in range(local_size):
        env = dict(os.environ)
        env.update({
            "RANK": str(node_id * local_size + lr),
            "LOCAL_RANK": str(lr),
            "WORLD_SIZE": str(world),
            "LOCAL_WORLD_SIZE": str(local_size),
            "MASTER_ADDR": master,
            "MASTER_PORT": port,
            "BPS_LOCAL_RANK": str(lr),
            "BPS_LOCAL_SIZE": str(local_size),
        })
        if os.environ.get("BPS_ENABLE_GDB", "0") == "1":
            # reference BYTEPS_ENABLE_GDB (launcher/launch.py:165-168)
            full = ["gdb", "-ex", "run", "-ex", "bt", "-batch", "--args"] + cmd
        else:
            full = cmd
        p = subprocess.Popen(full, env=env)
        if ranges:
            try:
                os.sched_setaffinity(p.pid, ranges[lr])
            except OSError:
                pass
        procs.append(p)

    def forward(sig, _frame):
        for p in procs:
            p.send_signal(sig)

    signal.signal(signal.SIGTERM, forward)
    rc = 0
    for p in procs:
        p.wait()
        rc = rc or p.returncode
    return rc


def main() -> int:
    role = os.environ.get(
        "BPS_ROLE", os.environ.get("DMLC_ROLE", "worker")).lower()
    if role == "worker":
        cmd = sys.argv[1:]
        if not cmd:
            print("usage: bpslaunch <command...>", file=sys.stderr)
            return 2
        return launch_worker(cmd)
    if role == "server":
        from byteps_amd.server import run_server
        run_server()
        return 0
    if role == "scheduler":
        from byteps_amd.launcher.scheduler import Scheduler
        port = int(os.environ.get(
            "BPS_ROOT_PORT", os.environ.get("DMLC_PS_ROOT_PORT", "9000")))
        sched = Scheduler(port=port).start()
        try:
            signal.pause()
        finally:
            sched.stop()
        return 0
    print("unknown BPS_ROLE %r" % role, file=sys.stderr)
    return 2


if __name__ == "__main__":
    sys.exit(main())
