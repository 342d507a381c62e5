"""Multi-process test harness: run a function on N ranks over gloo
(world_size > 1 on CPU — mirrors the reference's single-machine
forced-distributed test pattern, reference tests/meta_test.py:27-85)."""

from __future__ import annotations

import os
import pickle
import socket
import traceback

import torch.multiprocessing as mp


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _entry(rank: int, world: int, port: int, fn, args, q, extra_env):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["LOCAL_WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    for k, v in (extra_env or {}).items():
        os.environ[k] = str(v)
    try:
        result = fn(rank, world, *args)
        q.put((rank, "ok", pickle.dumps(result)))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))
        raise


def run_in_processes(fn, world: int, *args, extra_env=None, timeout: float = 300.0):
    """Spawn ``world`` processes running ``fn(rank, world, *args)`` with a
    gloo-compatible rendezvous on 127.0.0.1.  Returns [result_rank0, ...].
    Raises on any rank failure."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    procs = []
    for r in range(world):
        p = ctx.Process(target=_entry,
                        args=(r, world, port, fn, args, q, extra_env))
        p.start()
        procs.append(p)
    results = {}
    errors = []
    import queue as _queue
    for _ in range(world):
        try:
            rank, status, payload = q.get(timeout=timeout)
        except _queue.Empty:
            errors.append((-1, "harness timeout: a rank hung (deadlock?)"))
            break
        if status == "ok":
            results[rank] = pickle.loads(payload)
        else:
            errors.append((rank, payload))
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
            p.join(5)
    if errors:
        raise AssertionError(
            "\n".join("rank %d failed:\n%s" % (r, tb) for r, tb in errors))
    return [results[r] for r in range(world)]
