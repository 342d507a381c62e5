"""PS server process entry (reference byteps/server/__init__.py:1-27 —
which ctypes-loaded the server .so on import; here the server is started
explicitly via ``python -m byteps_amd.server``)."""

from __future__ import annotations

import os
import signal
import socket
import threading

from ..common.config import Config
from ..common.logging_util import get_logger

log = get_logger()


def run_server(port: int = 0, block: bool = True):
    """Start the native PS server; returns the Server object (its ``port``
    attribute reports the bound port when 0 was requested)."""
    from ..ops import core
    cfg = Config.from_env()
    port = port or int(os.environ.get("BPS_SERVER_PORT", "9100"))
    srv = core().Server(port, cfg.server_engine_threads,
                        cfg.server_enable_schedule)
    srv.start()
    log.info("byteps_amd server listening on :%d (%d engine threads, "
             "schedule=%s)", srv.port, cfg.server_engine_threads,
             cfg.server_enable_schedule)
    # announce to the scheduler if one is configured (reference ps-lite
    # scheduler rendezvous; static BPS_SERVER_URIS needs no announcement)
    if os.environ.get("BPS_ROOT_URI", os.environ.get("DMLC_PS_ROOT_URI")):
        try:
            from ..launcher.scheduler import register_server
            host = os.environ.get("BPS_SERVER_HOST") or \
                socket.gethostbyname(socket.gethostname())
            register_server(host, srv.port, cfg.root_uri, cfg.root_port)
            log.info("registered %s:%d with scheduler %s:%d", host,
                     srv.port, cfg.root_uri, cfg.root_port)
        except Exception as e:
            log.warning("scheduler registration failed: %s", e)
    if block:
        stop = threading.Event()
        signal.signal(signal.SIGTERM, lambda *a: stop.set())
        signal.signal(signal.SIGINT, lambda *a: stop.set())
        stop.wait()
        srv.stop()
    return srv
