"""Cluster rendezvous scheduler (reference: ps-lite's scheduler node via
DMLC_PS_ROOT_URI/PORT).

Control-plane only (never on the data path), so it is plain Python TCP:
servers register ``{"role": "server", "host": h, "port": p}``; workers
send ``{"role": "worker"}`` and block until ``num_servers`` servers have
registered, then receive the full server list.  Also provides a named
barrier service across processes.

Protocol: one JSON line per request, one JSON line per response.
"""

from __future__ import annotations

import json
import os
import socket
import socketserver
import threading
from typing import Dict, List, Tuple

from ..common.logging_util import get_logger

log = get_logger()


class _State:
    def __init__(self, num_servers: int):
        self.num_servers = num_servers
        self.servers: List[str] = []
        self.cv = threading.Condition()
        self.barriers: Dict[str, Tuple[int, int]] = {}   # name → (count, gen)


class _Handler(socketserver.StreamRequestHandler):
    def handle(self):
        st: _State = self.server.state  # type: ignore
        line = self.rfile.readline()
        if not line:
            return
        try:
            msg = json.loads(line)
        except json.JSONDecodeError:
            return
        role = msg.get("role")
        if role == "server":
            uri = "%s:%d" % (msg["host"], int(msg["port"]))
            with st.cv:
                if uri not in st.servers:
                    st.servers.append(uri)
                st.cv.notify_all()
            self._reply({"ok": True})
        elif role == "worker":
            with st.cv:
                st.cv.wait_for(lambda: len(st.servers) >= st.num_servers,
                               timeout=300)
                servers = list(st.servers)
            self._reply({"ok": len(servers) >= st.num_servers,
                         "servers": servers})
        elif role == "barrier":
            name = msg.get("name", "default")
            count = int(msg["count"])
            with st.cv:
                n, gen = st.barriers.get(name, (0, 0))
                n += 1
                if n >= count:
                    st.barriers[name] = (0, gen + 1)
                    st.cv.notify_all()
                else:
                    st.barriers[name] = (n, gen)
                    st.cv.wait_for(
                        lambda: st.barriers.get(name, (0, 0))[1] > gen,
                        timeout=300)
            self._reply({"ok": True})
        else:
            self._reply({"ok": False, "error": "unknown role"})

    def _reply(self, obj) -> None:
        self.wfile.write((json.dumps(obj) + "\n").encode())


class Scheduler:
    def __init__(self, port: int = 0, num_servers: int = 0):
        num_servers = num_servers or int(os.environ.get(
            "BPS_NUM_SERVER", os.environ.get("DMLC_NUM_SERVER", "0")))

        class _Srv(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self._srv = _Srv(("0.0.0.0", port), _Handler)
        self._srv.state = _State(num_servers)  # type: ignore
        self.port = self._srv.server_address[1]
        self._thread = threading.Thread(
            target=self._srv.serve_forever, daemon=True,
            name="bps-scheduler")

    def start(self) -> "Scheduler":
        self._thread.start()
        log.info("byteps_amd scheduler on :%d", self.port)
        return self

    def stop(self) -> None:
        self._srv.shutdown()
        self._srv.server_close()


def _rpc(host: str, port: int, msg: dict, timeout: float = 300.0) -> dict:
    with socket.create_connection((host, port), timeout=timeout) as s:
        s.sendall((json.dumps(msg) + "\n").encode())
        f = s.makefile()
        return json.loads(f.readline())


def register_server(host: str, port: int, sched_host: str,
                    sched_port: int) -> None:
    _rpc(sched_host, sched_port,
         {"role": "server", "host": host, "port": port})


def discover_servers(sched_host: str, sched_port: int) -> List[str]:
    r = _rpc(sched_host, sched_port, {"role": "worker"})
    if not r.get("ok"):
        raise RuntimeError("scheduler rendezvous failed: %r" % r)
    return r["servers"]


def barrier(sched_host: str, sched_port: int, count: int,
            name: str = "default") -> None:
    _rpc(sched_host, sched_port,
         {"role": "barrier", "name": name, "count": count})
