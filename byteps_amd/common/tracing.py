"""Chrome-trace profiling (reference common/global.cc:469-564,
docs/timeline.md).

Activated by ``BPS_TRACE_ON=1`` with ``BPS_TRACE_DIR``,
``BPS_TRACE_START_STEP``, ``BPS_TRACE_END_STEP``.  Emits
``<trace_dir>/<rank>/comm.json`` with one ``pid`` per communication unit
(``Comm.<key>``) — loadable in chrome://tracing or Perfetto, same layout
as the reference's output.
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Tuple


class Tracer:
    def __init__(self, cfg, rank: int) -> None:
        self.dir = os.path.join(cfg.trace_dir, str(rank))
        self.start_step = cfg.trace_start_step
        self.end_step = cfg.trace_end_step
        self.events: List[dict] = []
        self._open: Dict[Tuple[int, str], float] = {}
        self._lock = threading.Lock()
        self._flushed = False
        os.makedirs(self.dir, exist_ok=True)

    def _active(self, step: int) -> bool:
        return self.start_step <= step <= self.end_step

    def begin(self, key: int, stage: str, step: int) -> None:
        if not self._active(step):
            return
        with self._lock:
            self._open[(key, stage)] = time.perf_counter_ns() / 1000.0

    def end(self, key: int, stage: str, step: int) -> None:
        if not self._active(step):
            return
        with self._lock:
            ts = self._open.pop((key, stage), None)
            if ts is None:
                return
            now = time.perf_counter_ns() / 1000.0
            self.events.append({
                "name": stage,
                "ph": "X",
                "pid": "Comm.%d" % key,
                "tid": stage,
                "ts": ts,
                "dur": now - ts,
                "args": {"step": step},
            })
        if step == self.end_step:
            self.flush()

    def flush(self) -> None:
        with self._lock:
            if self._flushed and not self.events:
                return
            path = os.path.join(self.dir, "comm.json")
            with open(path, "w") as f:
                json.dump({"traceEvents": self.events,
                           "displayTimeUnit": "ms"}, f)
            self._flushed = True
