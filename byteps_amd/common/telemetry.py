"""Push-pull throughput telemetry (reference ``PushPullSpeed``,
common/global.cc:697-752): a sliding 10-second window of communicated
bytes, queryable as ``get_pushpull_speed() -> (timestamp_ms, MB/s)``."""

from __future__ import annotations

import collections
import threading
import time
from typing import Tuple

_WINDOW_SEC = 10.0


class SpeedMeter:
    def __init__(self) -> None:
        self._events = collections.deque()   # (t, nbytes)
        self._lock = threading.Lock()

    def record(self, nbytes: int) -> None:
        now = time.monotonic()
        with self._lock:
            self._events.append((now, nbytes))
            self._trim(now)

    def _trim(self, now: float) -> None:
        while self._events and now - self._events[0][0] > _WINDOW_SEC:
            self._events.popleft()

    def speed(self) -> Tuple[float, float]:
        """Returns (unix timestamp ms, MB/s over the window)."""
        now = time.monotonic()
        with self._lock:
            self._trim(now)
            total = sum(n for _, n in self._events)
            span = (now - self._events[0][0]) if self._events else 0.0
        mbps = (total / 1e6) / span if span > 0 else 0.0
        return time.time() * 1000.0, mbps


_meter = SpeedMeter()


def record(nbytes: int) -> None:
    _meter.record(nbytes)


def get_pushpull_speed() -> Tuple[float, float]:
    """Reference byteps_get_pushpull_speed (common/operations.cc:131-136)."""
    return _meter.speed()
