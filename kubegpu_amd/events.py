"""Placement event trace — ring buffer of recent scheduling decisions.

The reference's only observability is leveled logging.
Alongside metrics (metrics.py), this records one structured event per
schedule/release so an operator can answer "where did pod X land and
why" after the fact:

    {"ts", "event", "pod", "node", "gpus", "latency_ms",
     "predicted_ring_gbps"}

Bounded in-memory ring (default 1024). Set ``KUBEGPU_EVENT_LOG=<path>``
to additionally append each event as a JSON line to a file (best-effort;
I/O errors never affect scheduling).
"""

from __future__ import annotations

import json
import os
import threading
import time
from collections import deque
from typing import Deque, Dict, List, Optional


class EventTrace:
    def __init__(self, capacity: int = 1024, path: Optional[str] = None):
        self._lock = threading.Lock()
        self._ring: Deque[Dict] = deque(maxlen=capacity)
        self._path = path if path is not None else os.environ.get("KUBEGPU_EVENT_LOG")

    def record(self, event: str, **fields) -> None:
        rec = {"ts": round(time.time(), 3), "event": event, **fields}
        with self._lock:
            self._ring.append(rec)
        if self._path:
            try:
                with open(self._path, "a") as f:
                    f.write(json.dumps(rec) + "\n")
            except OSError:
                pass

    def recent(self, n: int = 100) -> List[Dict]:
        with self._lock:
            return list(self._ring)[-n:]

    def clear(self) -> None:
        with self._lock:
            self._ring.clear()


EVENTS = EventTrace()
