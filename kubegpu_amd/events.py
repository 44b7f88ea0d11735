"""Placement event trace — ring buffer of recent scheduling decisions.

The reference's only observability is leveled logging.
Alongside metrics (metrics.py), this records one structured event per
schedule/release so an operator can answer "where did pod X land and
why" after the fact:

    {"ts", "event", "pod", "node", "gpus", "latency_ms",
     "predicted_ring_gbps"}

Bounded in-memory ring (default 1024). Set ``KUBEGPU_EVENT_LOG=<path>``
to additionally append each event as a JSON line to a file.  File I/O
happens on a background flusher thread, never inline in record():
record() is called under the cluster scheduling lock, and disk latency
must not serialize scheduling.  Best-effort: I/O
errors never affect scheduling; flush() drains synchronously for tests
and shutdown.
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
        # pending file lines; flushed off-thread (bounded so a dead disk
        # cannot grow memory without limit)
        self._pending: Deque[str] = deque(maxlen=8192)
        self._flush_wake = threading.Event()
        self._flusher: Optional[threading.Thread] = None
        # serializes grab+write so concurrent drains (flusher thread +
        # an explicit flush()) cannot append batches out of order
        self._drain_lock = threading.Lock()

    def record(self, event: str, **fields) -> None:
        rec = {"ts": round(time.time(), 3), "event": event, **fields}
        path = self._path or os.environ.get("KUBEGPU_EVENT_LOG")
        with self._lock:
            self._ring.append(rec)
            if path:
                self._path = path
                self._pending.append(json.dumps(rec))
                self._ensure_flusher()
        if path:
            self._flush_wake.set()

    def _ensure_flusher(self) -> None:
        # called under self._lock
        if self._flusher is None or not self._flusher.is_alive():
            self._flusher = threading.Thread(
                target=self._flush_loop, name="event-flush", daemon=True
            )
            self._flusher.start()

    def _flush_loop(self) -> None:
        while True:
            self._flush_wake.wait(timeout=5.0)
            self._flush_wake.clear()
            self._drain()

    def _drain(self) -> None:
        with self._drain_lock:
            with self._lock:
                if not self._pending or not self._path:
                    return
                lines = list(self._pending)
                self._pending.clear()
                path = self._path
            try:
                with open(path, "a") as f:
                    f.write("\n".join(lines) + "\n")
            except OSError:
                pass  # best-effort: never affect scheduling

    def flush(self, timeout_s: float = 5.0) -> None:
        """Synchronously drain pending file writes (tests/shutdown)."""
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            with self._lock:
                empty = not self._pending
            if empty:
                return
            self._drain()

    def recent(self, n: int = 100) -> List[Dict]:
        with self._lock:
            return list(self._ring)[-n:]

    def clear(self) -> None:
        with self._lock:
            self._ring.clear()


EVENTS = EventTrace()
