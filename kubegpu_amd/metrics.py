"""Prometheus metrics (the reference has none; the build
adds pod-schedule latency and xGMI utilization per BASELINE.json).

Soft dependency: when prometheus_client is unavailable everything
degrades to in-memory counters so the control plane never hard-requires
a metrics stack.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

try:
    from prometheus_client import (
        CollectorRegistry,
        Counter,
        Gauge,
        Histogram,
        start_http_server,
    )

    _HAVE_PROM = True
except ImportError:  # pragma: no cover
    _HAVE_PROM = False


class Metrics:
    def __init__(self, registry=None) -> None:
        self._lock = threading.Lock()
        self.schedule_latencies: List[float] = []
        self.allocations = 0
        self.failures = 0
        self.gpu_health: Dict[str, bool] = {}
        self.gpu_in_use: Dict[str, bool] = {}
        if _HAVE_PROM:
            # per-instance registry: multiple Metrics objects (tests,
            # embedded schedulers) must not collide in the global one
            self.registry = registry if registry is not None else CollectorRegistry()
            self._h = Histogram(
                "kubegpu_amd_schedule_latency_seconds",
                "pod schedule latency",
                buckets=(1e-5, 1e-4, 5e-4, 1e-3, 5e-3, 1e-2, 0.1, 1.0),
                registry=self.registry,
            )
            self._alloc = Counter(
                "kubegpu_amd_allocations_total", "pod GPU allocations",
                registry=self.registry,
            )
            self._fail = Counter(
                "kubegpu_amd_schedule_failures_total", "schedule failures",
                registry=self.registry,
            )
            self._xgmi = Gauge(
                "kubegpu_amd_xgmi_link_gbps", "last probed xGMI ring bandwidth GB/s",
                registry=self.registry,
            )
            self._gpu_health = Gauge(
                "kubegpu_amd_gpu_healthy",
                "1 = GPU healthy, 0 = unhealthy (ECC/vanished)",
                ["uuid"],
                registry=self.registry,
            )
            self._gpu_ecc = Gauge(
                "kubegpu_amd_gpu_ecc_uncorrectable",
                "accumulated uncorrectable ECC errors",
                ["uuid"],
                registry=self.registry,
            )
            self._gpu_in_use = Gauge(
                "kubegpu_amd_gpu_in_use",
                "1 = GPU held by a live allocation (manager path)",
                ["uuid"],
                registry=self.registry,
            )

    def observe_schedule(self, seconds: float) -> None:
        with self._lock:
            self.schedule_latencies.append(seconds)
        if _HAVE_PROM:
            self._h.observe(seconds)

    def inc_allocation(self) -> None:
        with self._lock:
            self.allocations += 1
        if _HAVE_PROM:
            self._alloc.inc()

    def inc_failure(self) -> None:
        with self._lock:
            self.failures += 1
        if _HAVE_PROM:
            self._fail.inc()

    def set_xgmi_gbps(self, gbps: float) -> None:
        if _HAVE_PROM:
            self._xgmi.set(gbps)

    def set_gpu_health(self, uuid: str, healthy: bool, ecc_uncorrectable: int = 0) -> None:
        with self._lock:
            self.gpu_health[uuid] = healthy
        if _HAVE_PROM:
            self._gpu_health.labels(uuid=uuid).set(1.0 if healthy else 0.0)
            self._gpu_ecc.labels(uuid=uuid).set(float(ecc_uncorrectable))

    def set_gpu_in_use(self, uuid: str, in_use: bool) -> None:
        with self._lock:
            self.gpu_in_use[uuid] = in_use
        if _HAVE_PROM:
            self._gpu_in_use.labels(uuid=uuid).set(1.0 if in_use else 0.0)

    def percentile(self, q: float) -> Optional[float]:
        with self._lock:
            if not self.schedule_latencies:
                return None
            data = sorted(self.schedule_latencies)
            idx = min(len(data) - 1, int(q * len(data)))
            return data[idx]

    def serve(self, port: int = 9400) -> bool:
        if _HAVE_PROM:
            start_http_server(port, registry=self.registry)
            return True
        return False


METRICS = Metrics()
