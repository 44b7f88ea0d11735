"""Cluster — the end-to-end scheduling flow the two-repo reference split.

Emulates the KubeDevice core scheduler loop: for a pod,
run the device scheduler's fit predicate per candidate node, pick the
best node, run pod_allocate, have the group-scheduler core bind concrete
GPUs into AllocateFrom, commit accounting, and (when a device-plugin
manager is attached for the node) produce the container's device nodes +
env at create time.

Node choice is xGMI-aware: each candidate is scored by the ring
bottleneck bandwidth of the subset it would give the pod, with remaining
xGMI connectivity as tie-breaker (anti-fragmentation bin-packing —
BASELINE.json configs 3-4).
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..api.types import NodeInfo, PodInfo
from ..deviceplugin.manager import AMDGPUManager
from ..discovery import GpusInfo
from ..events import EVENTS
from ..scheduler.scheduler import AMDGPUScheduler
from ..scheduler.translate import SchedulingError
from .group_scheduler import GroupScheduler


@dataclass
class ScheduleResult:
    pod_name: str
    node_name: str
    uuids: List[str]
    latency_s: float


class Cluster:
    """In-process cluster: device scheduler + group core + node plugins."""

    def __init__(self, scheduler: Optional[AMDGPUScheduler] = None,
                 policy: str = "xgmi"):
        import threading

        # One pod binds at a time (the reference's core serializes
        # scheduler-plugin calls; without this the trial
        # bind and the commit race between threads and two pods can
        # claim the same GPUs).  RLock: the class index is mutated by
        # add/remove/release/reindex under the same lock schedule()
        # holds while iterating it.
        self._sched_lock = threading.RLock()
        self.scheduler = scheduler or AMDGPUScheduler()
        self.core = GroupScheduler(policy=policy)
        self.policy = policy
        self.node_infos: Dict[str, NodeInfo] = {}
        self.managers: Dict[str, AMDGPUManager] = {}
        # Incremental equivalence classes: signature -> node names.  A
        # node's signature only changes at the events this class owns
        # (bind commit, release, node add/remove), so per-pod work is
        # O(distinct states), not O(nodes) — flat out to multi-thousand
        #-node fleets (docs/ROADMAP.md #5).
        from sortedcontainers import SortedSet

        self._sorted_set = SortedSet
        self._node_sig: Dict[str, Tuple] = {}
        self._first_node: Optional[str] = None  # cached min(node_infos)
        # SortedSet members: O(log n) add/discard, O(1) min/max, so the
        # per-class representative (the member the full candidate sort
        # would pick) costs nothing to maintain even when scheduling
        # always consumes the current representative
        self._classes: Dict[Tuple, "SortedSet"] = {}

    # -- cluster state -----------------------------------------------------

    def add_node(
        self,
        node_info: NodeInfo,
        gpus_info: Optional[GpusInfo] = None,
        manager: Optional[AMDGPUManager] = None,
    ) -> None:
        with self._sched_lock:
            name = node_info.name
            self.scheduler.add_node(name, node_info, gpus_info)
            self.core.register_node(node_info, gpus_info)
            self.node_infos[name] = node_info
            if manager is not None:
                self.managers[name] = manager
            if self._first_node is None or name < self._first_node:
                self._first_node = name
            self.reindex_node(name)

    def reindex_node(self, name: str) -> None:
        """Re-home *name* in the signature classes after a state change.

        Called automatically on add/remove/bind/release; call it
        manually only after mutating a node's core state out-of-band.
        """
        with self._sched_lock:
            self._reindex_locked(name)

    def _reindex_locked(self, name: str) -> None:
        new_sig = self.core.state_signature(name)
        if new_sig is None:
            new_sig = ("__unregistered__", name)
        old = self._node_sig.get(name)
        if old == new_sig:
            return
        if old is not None:
            members = self._classes.get(old)
            if members is not None:
                members.discard(name)
                if not members:
                    del self._classes[old]
        if new_sig not in self._classes:
            self._classes[new_sig] = self._sorted_set()
        self._classes[new_sig].add(name)
        self._node_sig[name] = new_sig

    def _drop_node_index(self, name: str) -> None:
        old = self._node_sig.pop(name, None)
        if old is not None:
            members = self._classes.get(old)
            if members is not None:
                members.discard(name)
                if not members:
                    del self._classes[old]

    def add_node_from_manager(self, name: str, manager: AMDGPUManager) -> NodeInfo:
        """Discovery -> advertise -> register, in one step."""
        ni = NodeInfo(name=name)
        manager.start()
        manager.update_node_info(ni)
        self.add_node(ni, manager._last_info, manager)
        return ni

    def remove_node(self, name: str) -> None:
        with self._sched_lock:
            self.scheduler.remove_node(name)
            self.core.remove_node(name)
            self.node_infos.pop(name, None)
            self.managers.pop(name, None)
            if name == self._first_node:
                self._first_node = min(self.node_infos) if self.node_infos else None
            self._drop_node_index(name)

    # -- scheduling --------------------------------------------------------

    def schedule(self, pod: PodInfo) -> ScheduleResult:
        """Fit -> choose node -> allocate -> bind -> commit."""
        with self._sched_lock:
            return self._schedule_locked(pod)

    def _schedule_locked(self, pod: PodInfo) -> ScheduleResult:
        t0 = time.perf_counter()
        candidates: List[Tuple[Tuple[float, int], str, PodInfo, List[str]]] = []
        # The topology-aware translation synthesizes against the
        # cluster-wide canonical-tree cache, so it is node-independent
        # (the flat knob==0 path wraps against per-node advertisement
        # instead): translate once, bind per candidate node.
        from ..scheduler.translate import GPU_TOPOLOGY_GENERATION

        shared: Optional[PodInfo] = None
        if pod.requests.get(GPU_TOPOLOGY_GENERATION) in (None, 1) and self.node_infos:
            shared = pod.copy()
            first_ni = self.node_infos[self._first_node]
            try:
                self.scheduler.pod_allocate(first_ni, shared)
            except SchedulingError:
                raise SchedulingError(f"no node fits pod {pod.name}")
        # Equivalence-class dedup: nodes with identical topology
        # fingerprint AND identical free-position sets produce identical
        # bind results and scores, so only one representative per class
        # needs a (relatively expensive) trial bind.  The classes are
        # maintained INCREMENTALLY at bind/release/add/remove events, so
        # this loop is O(distinct states), never O(nodes).
        # Representative = the class member the full sort below would
        # have picked among its (tied) members: min name for the
        # first-fit "naive" policy, max name for the reverse
        # (score, name) sort of "xgmi".
        last = self.policy != "naive"  # naive: min name; xgmi: max name
        rep_names = sorted(
            members[-1] if last else members[0]
            for members in self._classes.values()
        )
        for name in rep_names:
            ni = self.node_infos[name]
            # fit == "a translation + binding exists": the bind attempt
            # below subsumes the pod_fits_device predicate (which stays
            # available for API parity / external callers).
            try:
                if shared is not None:
                    trial = shared.copy()
                else:
                    trial = pod.copy()
                    self.scheduler.pod_allocate(ni, trial)
                uuids = self.core.bind_pod(name, trial, commit=False)
            except SchedulingError:
                continue
            state = self.core.nodes[name]
            idxs = [state.gpus[u].index for u in uuids]
            ring_bw = state.scorer.ring_bw(idxs) if idxs else 0.0
            remaining = [
                state.gpus[u].index for u in state.free_uuids() if u not in set(uuids)
            ]
            frag = state.scorer.edges(remaining)
            score = (ring_bw, frag)
            candidates.append((score, name, trial, uuids))
        if not candidates:
            raise SchedulingError(f"no node fits pod {pod.name}")
        if self.policy == "naive":
            # reference-like: no bandwidth model across nodes — first
            # fitting node in name order (the external core's arbitrary
            # choice; fit returned score 0.0, gpu_scheduler.go:43)
            candidates.sort(key=lambda c: c[1])
        else:
            candidates.sort(key=lambda c: (c[0], c[1]), reverse=True)
        best_score, node_name, bound_pod, uuids = candidates[0]
        # adopt the winning translation/bindings into the caller's pod
        pod.running_containers = bound_pod.running_containers
        pod.init_containers = bound_pod.init_containers
        pod.node_name = node_name
        self.core.take_pod_resources(node_name, pod)
        self.reindex_node(node_name)  # its free set changed
        latency = time.perf_counter() - t0
        ring = best_score[0] if best_score else 0.0
        EVENTS.record(
            "schedule",
            pod=pod.name,
            node=node_name,
            gpus=list(uuids),
            latency_ms=round(latency * 1e3, 4),
            predicted_ring_gbps=None if ring >= 1e9 else round(ring, 1),
        )
        return ScheduleResult(
            pod_name=pod.name,
            node_name=node_name,
            uuids=uuids,
            latency_s=latency,
        )

    def release(self, pod: PodInfo) -> None:
        if pod.node_name:
            with self._sched_lock:
                self.core.return_pod_resources(pod.node_name, pod)
                self._reindex_locked(pod.node_name)
            # node side: clear the device plugin's in_use flags (the
            # other half of allocate's in_use=True — no write-only state)
            mgr = self.managers.get(pod.node_name)
            if mgr is not None:
                for cont in list(pod.running_containers.values()) + list(
                    pod.init_containers.values()
                ):
                    mgr.release(pod, cont)
            EVENTS.record("release", pod=pod.name, node=pod.node_name)

    # -- container create (node side) --------------------------------------

    def container_allocate(self, pod: PodInfo, container_name: str):
        """(mounts, devices, envs) for one container of a scheduled pod."""
        mgr = self.managers.get(pod.node_name or "")
        if mgr is None:
            raise SchedulingError(f"no device manager for node {pod.node_name}")
        cont = pod.running_containers.get(container_name) or pod.init_containers.get(
            container_name
        )
        if cont is None:
            raise KeyError(container_name)
        return mgr.allocate(pod, cont)
