"""AMDGPUManager — node-side device plugin.

Behavioral parity with the reference's NvidiaGPUManager
(/root/reference/nvidiagpuplugin/gpu/nvidia/nvidia_gpu_manager.go):

* mutex-guarded inventory with a 5-minute discovery cache (:110-121)
* mark/sweep re-discovery that tolerates GPUs vanishing while preserving
  in-use flags (:132-155)
* two-pass topology grouping into hierarchical resource names (:63-91,
  158-180) — here derived from the explicit xGMI link graph instead of
  NVML P2P levels:
      pass 0 (gpugrp0): maximal groups closed under single-hop xGMI
      pass 1 (gpugrp1): groups closed under any xGMI path or shared NUMA
* update_node_info publishes amd.com/gpu counts plus per-GPU
  `<tree>/cards=1` and `<tree>/memory=<bytes>` (:191-213)
* allocate resolves AllocateFrom values to concrete GPUs and returns
  /dev/kfd + per-GPU /dev/dri render nodes and ROCR_VISIBLE_DEVICES
  (:216-241 — except the MI355X path needs no vendor runtime hook and no
  REST daemon: device paths are computed directly).
"""

from __future__ import annotations

import re
import threading
import time
from typing import Dict, List, Optional, Tuple

from ..api import utils
from ..api.device import Device
from ..api.types import (
    ContainerInfo,
    Mount,
    NodeInfo,
    PodInfo,
    add_group_resource,
)
from ..discovery import (
    Backend,
    DiscoveryError,
    GpuInfo,
    GpusInfo,
    default_backend,
)
from ..plugintypes import RESOURCE_GPU

# Extracts the concrete GPU id out of a bound cards resource name
# (cf. the reference's UUID regex at nvidia_gpu_manager.go:225).
ALLOCATE_RE = re.compile(r".*/gpugrp1/[^/]+/gpugrp0/[^/]+/gpu/([^/]+)/cards$")

DISCOVERY_CACHE_S = 300.0  # 5 minutes, like the reference
VANISHED_TTL_S = 600.0  # how long a vanished GPU stays visible-Unhealthy


class _UnionFind:
    def __init__(self, n: int):
        self.parent = list(range(n))

    def find(self, x: int) -> int:
        while self.parent[x] != x:
            self.parent[x] = self.parent[self.parent[x]]
            x = self.parent[x]
        return x

    def union(self, a: int, b: int) -> None:
        ra, rb = self.find(a), self.find(b)
        if ra != rb:
            self.parent[max(ra, rb)] = min(ra, rb)


class AMDGPUManager(Device):
    """MI355X device plugin manager."""

    def __init__(self, backend: Optional[Backend] = None):
        self._lock = threading.RLock()
        self._backend = backend if backend is not None else default_backend()
        self.gpus: Dict[str, GpuInfo] = {}  # uuid -> GpuInfo
        self.path_to_id: Dict[str, str] = {}  # render path -> uuid
        self.bdf_to_id: Dict[str, str] = {}
        self.index_to_id: Dict[int, str] = {}
        self._last_get_time: float = 0.0
        self._last_info: Optional[GpusInfo] = None
        self._discovering: bool = False  # single-flight refresh guard
        self._fetch_cv = threading.Condition(self._lock)
        self._fetch_gen = 0  # bumped on every completed fetch
        # Recently-vanished GPUs kept as tombstones so the kubelet
        # device plugin can report them Unhealthy (capacity visible,
        # allocatable 0) instead of silently shrinking the node, until
        # the grace window expires: uuid -> (last GpuInfo, swept-at).
        self.vanished: Dict[str, tuple] = {}

    # -- Device interface --------------------------------------------------

    def new(self) -> None:
        with self._lock:
            self.gpus = {}
            self.path_to_id = {}
            self.bdf_to_id = {}
            self.index_to_id = {}
            self._last_get_time = 0.0

    def start(self) -> None:
        # Discovery errors must not fail node start (reference ignores
        # them at Start, nvidia_gpu_manager.go:185-188).
        try:
            self.update_gpu_info()
        except DiscoveryError as e:
            utils.errorf("GPU discovery failed at start (node keeps 0 GPUs): %s", e)

    def get_name(self) -> str:
        return "amdgpu"

    # -- discovery ---------------------------------------------------------

    def update_gpu_info(self, force: bool = False) -> None:
        """Fetch inventory (with 5-min cache), mark/sweep, regroup.

        The backend fetch (a subprocess on the amdsmi path, and slow on
        a fully loaded GPU) runs OUTSIDE the manager lock, single-flight:
        concurrent callers serve the previous (stale) state instead of
        queueing behind the fetch — Allocate/GetPreferredAllocation must
        never block tens of seconds on a refresh."""
        with self._lock:
            now = time.monotonic()
            if not force and self._last_info is not None and (
                now - self._last_get_time < DISCOVERY_CACHE_S
            ):
                return
            if self._discovering:
                if not force:
                    return  # stale-while-revalidate
                # force=True must actually refresh: wait
                # for the in-flight fetch; if it succeeded, that IS the
                # refresh — otherwise fall through and fetch ourselves.
                start_gen = self._fetch_gen
                while self._discovering:
                    self._fetch_cv.wait(timeout=30.0)
                if self._fetch_gen > start_gen and self._last_info is not None:
                    return
            self._discovering = True
        try:
            info = self._backend.get_devices()  # may raise DiscoveryError
        except BaseException:
            with self._lock:
                self._discovering = False
                self._fetch_cv.notify_all()
            raise
        with self._lock:
            self._discovering = False
            self._fetch_gen += 1
            self._fetch_cv.notify_all()
            # re-read the clock: the fetch above may be slow, and the
            # tombstone sweep below must stamp vanish time at NOW, not
            # at fetch start (otherwise VANISHED_TTL shrinks)
            now = time.monotonic()
            self._last_get_time = now
            self._last_info = info

            # mark...
            for gpu in self.gpus.values():
                gpu.found = False
            # ...rebuild/refresh...
            for dev in info.devices:
                prev = self.gpus.get(dev.uuid)
                if prev is not None:
                    in_use = prev.in_use  # survives re-discovery (:143-145)
                else:
                    in_use = False
                dev.found = True
                dev.in_use = in_use
                self.gpus[dev.uuid] = dev
            # ...sweep (tombstone first, drop after the grace window).
            for uuid in [u for u, g in self.gpus.items() if not g.found]:
                utils.logf(2, "GPU %s vanished; removing from inventory", uuid)
                self.vanished[uuid] = (self.gpus[uuid], now)
                del self.gpus[uuid]
            for uuid in list(self.vanished):
                if uuid in self.gpus:  # came back
                    del self.vanished[uuid]
                elif now - self.vanished[uuid][1] > VANISHED_TTL_S:
                    del self.vanished[uuid]

            self.path_to_id = {g.render_path: u for u, g in self.gpus.items() if g.render_path}
            self.bdf_to_id = {g.bdf: u for u, g in self.gpus.items() if g.bdf}
            self.index_to_id = {g.index: u for u, g in self.gpus.items()}

            self._topology_discovery(info)

    def device_health(self) -> Dict[str, bool]:
        """uuid -> healthy for every advertisable device.

        Present GPUs are healthy unless they report uncorrectable ECC
        errors (GpuInfo.healthy); tombstoned (recently-vanished) GPUs
        stay visible but always unhealthy so kubelet degrades
        allocatable instead of the node silently shrinking."""
        with self._lock:
            out = {u: g.healthy for u, g in self.gpus.items()}
            for u in self.vanished:
                out.setdefault(u, False)
            return out

    def gpu_or_tombstone(self, uuid: str) -> Optional[GpuInfo]:
        with self._lock:
            g = self.gpus.get(uuid)
            if g is not None:
                return g
            t = self.vanished.get(uuid)
            return t[0] if t else None

    def _topology_discovery(self, info: GpusInfo) -> None:
        """Derive the two-level gpugrp names from the xGMI graph.

        Reference analog: two topologyDiscovery passes over NVML link
        levels (nvidia_gpu_manager.go:63-91,178-180).  Here the link
        graph is explicit, so the passes are two union-find closures.
        """
        devs = sorted(info.devices, key=lambda g: g.index)
        n = len(devs)
        if n == 0:
            return
        pos = {g.index: i for i, g in enumerate(devs)}

        # pass 0: single-hop xGMI
        uf0 = _UnionFind(n)
        # pass 1: any xGMI path, or same NUMA domain
        uf1 = _UnionFind(n)
        for g in devs:
            for l in g.links:
                if l.peer_index not in pos:
                    continue
                a, b = pos[g.index], pos[l.peer_index]
                # INTERNAL = same-package fabric between CPX/DPX
                # partitions of one OAM: tighter than any xGMI hop, so
                # it groups at level 0 alongside single-hop xGMI
                if l.type in ("XGMI", "INTERNAL") and l.hops <= 1:
                    uf0.union(a, b)
                if l.type in ("XGMI", "INTERNAL"):
                    uf1.union(a, b)
        for i, gi in enumerate(devs):
            for j in range(i + 1, n):
                if devs[j].numa_node == gi.numa_node:
                    uf1.union(i, j)

        def group_ids(uf: _UnionFind) -> Dict[int, int]:
            roots: Dict[int, int] = {}
            out: Dict[int, int] = {}
            for i in range(n):
                r = uf.find(i)
                if r not in roots:
                    roots[r] = len(roots)
                out[i] = roots[r]
            return out

        g0 = group_ids(uf0)
        g1 = group_ids(uf1)
        for i, dev in enumerate(devs):
            name = f"gpugrp1/{g1[i]}/gpugrp0/{g0[i]}/gpu/{dev.uuid}"
            gpu = self.gpus.get(dev.uuid)
            if gpu is not None:
                gpu.name = name
                gpu.topo_done = True

    # -- node advertisement -------------------------------------------------

    def update_node_info(self, node_info: NodeInfo) -> None:
        try:
            self.update_gpu_info()
        except DiscoveryError as e:
            utils.errorf("discovery failed in update_node_info: %s", e)
            # zero the count but keep the node alive (:193-197)
            for rl in (node_info.capacity, node_info.allocatable,
                       node_info.kube_cap, node_info.kube_alloc):
                rl[RESOURCE_GPU] = 0
            return
        with self._lock:
            # capacity counts every discovered GPU; allocatable excludes
            # ECC-unhealthy ones (a GPU with uncorrectable errors stays
            # visible but must not take new pods — mirrors the kubelet
            # plugin's per-device Unhealthy advertisement)
            count = len(self.gpus)
            healthy_count = sum(1 for g in self.gpus.values() if g.healthy)
            for rl in (node_info.capacity, node_info.kube_cap):
                rl[RESOURCE_GPU] = count
            for rl in (node_info.allocatable, node_info.kube_alloc):
                rl[RESOURCE_GPU] = healthy_count
            for uuid in utils.sorted_string_keys(self.gpus):
                gpu = self.gpus[uuid]
                if not gpu.name:
                    continue
                add_group_resource(node_info.capacity, f"{gpu.name}/cards", 1)
                add_group_resource(
                    node_info.capacity, f"{gpu.name}/memory", gpu.memory.vram_total_bytes
                )
                if not gpu.healthy:
                    continue
                add_group_resource(node_info.allocatable, f"{gpu.name}/cards", 1)
                add_group_resource(
                    node_info.allocatable, f"{gpu.name}/memory", gpu.memory.vram_total_bytes
                )

    # -- allocation ---------------------------------------------------------

    def allocate(
        self, pod: PodInfo, container: ContainerInfo
    ) -> Tuple[List[Mount], List[str], Dict[str, str]]:
        """Map AllocateFrom bindings to device nodes + env.

        Returns (mounts, devices, envs): /dev/kfd plus each allocated
        GPU's /dev/dri/renderD* (and card node when known), with
        ROCR_VISIBLE_DEVICES naming the allocated GPU UUIDs.
        """
        with self._lock:
            uuids: List[str] = []
            for req_name in utils.sorted_string_keys(container.allocate_from):
                concrete = container.allocate_from[req_name]
                m = ALLOCATE_RE.match(concrete)
                if not m:
                    continue
                uuid = m.group(1)
                if uuid not in uuids:
                    uuids.append(uuid)
            devices: List[str] = []
            visible: List[str] = []
            if uuids:
                devices.append("/dev/kfd")
            for uuid in uuids:
                gpu = self.gpus.get(uuid)
                if gpu is None:
                    utils.errorf("allocate: unknown GPU %s for pod %s", uuid, pod.name)
                    raise KeyError(f"unknown GPU uuid {uuid}")
                gpu.in_use = True
                if gpu.render_path:
                    devices.append(gpu.render_path)
                if gpu.card_path:
                    devices.append(gpu.card_path)
                visible.append(uuid)
            envs: Dict[str, str] = {}
            if visible:
                envs["ROCR_VISIBLE_DEVICES"] = ",".join(visible)
            return [], devices, envs

    def release(self, pod: PodInfo, container: ContainerInfo) -> List[str]:
        """Inverse of allocate: clear in_use for the container's bound
        GPUs (pod ended / binding returned).  While held, in_use still
        survives re-discovery (reference invariant,
        nvidia_gpu_manager.go:143-145); this is the missing other half —
        round-1 set the flag and never cleared it.  Returns the released
        uuids."""
        with self._lock:
            released: List[str] = []
            for req_name in utils.sorted_string_keys(container.allocate_from):
                m = ALLOCATE_RE.match(container.allocate_from[req_name])
                if not m:
                    continue
                uuid = m.group(1)
                gpu = self.gpus.get(uuid)
                if gpu is not None and gpu.in_use:
                    gpu.in_use = False
                    released.append(uuid)
            return released

    def release_uuids(self, uuids: List[str]) -> None:
        """Clear in_use for an explicit uuid list (operator tooling)."""
        with self._lock:
            for uuid in uuids:
                gpu = self.gpus.get(uuid)
                if gpu is not None:
                    gpu.in_use = False

    def in_use_uuids(self) -> List[str]:
        with self._lock:
            return sorted(u for u, g in self.gpus.items() if g.in_use)


def create_device_plugin(backend: Optional[Backend] = None) -> AMDGPUManager:
    """Factory (parity: CreateDevicePlugin, plugin/nvidiagpu.go:8-10)."""
    mgr = AMDGPUManager(backend=backend)
    mgr.new()
    return mgr
