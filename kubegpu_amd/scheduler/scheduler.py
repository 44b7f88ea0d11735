"""AMDGPUScheduler — the DeviceScheduler implementation.

Parity with gpuschedulerplugin/gpu_scheduler.go:21-71:

* add_node forces the node's allocatable list to the two-level scheme
  using a synthetic template, then caches the canonical tree (:21-28)
* pod_fits_device / pod_allocate both run the pod translation — "fit"
  means "a topology-aware translation exists"; concrete per-device
  binding belongs to the group-scheduler core (using_group_scheduler()
  is True, :69-71), which this repo also owns (kubegpu_amd.core)
* take/return_pod_resources are no-ops here — accounting lives in the
  core (:57-63)

MI355X extension: add_node optionally registers the node's discovered
xGMI link graph (GpusInfo) so the core can score concrete subsets by ring
bandwidth instead of group names alone.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..api import utils
from ..api.devicescheduler import DeviceScheduler, PredicateFailureReason
from ..api.resource import WILDCARD
from ..api.types import NodeInfo, PodInfo
from ..discovery import GpusInfo
from ..plugintypes import RESOURCE_GPU
from .translate import (
    SchedulingError,
    TWO_LEVEL_TEMPLATE,
    translate_gpu_resources,
    translate_pod_gpu_resources,
)
from .treecache import NodeTreeCache


def _concretize_wildcards(resources: Dict[str, int]) -> None:
    """Node-side lists must be concrete: a flat node's synthesized group
    levels become group 0 (single implicit group)."""
    for name in list(resources.keys()):
        if f"/{WILDCARD}/" in name:
            resources[name.replace(f"/{WILDCARD}/", "/0/")] = resources.pop(name)


class AMDGPUScheduler(DeviceScheduler):
    def __init__(self, cache: Optional[NodeTreeCache] = None, group_core=None):
        self.cache = cache if cache is not None else NodeTreeCache()
        self.topologies: Dict[str, GpusInfo] = {}
        # optional kubegpu_amd.core.GroupScheduler: lets pod_fits_device
        # run the concrete binder when run_group_scheduler=True (the
        # reference leaves this to the external core; we own it)
        self.group_core = group_core

    # -- node lifecycle ----------------------------------------------------

    def add_node(
        self,
        node_name: str,
        node_info: NodeInfo,
        gpus_info: Optional[GpusInfo] = None,
    ) -> None:
        num = node_info.kube_alloc.get(RESOURCE_GPU, node_info.allocatable.get(RESOURCE_GPU, 0))
        translate_gpu_resources(num, TWO_LEVEL_TEMPLATE, node_info.allocatable)
        _concretize_wildcards(node_info.allocatable)
        self.cache.add_node_resources(node_name, node_info.allocatable)
        if gpus_info is not None:
            self.topologies[node_name] = gpus_info

    def remove_node(self, node_name: str) -> None:
        self.cache.remove_node(node_name)
        self.topologies.pop(node_name, None)

    # -- pod scheduling ----------------------------------------------------

    def pod_fits_device(
        self,
        node_info: NodeInfo,
        pod_info: PodInfo,
        fill_allocate_from: bool = False,
        run_group_scheduler: bool = False,
    ) -> Tuple[bool, List[PredicateFailureReason], float]:
        target = pod_info if fill_allocate_from else pod_info.copy()
        try:
            translate_pod_gpu_resources(node_info, target, self.cache)
            if run_group_scheduler and self.group_core is not None:
                # dry-run concrete binding on this node (commit=False)
                self.group_core.bind_pod(node_info.name, target.copy(), commit=False)
        except SchedulingError as e:
            utils.logf(3, "pod %s does not fit: %s", pod_info.name, e)
            reason = PredicateFailureReason(
                resource_name=RESOURCE_GPU,
                requested=pod_info.requests.get(RESOURCE_GPU, 0),
                used=0,
                capacity=node_info.allocatable.get(RESOURCE_GPU, 0),
            )
            return False, [reason], 0.0
        return True, [], 0.0

    def pod_allocate(self, node_info: NodeInfo, pod_info: PodInfo) -> None:
        try:
            translate_pod_gpu_resources(node_info, pod_info, self.cache)
        except SchedulingError as e:
            raise SchedulingError(
                f"pod_allocate: no translation for pod {pod_info.name}: {e}"
            ) from e

    def take_pod_resources(self, node_info: NodeInfo, pod_info: PodInfo) -> None:
        pass  # accounting lives in the group-scheduler core

    def return_pod_resources(self, node_info: NodeInfo, pod_info: PodInfo) -> None:
        pass

    def get_name(self) -> str:
        return "amdgpu-topology"

    def using_group_scheduler(self) -> bool:
        return True


def create_device_scheduler_plugin() -> AMDGPUScheduler:
    """Factory (parity: CreateDeviceSchedulerPlugin,
    gpuschedulerplugin/plugin/gpuscheduler.go:8-11)."""
    return AMDGPUScheduler()
