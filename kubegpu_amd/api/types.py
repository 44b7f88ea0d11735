"""Re-owned KubeDevice-API core types.

The reference (microsoft/KubeGPU) consumes these from the external
`github.com/Microsoft/KubeDevice-API` repo; the surface is reconstructed
from its call sites (see SURVEY.md §1 "The external KubeDevice-API
surface", citing e.g. /root/reference/gpuschedulerplugin/gpu.go:16,94 and
nvidiagpuplugin/gpu/nvidia/nvidia_gpu_manager.go:200-213).  We own the API
here, expressed as plain Python types.

Resource-name grammar (the contract every layer shares):

    <DeviceGroupPrefix>/gpugrp1/<H>/gpugrp0/<G>/gpu/<ID>/cards  = 1
    <DeviceGroupPrefix>/gpugrp1/<H>/gpugrp0/<G>/gpu/<ID>/memory = <bytes>

with ``DeviceGroupPrefix == "resource/group"`` (reference fixture:
gpuschedulerplugin/gpu_test.go:79-84).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional

# A resource name, e.g. "amd.com/gpu" or
# "resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU-abc/cards".
ResourceName = str

# Quantity per resource name (counts, bytes).
ResourceList = Dict[ResourceName, int]

# Request-name -> concrete-device-resource-name binding produced by the
# group scheduler core and consumed by Device.allocate
# (cf. nvidia_gpu_manager.go:216-241).
ResourceLocation = Dict[ResourceName, ResourceName]

# Prefix under which grouped (topology-encoded) resources are published.
DEVICE_GROUP_PREFIX: str = "resource/group"


def add_group_resource(rlist: ResourceList, suffix: str, val: int) -> None:
    """Insert ``DEVICE_GROUP_PREFIX/<suffix> = val`` into *rlist*.

    API parity: types.AddGroupResource (used at
    nvidia_gpu_manager.go:206-209 to publish per-GPU cards/memory).
    """
    rlist[f"{DEVICE_GROUP_PREFIX}/{suffix}"] = val


@dataclass
class NodeInfo:
    """Per-node resource advertisement (types.NodeInfo).

    Capacity/Allocatable carry the full topology-encoded tree;
    KubeCap/KubeAlloc carry what the stock kubelet sees (flat counts).
    Reference population: nvidia_gpu_manager.go:191-213.
    """

    name: str = ""
    capacity: ResourceList = field(default_factory=dict)
    allocatable: ResourceList = field(default_factory=dict)
    kube_cap: ResourceList = field(default_factory=dict)
    kube_alloc: ResourceList = field(default_factory=dict)

    def copy(self) -> "NodeInfo":
        return NodeInfo(
            name=self.name,
            capacity=dict(self.capacity),
            allocatable=dict(self.allocatable),
            kube_cap=dict(self.kube_cap),
            kube_alloc=dict(self.kube_alloc),
        )


def new_node_info(name: str = "") -> NodeInfo:
    """types.NewNodeInfo() (cf. cmd/main.go:37)."""
    return NodeInfo(name=name)


@dataclass
class ContainerInfo:
    """types.ContainerInfo (cf. gpu.go:75-92, nvidia_gpu_manager.go:221).

    requests       device-plugin-visible requests (grouped names)
    kube_requests  what the pod spec literally asked for (flat names)
    dev_requests   the scheduler's rewritten, topology-aware requests
    allocate_from  request-name -> concrete-device-name bindings
    """

    requests: ResourceList = field(default_factory=dict)
    kube_requests: ResourceList = field(default_factory=dict)
    dev_requests: ResourceList = field(default_factory=dict)
    allocate_from: ResourceLocation = field(default_factory=dict)

    def copy(self) -> "ContainerInfo":
        return ContainerInfo(
            requests=dict(self.requests),
            kube_requests=dict(self.kube_requests),
            dev_requests=dict(self.dev_requests),
            allocate_from=dict(self.allocate_from),
        )


@dataclass
class PodInfo:
    """types.PodInfo (cf. gpu.go:94-127, gpu_test.go:61-71)."""

    name: str = ""
    requests: ResourceList = field(default_factory=dict)
    init_containers: Dict[str, ContainerInfo] = field(default_factory=dict)
    running_containers: Dict[str, ContainerInfo] = field(default_factory=dict)
    node_name: Optional[str] = None

    def copy(self) -> "PodInfo":
        return PodInfo(
            name=self.name,
            requests=dict(self.requests),
            init_containers={k: v.copy() for k, v in self.init_containers.items()},
            running_containers={k: v.copy() for k, v in self.running_containers.items()},
            node_name=self.node_name,
        )


@dataclass
class Mount:
    """A host-path mount to inject into the container (device.Mount)."""

    host_path: str
    container_path: str
    read_only: bool = True

