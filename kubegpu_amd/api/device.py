"""The Device interface (KubeDevice-API `device` parity).

Reconstructed from reference call sites: device.Device{New, Start,
GetName, UpdateNodeInfo, Allocate} (nvidia_gpu_manager.go:35-38,216 and
cmd/main.go:23).  Allocate returns (mounts, devices, envs):
for the MI355X path, devices are /dev/kfd plus the per-GPU
/dev/dri/renderD* nodes and envs carry ROCR_VISIBLE_DEVICES — no vendor
runtime hook is involved (north star, BASELINE.json).
"""

from __future__ import annotations

import abc
from typing import Dict, List, Tuple

from .types import ContainerInfo, Mount, NodeInfo, PodInfo


class Device(abc.ABC):
    """Node-side device plugin interface."""

    @abc.abstractmethod
    def new(self) -> None:
        """Initialise internal state (device.Device.New)."""

    @abc.abstractmethod
    def start(self) -> None:
        """Begin discovery; must not fail the node when the GPU stack is
        absent (reference ignores discovery errors at Start,
        nvidia_gpu_manager.go:185-188)."""

    @abc.abstractmethod
    def get_name(self) -> str:
        """Plugin name."""

    @abc.abstractmethod
    def update_node_info(self, node_info: NodeInfo) -> None:
        """Publish capacity/allocatable (topology tree + flat counts)."""

    @abc.abstractmethod
    def allocate(
        self, pod: PodInfo, container: ContainerInfo
    ) -> Tuple[List[Mount], List[str], Dict[str, str]]:
        """Resolve container.allocate_from into concrete (mounts, device
        paths, env vars) at container-create time."""
