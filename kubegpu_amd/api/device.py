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


def create_device_from_plugin(path: str) -> Device:
    """Load a device plugin from a Python file path and instantiate it.

    Parity with device.CreateDeviceFromPlugin loading a Go buildmode
    plugin .so (reference cmd/main.go:23): the module must expose
    ``create_device_plugin() -> Device``.
    """
    import importlib.util
    import os

    if not os.path.exists(path):
        raise ImportError(f"device plugin path does not exist: {path}")
    spec = importlib.util.spec_from_file_location("kubegpu_amd_plugin", path)
    if spec is None or spec.loader is None:
        raise ImportError(f"cannot load device plugin from {path}")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    factory = getattr(mod, "create_device_plugin", None)
    if factory is None:
        raise AttributeError(f"{path} does not export create_device_plugin()")
    dev = factory()
    if not isinstance(dev, Device):
        raise TypeError(f"{path}: create_device_plugin() returned {type(dev)}")
    return dev
