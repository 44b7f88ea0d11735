"""The DeviceScheduler interface (KubeDevice-API `devicescheduler` parity).

Reconstructed from the reference implementation's method set
(gpuschedulerplugin/gpu_scheduler.go:21-71).
"""

from __future__ import annotations

import abc
from dataclasses import dataclass
from typing import List, Tuple

from .types import NodeInfo, PodInfo


@dataclass(frozen=True)
class PredicateFailureReason:
    """Why a pod does not fit a node (devicescheduler.PredicateFailureReason)."""

    resource_name: str
    requested: int
    used: int
    capacity: int

    def get_reason(self) -> str:
        return (
            f"Insufficient {self.resource_name}: requested {self.requested}, "
            f"used {self.used}, capacity {self.capacity}"
        )


class DeviceScheduler(abc.ABC):
    """Scheduler-side device scheduler interface."""

    @abc.abstractmethod
    def add_node(self, node_name: str, node_info: NodeInfo) -> None: ...

    @abc.abstractmethod
    def remove_node(self, node_name: str) -> None: ...

    @abc.abstractmethod
    def pod_fits_device(
        self, node_info: NodeInfo, pod_info: PodInfo, fill_allocate_from: bool, run_group_scheduler: bool
    ) -> Tuple[bool, List[PredicateFailureReason], float]:
        """(fits, failure reasons, score)."""

    @abc.abstractmethod
    def pod_allocate(self, node_info: NodeInfo, pod_info: PodInfo) -> None:
        """Finalize translation at bind time; raise on failure."""

    @abc.abstractmethod
    def take_pod_resources(self, node_info: NodeInfo, pod_info: PodInfo) -> None: ...

    @abc.abstractmethod
    def return_pod_resources(self, node_info: NodeInfo, pod_info: PodInfo) -> None: ...

    @abc.abstractmethod
    def get_name(self) -> str: ...

    @abc.abstractmethod
    def using_group_scheduler(self) -> bool: ...
