"""Kubelet-facing device-plugin server (stock-Kubernetes path)."""

from . import dpapi  # noqa: F401
from .kubelet_plugin import DevicePluginServicer, KubeletDevicePlugin  # noqa: F401
