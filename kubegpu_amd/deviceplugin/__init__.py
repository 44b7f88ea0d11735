"""Node-side device plugin (cf. nvidiagpuplugin/gpu/nvidia/)."""

from .manager import ALLOCATE_RE, AMDGPUManager, create_device_plugin  # noqa: F401
