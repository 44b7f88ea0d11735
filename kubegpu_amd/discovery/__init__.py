"""GPU discovery: amdsmi / sysfs / fake backends and the inventory schema."""

from .types import (  # noqa: F401
    LINK_PCIE,
    LINK_XGMI,
    PCIE_GBPS_DEFAULT,
    XGMI_LINK_GBPS_DEFAULT,
    GpuInfo,
    GpusInfo,
    LinkInfo,
    MemoryInfo,
    VersionInfo,
    direct_xgmi_pairs,
)
from .backends import (  # noqa: F401
    AmdSmiBackend,
    Backend,
    CrashingBackend,
    DiscoveryError,
    FakeBackend,
    SysfsBackend,
    default_backend,
)
from . import fixtures  # noqa: F401
