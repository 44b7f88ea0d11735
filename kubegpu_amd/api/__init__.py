"""Re-owned KubeDevice-API surface."""

from .types import (  # noqa: F401
    DEVICE_GROUP_PREFIX,
    ContainerInfo,
    Mount,
    NodeInfo,
    PodInfo,
    ResourceList,
    ResourceLocation,
    ResourceName,
    add_group_resource,
    new_node_info,
)
from .device import Device  # noqa: F401
from .devicescheduler import DeviceScheduler, PredicateFailureReason  # noqa: F401
from .resource import (  # noqa: F401
    WILDCARD,
    matches,
    parse_cards_name,
    translate_resource,
)
from . import utils  # noqa: F401
