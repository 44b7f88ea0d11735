"""Group-scheduler core + in-process cluster orchestration."""

from .group_scheduler import GroupScheduler, NodeState  # noqa: F401
from .cluster import Cluster, ScheduleResult  # noqa: F401
