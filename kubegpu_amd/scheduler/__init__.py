"""Topology-aware device scheduler (cf. gpuschedulerplugin/)."""

from .translate import (  # noqa: F401
    GPU_TOPOLOGY_GENERATION,
    SchedulingError,
    TWO_LEVEL_TEMPLATE,
    convert_to_best_gpu_requests,
    pod_num_gpus,
    set_gpu_reqs,
    synth_name,
    translate_gpu_container_resources,
    translate_gpu_resources,
    translate_pod_gpu_resources,
    tree_slots,
)
from .treecache import (  # noqa: F401
    LabeledLayout,
    NodeTreeCache,
    compute_tree_score,
    parse_node_resources,
    tree_key,
)
from .scheduler import AMDGPUScheduler, create_device_scheduler_plugin  # noqa: F401
from . import xgmi  # noqa: F401
