"""GPU request translation.

Parity with gpuschedulerplugin/gpu.go:16-127,247-324:
flat `amd.com/gpu: N` pod requests are rewritten into topology-encoded
per-card requests, either

* **topology-aware** (default / `gpu/gpu-generate-topology` in {unset,1}):
  pick the best cached canonical tree and synthesize concrete
  `resource/group/gpugrp1/<hi>/gpugrp0/<gi>/gpu/<k>/cards` requests,
  packing the densest groups first (DFS over descending-sorted children,
  gpu.go:247-271), or
* **flat** (`gpu-generate-topology` == 0): per-card requests wrapped with
  wildcard group levels that match whatever the node advertises
  (gpu.go:16-66 via resource.TranslateResource), or
* error for any other knob value (gpu.go:102-126).
"""

from __future__ import annotations

import re
from typing import List, Tuple

from ..api import utils
from ..api.resource import translate_resource
from ..api.types import (
    DEVICE_GROUP_PREFIX,
    ContainerInfo,
    NodeInfo,
    PodInfo,
    ResourceList,
)
from ..plugintypes import RESOURCE_GPU, SortedTreeNode
from .treecache import NodeTreeCache

# Per-pod config knob (gpu_scheduler.go:14): unset/1 = auto topology,
# 0 = flat/no-topology, anything else = error.
GPU_TOPOLOGY_GENERATION = "gpu/gpu-generate-topology"

# Synthetic 2-level template used to force node allocatable lists to the
# two-level scheme (gpu_scheduler.go:21-28).
TWO_LEVEL_TEMPLATE: ResourceList = {
    f"{DEVICE_GROUP_PREFIX}/gpugrp1/A/gpugrp0/B/gpu/GPU0/cards": 1
}

_GPU_PATH_RE = re.compile(r"(^|/)gpu/[^/]+/cards$")


class SchedulingError(RuntimeError):
    pass


def set_gpu_reqs(cont: ContainerInfo) -> int:
    """DevRequests[gpu] = max(device-reported, kube-reported)
    (gpu.go:80-92)."""
    dev = cont.dev_requests.get(RESOURCE_GPU, cont.requests.get(RESOURCE_GPU, 0))
    kube = cont.kube_requests.get(RESOURCE_GPU, 0)
    n = max(dev, kube)
    if n > 0:
        cont.dev_requests[RESOURCE_GPU] = n
    return n


def strip_gpu_dev_requests(cont: ContainerInfo) -> None:
    """Remove all per-card `*/gpu/*` entries before re-synthesis
    (gpu.go:275-283)."""
    for name in list(cont.dev_requests.keys()):
        if name == RESOURCE_GPU or _GPU_PATH_RE.search(name):
            del cont.dev_requests[name]


def translate_gpu_resources(
    num_gpus: int, template_resources: ResourceList, resources: ResourceList
) -> None:
    """3-stage rewrite of *resources* in place (gpu.go:16-66).

    Stage 1 expands a flat `amd.com/gpu: N` into per-card
    `resource/group/gpu/<i>/cards` entries; stages 2-3 wrap with the
    gpugrp0 / gpugrp1 levels advertised by *template_resources*.
    """
    flat = resources.pop(RESOURCE_GPU, 0)
    n = max(int(flat), int(num_gpus))
    existing_cards = any(_GPU_PATH_RE.search(name) for name in resources)
    if n > 0 and not existing_cards:
        for i in range(n):
            resources[f"{DEVICE_GROUP_PREFIX}/gpu/{i}/cards"] = 1
    translate_resource(template_resources, resources, "gpugrp0", "gpu")
    translate_resource(template_resources, resources, "gpugrp1", "gpugrp0")


def translate_gpu_container_resources(
    template_resources: ResourceList, cont: ContainerInfo
) -> None:
    """Per-container flat translation (gpu.go:75-78)."""
    n = set_gpu_reqs(cont)
    translate_gpu_resources(n, template_resources, cont.dev_requests)


def tree_slots(tree: SortedTreeNode) -> List[Tuple[int, int]]:
    """Flatten the canonical tree into an ordered list of (hi, gi)
    position slots, one per card, densest groups first.

    The DFS order over descending-sorted children IS the packing policy
    (gpu.go:247-271): consuming slots left to right fills the densest /
    highest-scoring gpugrp0 first.
    """
    slots: List[Tuple[int, int]] = []
    for hi, h_node in enumerate(tree.children):
        if h_node.children:
            for gi, g_node in enumerate(h_node.children):
                slots.extend([(hi, gi)] * g_node.val)
        else:
            slots.extend([(hi, 0)] * h_node.val)
    if not tree.children:
        slots.extend([(0, 0)] * tree.val)
    return slots


def synth_name(hi: int, gi: int, card: int) -> str:
    """The synthesized request grammar (gpu.go:286):
    resource/group/gpugrp1/<hi>/gpugrp0/<gi>/gpu/<card>/cards."""
    return f"{DEVICE_GROUP_PREFIX}/gpugrp1/{hi}/gpugrp0/{gi}/gpu/{card}/cards"


def translate_to_tree(
    tree: SortedTreeNode, cont: ContainerInfo, slots: List[Tuple[int, int]], offset: int
) -> int:
    """Rewrite one container's dev requests against *tree* starting at
    slot *offset*; returns the new offset (gpu.go:273-291)."""
    n = set_gpu_reqs(cont)
    strip_gpu_dev_requests(cont)
    if n == 0:
        return offset
    if offset + n > len(slots):
        raise SchedulingError(
            f"tree with {len(slots)} cards cannot hold {offset + n} requested"
        )
    for k in range(n):
        hi, gi = slots[offset + k]
        cont.dev_requests[synth_name(hi, gi, offset + k)] = 1
    return offset + n


def pod_num_gpus(pod: PodInfo) -> int:
    """Pod GPU demand = max(Σ running, max init) (gpu.go:295-303).

    Init containers run sequentially BEFORE the app containers start, so
    the effective demand is the larger of the app containers' total and
    the biggest single init container — the same effective-request math
    stock Kubernetes uses.  (The reference sums running, then raises to
    any larger init request; it never adds the two.)
    """
    running = sum(set_gpu_reqs(c) for c in pod.running_containers.values())
    init = max((set_gpu_reqs(c) for c in pod.init_containers.values()), default=0)
    return max(running, init)


def convert_to_best_gpu_requests(pod: PodInfo, cache: NodeTreeCache) -> None:
    """Rewrite every container against the best cached tree
    (gpu.go:294-324)."""
    num = pod_num_gpus(pod)
    if num == 0:
        return
    tree = cache.find_best_tree(num)
    if tree is None:
        raise SchedulingError(
            f"no cached node topology can hold {num} GPUs for pod {pod.name}"
        )
    slots = tree_slots(tree)
    offset = 0
    for name in utils.sorted_string_keys(pod.running_containers):
        offset = translate_to_tree(tree, pod.running_containers[name], slots, offset)
    # Init containers run one at a time; each starts from the pod's first
    # slots (they may overlap the running containers' slots and each
    # other — demand is max(Σ running, max init), gpu.go:295-303).
    for name in utils.sorted_string_keys(pod.init_containers):
        translate_to_tree(tree, pod.init_containers[name], slots, 0)


def translate_pod_gpu_resources(
    node_info: NodeInfo, pod: PodInfo, cache: NodeTreeCache
) -> None:
    """Dispatch on the gpu-generate-topology knob (gpu.go:94-127).

    Exact reference semantics (gpu.go:104-123): knob unset/1 tries the
    best cached tree, and when NO tree holds the demand it falls back
    to the flat/no-topology translation instead of failing — whether
    the pod then fits is the group core's per-node decision.  knob 0
    goes flat directly; any other value is an error.
    """
    knob = pod.requests.get(GPU_TOPOLOGY_GENERATION)
    if knob is None or knob == 1:
        try:
            convert_to_best_gpu_requests(pod, cache)
            return
        except SchedulingError:
            pass  # !found -> flat fallback (gpu.go:113-116)
    elif knob != 0:
        raise SchedulingError(
            f"invalid {GPU_TOPOLOGY_GENERATION} value {knob} for pod {pod.name}"
        )
    for cont in list(pod.running_containers.values()) + list(
        pod.init_containers.values()
    ):
        translate_gpu_container_resources(node_info.allocatable, cont)
