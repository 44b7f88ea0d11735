"""Request-translation tests (cf. gpuschedulerplugin/gpu.go:16-127)."""

import pytest

from kubegpu_amd.api.types import ContainerInfo, NodeInfo, PodInfo
from kubegpu_amd.plugintypes import RESOURCE_GPU
from kubegpu_amd.scheduler import (
    GPU_TOPOLOGY_GENERATION,
    SchedulingError,
    TWO_LEVEL_TEMPLATE,
    NodeTreeCache,
    set_gpu_reqs,
    translate_gpu_container_resources,
    translate_gpu_resources,
    translate_pod_gpu_resources,
)


def test_set_gpu_reqs_takes_max():
    c = ContainerInfo(
        requests={RESOURCE_GPU: 2},
        kube_requests={RESOURCE_GPU: 3},
    )
    assert set_gpu_reqs(c) == 3
    assert c.dev_requests[RESOURCE_GPU] == 3


def test_flat_expansion_two_level():
    """Flat amd.com/gpu: 2 expands to wildcard-wrapped per-card names."""
    c = ContainerInfo(kube_requests={RESOURCE_GPU: 2})
    translate_gpu_container_resources(TWO_LEVEL_TEMPLATE, c)
    assert RESOURCE_GPU not in c.dev_requests
    assert c.dev_requests == {
        "resource/group/gpugrp1/*/gpugrp0/*/gpu/0/cards": 1,
        "resource/group/gpugrp1/*/gpugrp0/*/gpu/1/cards": 1,
    }


def test_flat_expansion_flat_node():
    """Against a flat (non-grouped) template, names stay ungrouped."""
    c = ContainerInfo(kube_requests={RESOURCE_GPU: 1})
    translate_gpu_container_resources({}, c)
    assert c.dev_requests == {"resource/group/gpu/0/cards": 1}


def test_node_allocatable_forcing():
    """A flat node's allocatable is forced to the 2-level scheme
    (cf. gpu_scheduler.go:21-28)."""
    alloc = {RESOURCE_GPU: 2}
    translate_gpu_resources(2, TWO_LEVEL_TEMPLATE, alloc)
    assert set(alloc) == {
        "resource/group/gpugrp1/*/gpugrp0/*/gpu/0/cards",
        "resource/group/gpugrp1/*/gpugrp0/*/gpu/1/cards",
    }


def _cached_node(cache: NodeTreeCache):
    res = {}
    for g in range(8):
        res[f"resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU{g}/cards"] = 1
    cache.add_node_resources("node0", res)


def test_topology_knob_dispatch():
    cache = NodeTreeCache()
    _cached_node(cache)
    ni = NodeInfo(name="node0")

    # knob unset -> topology synthesis
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    translate_pod_gpu_resources(ni, pod, cache)
    names = sorted(pod.running_containers["c"].dev_requests)
    assert names == [
        "resource/group/gpugrp1/0/gpugrp0/0/gpu/0/cards",
        "resource/group/gpugrp1/0/gpugrp0/0/gpu/1/cards",
    ]

    # knob = 0 -> flat path
    pod0 = PodInfo(
        name="p0",
        requests={GPU_TOPOLOGY_GENERATION: 0},
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 1})},
    )
    translate_pod_gpu_resources(ni, pod0, cache)
    assert list(pod0.running_containers["c"].dev_requests) == [
        "resource/group/gpu/0/cards"
    ]

    # invalid knob -> error (gpu.go:102-126)
    podx = PodInfo(
        name="px",
        requests={GPU_TOPOLOGY_GENERATION: 7},
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 1})},
    )
    with pytest.raises(SchedulingError):
        translate_pod_gpu_resources(ni, podx, cache)


def test_no_tree_big_pod_falls_back_to_flat():
    """When no cached tree holds the demand, translation falls back to
    the flat/no-topology path instead of erroring — reference semantics
    (gpu.go:104-116: `if !found || req == 0` both land on the flat
    loop).  Whether the pod then fits is the group core's decision."""
    cache = NodeTreeCache()
    _cached_node(cache)  # 8-card tree, demand is 9
    ni = NodeInfo(name="node0")
    pod = PodInfo(
        name="big",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 9})},
    )
    translate_pod_gpu_resources(ni, pod, cache)
    reqs = pod.running_containers["c"].dev_requests
    assert len(reqs) == 9
    assert all("/gpu/" in r and r.endswith("/cards") for r in reqs)


def test_big_pod_still_unschedulable_end_to_end():
    """The flat fallback does not make an oversized pod schedulable: the
    binder has only 8 concrete GPUs."""
    from kubegpu_amd.core import Cluster
    from kubegpu_amd.deviceplugin import create_device_plugin
    from kubegpu_amd.discovery import FakeBackend, fixtures

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(
        name="big",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 9})},
    )
    with pytest.raises(SchedulingError):
        cluster.schedule(pod)


def test_multi_container_disjoint_slots():
    cache = NodeTreeCache()
    _cached_node(cache)
    ni = NodeInfo(name="node0")
    pod = PodInfo(
        name="p",
        running_containers={
            "a": ContainerInfo(kube_requests={RESOURCE_GPU: 2}),
            "b": ContainerInfo(kube_requests={RESOURCE_GPU: 3}),
        },
        init_containers={"i": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    translate_pod_gpu_resources(ni, pod, cache)
    a = set(pod.running_containers["a"].dev_requests)
    b = set(pod.running_containers["b"].dev_requests)
    assert not a & b
    assert len(a) == 2 and len(b) == 3
    # init containers restart from slot 0 (max-over-init accounting)
    i = sorted(pod.init_containers["i"].dev_requests)
    assert i[0].endswith("/gpu/0/cards")


def test_synth_parse_roundtrip():
    """The synthesized request grammar and the parser are inverses."""
    from kubegpu_amd.api.resource import parse_cards_name
    from kubegpu_amd.scheduler.translate import synth_name

    for hi in range(3):
        for gi in range(4):
            for card in (0, 7, 123):
                name = synth_name(hi, gi, card)
                prefix, h, g, cid = parse_cards_name(name)
                assert (prefix, h, g, cid) == (
                    "resource/group", str(hi), str(gi), str(card)
                )


def test_pod_demand_is_max_of_running_and_init():
    """Pod GPU demand = max(Σ running, max init) — parity with
    /root/reference/gpuschedulerplugin/gpu.go:295-303: the reference sums
    the running containers, then raises the total to any larger single
    init request.  It never ADDS init on top (round-1 regression)."""
    from kubegpu_amd.scheduler.translate import pod_num_gpus

    def pod(run, init):
        return PodInfo(
            name="p",
            running_containers={
                f"r{i}": ContainerInfo(kube_requests={RESOURCE_GPU: n})
                for i, n in enumerate(run)
            },
            init_containers={
                f"i{i}": ContainerInfo(kube_requests={RESOURCE_GPU: n})
                for i, n in enumerate(init)
            },
        )

    assert pod_num_gpus(pod([4, 4], [8])) == 8      # not 16
    assert pod_num_gpus(pod([8], [8])) == 8         # not 16
    assert pod_num_gpus(pod([2], [6])) == 6         # init dominates
    assert pod_num_gpus(pod([2], [1, 6, 3])) == 6   # max over init
    assert pod_num_gpus(pod([3], [])) == 3
    assert pod_num_gpus(pod([], [5])) == 5


def test_translate_8run_8init_fits_8gpu_tree():
    """An 8-GPU pod with an 8-GPU init container translates against an
    8-card tree (it would have demanded 16 and failed under the round-1
    sum semantics; fits in the reference and stock Kubernetes)."""
    cache = NodeTreeCache()
    _cached_node(cache)  # 8-card canonical tree
    ni = NodeInfo(name="node0")
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 8})},
        init_containers={"i": ContainerInfo(kube_requests={RESOURCE_GPU: 8})},
    )
    translate_pod_gpu_resources(ni, pod, cache)
    assert len(pod.running_containers["c"].dev_requests) == 8
    assert len(pod.init_containers["i"].dev_requests) == 8
