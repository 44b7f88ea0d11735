"""Real captured MI355X inventory as a CPU test fixture.

The reference's device-plugin tests are built on real captured JSON
payloads (two nvidia-docker dumps, nvidia_gpu_manager_test.go:16-17).
Same strategy here: tests/fixtures/real_1x_mi355x.json is the verbatim
`amdsmiinfo json` output of a production MI355X box (captured via gpurun,
round 1), driven through the whole discovery -> manager -> scheduler
stack without a GPU.
"""

import json
import os

from kubegpu_amd.api.types import ContainerInfo, NodeInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, GpusInfo
from kubegpu_amd.plugintypes import RESOURCE_GPU

FIXTURE = os.path.join(os.path.dirname(__file__), "fixtures", "real_1x_mi355x.json")


def _load():
    with open(FIXTURE) as f:
        return GpusInfo.from_json(f.read())


def test_real_payload_parses():
    info = _load()
    assert len(info.devices) == 1
    g = info.devices[0]
    assert g.model == "AMD Instinct MI355 OAM"
    assert g.gfx_target == "gfx950" and g.device_id == "0x75a3"
    assert g.compute_units == 256  # 256 CUs in 8 XCDs
    assert g.memory.vram_total_bytes == 309220868096  # 288 GiB HBM3E
    assert g.memory.vram_bandwidth_gbps == 8192.0  # 8 TB/s
    assert g.compute_partition == "SPX" and g.memory_partition == "NPS1"
    assert g.render_path == "/dev/dri/renderD184"
    assert g.healthy


def test_real_payload_through_full_stack():
    """Capture -> manager -> advertise -> schedule -> allocate, no GPU."""
    mgr = create_device_plugin(FakeBackend(_load()))
    mgr.start()
    ni = NodeInfo(name="captured")
    mgr.update_node_info(ni)
    assert ni.kube_alloc[RESOURCE_GPU] == 1
    cards = [k for k in ni.allocatable if k.endswith("/cards")]
    assert len(cards) == 1 and "93ff75a3-0000-1000-8091-62812952020a" in cards[0]
    mem = [k for k in ni.allocatable if k.endswith("/memory")]
    assert ni.allocatable[mem[0]] == 309220868096

    cluster = Cluster()
    cluster.add_node(ni, mgr._last_info, mgr)
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 1})},
    )
    res = cluster.schedule(pod)
    mounts, devices, envs = cluster.container_allocate(pod, "c")
    assert devices == ["/dev/kfd", "/dev/dri/renderD184", "/dev/dri/card56"]
    assert envs["ROCR_VISIBLE_DEVICES"] == "93ff75a3-0000-1000-8091-62812952020a"


def test_round2_capture_parses_with_runtime_versions():
    """Round-2 verbatim capture (`amdsmiinfo json`, gpurun session F):
    the version block is runtime-queried — rocm from
    /opt/rocm/.info/version, amdsmi via amdsmi_get_lib_version — not the
    round-1 hardcode."""
    path = os.path.join(os.path.dirname(__file__), "fixtures",
                        "real_1x_mi355x_r2.json")
    with open(path) as f:
        raw = json.load(f)
    assert raw["version"]["rocm"] == "7.2.0"
    assert raw["version"]["amdsmi"] == "26.2.1"  # a real lib version
    info = GpusInfo.from_json(json.dumps(raw))
    g = info.devices[0]
    assert g.gfx_target == "gfx950"
    assert g.compute_units == 256
    assert g.memory.vram_total_bytes > 300e9
    # drives the manager end to end like the round-1 capture
    mgr = create_device_plugin(FakeBackend(info))
    mgr.start()
    ni = NodeInfo(name="captured-r2")
    mgr.update_node_info(ni)
    assert ni.kube_alloc[RESOURCE_GPU] == 1


def test_post_cardfix_capture_omits_phantom_card_node():
    """Verbatim capture from the card-fix binary (session P): the
    containerized box injects renderD but not card*, so card_path must
    be empty — the manager and CDI then never name a missing node."""
    path = os.path.join(os.path.dirname(__file__), "fixtures",
                        "real_1x_mi355x_r2b.json")
    with open(path) as f:
        info = GpusInfo.from_json(f.read())
    g = info.devices[0]
    assert g.card_path == ""
    assert g.render_path.startswith("/dev/dri/renderD")
    # allocate path: /dev/kfd + render only, no phantom card
    mgr = create_device_plugin(FakeBackend(info))
    cluster = Cluster()
    cluster.add_node_from_manager("n", mgr)
    pod = PodInfo(name="p", running_containers={
        "c": ContainerInfo(kube_requests={RESOURCE_GPU: 1})})
    cluster.schedule(pod)
    _, devices, _ = cluster.container_allocate(pod, "c")
    assert devices[0] == "/dev/kfd"
    assert all("card" not in d for d in devices[1:])
    # CDI spec from the same capture names no card node either
    from kubegpu_amd.deviceplugin.cdi import cdi_spec
    spec = cdi_spec(info)
    for dev in spec["devices"]:
        for n in dev["containerEdits"]["deviceNodes"]:
            assert "card" not in n["path"]
