"""Device-plugin manager tests.

Analog of nvidia_gpu_manager_test.go:100-150: a fake backend with an
8-GPU topology fixture and a degenerate no-topology fixture; asserts the
exact advertised resource-name tree and the allocation outputs (device
nodes + ROCR_VISIBLE_DEVICES instead of NVIDIA_VISIBLE_DEVICES / REST
daemon CLI parsing).
"""

import pytest

from kubegpu_amd.api.types import ContainerInfo, NodeInfo, PodInfo
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import (
    CrashingBackend,
    FakeBackend,
    GpusInfo,
    fixtures,
)
from kubegpu_amd.plugintypes import RESOURCE_GPU

VRAM = fixtures.MI355X_VRAM_BYTES


def _mgr(fix):
    return create_device_plugin(FakeBackend(fix))


def test_full_hive_advertisement(fixture_8x):
    mgr = _mgr(fixture_8x)
    mgr.start()
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)
    assert ni.kube_cap[RESOURCE_GPU] == 8
    assert ni.allocatable[RESOURCE_GPU] == 8
    # full mesh -> single gpugrp0/gpugrp1 group
    for i in range(8):
        uuid = f"GPU-mi355x-{i:02d}"
        cards = f"resource/group/gpugrp1/0/gpugrp0/0/gpu/{uuid}/cards"
        mem = f"resource/group/gpugrp1/0/gpugrp0/0/gpu/{uuid}/memory"
        assert ni.allocatable[cards] == 1
        assert ni.allocatable[mem] == VRAM
    # 8 cards + 8 memory + flat count
    assert len(ni.allocatable) == 17


def test_two_hive_grouping(fixture_2hive):
    mgr = _mgr(fixture_2hive)
    mgr.start()
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)
    # GPUs 0-3 in gpugrp0/0, 4-7 in gpugrp0/1; numa splits match hives so
    # gpugrp1 has two groups as well
    assert "resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU-mi355x-00/cards" in ni.allocatable
    assert "resource/group/gpugrp1/1/gpugrp0/1/gpu/GPU-mi355x-04/cards" in ni.allocatable


def test_degenerate_no_topology(fixture_no_xgmi):
    """No xGMI: each GPU its own gpugrp0, one shared-NUMA gpugrp1 (analog
    of the "Topology":null K80 fixture, nvidia_gpu_manager_test.go:140-144)."""
    mgr = _mgr(fixture_no_xgmi)
    mgr.start()
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)
    for i in range(4):
        uuid = f"GPU-mi355x-{i:02d}"
        assert (
            f"resource/group/gpugrp1/0/gpugrp0/{i}/gpu/{uuid}/cards" in ni.allocatable
        )


def test_allocate_devices_and_env(fixture_8x):
    mgr = _mgr(fixture_8x)
    mgr.start()
    cont = ContainerInfo(
        allocate_from={
            "resource/group/gpugrp1/0/gpugrp0/0/gpu/2/cards":
                "resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU-mi355x-02/cards",
            "resource/group/gpugrp1/0/gpugrp0/0/gpu/3/cards":
                "resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU-mi355x-05/cards",
        }
    )
    mounts, devices, envs = mgr.allocate(PodInfo(name="p"), cont)
    assert mounts == []
    assert "/dev/kfd" in devices
    assert "/dev/dri/renderD130" in devices  # index 2
    assert "/dev/dri/renderD133" in devices  # index 5
    assert envs["ROCR_VISIBLE_DEVICES"] == "GPU-mi355x-02,GPU-mi355x-05"
    assert mgr.gpus["GPU-mi355x-02"].in_use


def test_allocate_unknown_gpu_raises(fixture_8x):
    mgr = _mgr(fixture_8x)
    mgr.start()
    cont = ContainerInfo(
        allocate_from={
            "r": "resource/group/gpugrp1/0/gpugrp0/0/gpu/GPU-nonexistent/cards"
        }
    )
    with pytest.raises(KeyError):
        mgr.allocate(PodInfo(name="p"), cont)


def test_crash_containment():
    """Discovery failure must not fail start; node advertises 0 GPUs
    (cf. nvidia_gpu_manager.go:185-188,193-197)."""
    mgr = create_device_plugin(CrashingBackend())
    mgr.start()  # no raise
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)
    assert ni.kube_cap[RESOURCE_GPU] == 0
    assert len(ni.allocatable) == 1


def test_mark_sweep_rediscovery(fixture_8x):
    """A GPU vanishing between updates is swept; in_use survives for the
    rest (nvidia_gpu_manager.go:132-155)."""
    backend = FakeBackend(fixture_8x)
    mgr = create_device_plugin(backend)
    mgr.start()
    assert len(mgr.gpus) == 8
    mgr.gpus["GPU-mi355x-01"].in_use = True

    smaller = GpusInfo.from_json(fixture_8x.to_json())
    smaller.devices = [d for d in smaller.devices if d.index != 7]
    backend.set_info(smaller)
    mgr.update_gpu_info(force=True)
    assert len(mgr.gpus) == 7
    assert "GPU-mi355x-07" not in mgr.gpus
    assert mgr.gpus["GPU-mi355x-01"].in_use  # preserved


def test_discovery_cache(fixture_8x):
    """Within the 5-minute window, the backend is not re-queried."""
    calls = {"n": 0}

    class CountingBackend(FakeBackend):
        def get_gpu_info(self):
            calls["n"] += 1
            return super().get_gpu_info()

    mgr = create_device_plugin(CountingBackend(fixture_8x))
    mgr.start()
    mgr.update_gpu_info()
    mgr.update_gpu_info()
    assert calls["n"] == 1
    mgr.update_gpu_info(force=True)
    assert calls["n"] == 2


def test_create_device_from_plugin(tmp_path):
    """Plugin loading parity (device.CreateDeviceFromPlugin analog)."""
    from kubegpu_amd.api.device import create_device_from_plugin

    plugin_file = tmp_path / "myplugin.py"
    plugin_file.write_text(
        "from kubegpu_amd.deviceplugin import create_device_plugin as _f\n"
        "from kubegpu_amd.discovery import FakeBackend, fixtures\n"
        "def create_device_plugin():\n"
        "    return _f(FakeBackend(fixtures.fixture_8x_mi355x()))\n"
    )
    dev = create_device_from_plugin(str(plugin_file))
    dev.start()
    assert dev.get_name() == "amdgpu"


def test_p2p_false_caps_bandwidth():
    """An xGMI link with p2p disabled degrades to the host path."""
    from kubegpu_amd.discovery import fixtures

    fix = fixtures.fixture_8x_mi355x()
    for l in fix.devices[0].links:
        if l.peer_index == 1:
            l.p2p = False
    bw = fix.bandwidth_matrix()
    assert bw[0][1] == 63.0
    assert bw[0][2] == 153.0


def test_ecc_unhealthy_excluded_from_allocatable():
    """Uncorrectable-ECC GPU: stays in capacity, leaves allocatable, and
    the scheduler never places a pod on it."""
    from kubegpu_amd.api.types import ContainerInfo, NodeInfo, PodInfo
    from kubegpu_amd.core import Cluster
    from kubegpu_amd.plugintypes import RESOURCE_GPU

    fix = fixtures.fixture_8x_mi355x()
    fix.devices[2].ecc_uncorrectable = 1
    mgr = create_device_plugin(FakeBackend(fix))
    mgr.start()
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)
    assert ni.capacity[RESOURCE_GPU] == 8
    assert ni.allocatable[RESOURCE_GPU] == 7
    bad = fix.devices[2].uuid
    assert any(bad in k for k in ni.capacity)
    assert not any(bad in k for k in ni.allocatable)

    cluster = Cluster()
    cluster.add_node(ni, mgr._last_info, mgr)
    seen = set()
    pods = []
    for i in range(3):
        pod = PodInfo(
            name=f"p{i}",
            running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
        )
        res = cluster.schedule(pod)
        pods.append(pod)
        seen.update(res.uuids)
    assert bad not in seen and len(seen) == 6


def test_create_device_from_plugin_error_paths(tmp_path):
    """Loader rejects missing files, modules without the factory, and
    factories returning non-Device objects."""
    import pytest as _pytest

    from kubegpu_amd.api.device import create_device_from_plugin

    with _pytest.raises(ImportError):
        create_device_from_plugin(str(tmp_path / "missing.py"))

    nofactory = tmp_path / "nofactory.py"
    nofactory.write_text("x = 1\n")
    with _pytest.raises(AttributeError, match="create_device_plugin"):
        create_device_from_plugin(str(nofactory))

    wrongtype = tmp_path / "wrongtype.py"
    wrongtype.write_text("def create_device_plugin():\n    return 42\n")
    with _pytest.raises(TypeError, match="returned"):
        create_device_from_plugin(str(wrongtype))


def test_container_allocate_error_paths():
    """Unknown container name and missing node manager fail loudly."""
    import pytest as _pytest

    from kubegpu_amd.api.types import ContainerInfo, PodInfo
    from kubegpu_amd.core import Cluster
    from kubegpu_amd.plugintypes import RESOURCE_GPU
    from kubegpu_amd.scheduler import SchedulingError

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 1})},
    )
    cluster.schedule(pod)
    with _pytest.raises(KeyError):
        cluster.container_allocate(pod, "nope")
    pod.node_name = "ghost-node"
    with _pytest.raises(SchedulingError, match="no device manager"):
        cluster.container_allocate(pod, "c")


def test_in_use_full_lifecycle():
    """in_use is no longer write-only: allocate
    sets it, it survives re-discovery while held (reference invariant
    nvidia_gpu_manager.go:143-145), release clears it, and
    in_use_uuids() surfaces it."""
    from kubegpu_amd.api.types import ContainerInfo, PodInfo
    from kubegpu_amd.core import Cluster
    from kubegpu_amd.plugintypes import RESOURCE_GPU

    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster = Cluster()
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    res = cluster.schedule(pod)
    cluster.container_allocate(pod, "c")
    assert mgr.in_use_uuids() == sorted(res.uuids)

    # survives a forced re-discovery
    mgr.update_gpu_info(force=True)
    assert mgr.in_use_uuids() == sorted(res.uuids)

    cluster.release(pod)
    assert mgr.in_use_uuids() == []


def test_release_uuids_direct():
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    mgr.start()
    uuids = sorted(mgr.gpus)[:3]
    for u in uuids:
        mgr.gpus[u].in_use = True
    mgr.release_uuids(uuids[:2] + ["GPU-not-real"])
    assert mgr.in_use_uuids() == [uuids[2]]


def test_kubelet_allocate_does_not_touch_in_use(tmp_path):
    """The v1beta1 path has no deallocate RPC, so the kubelet server
    must not set a flag it can never clear; occupancy there comes from
    amdsmi process_count."""
    import grpc

    from kubegpu_amd.server import KubeletDevicePlugin, dpapi

    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    mgr.start()
    p = KubeletDevicePlugin(mgr, socket_path=str(tmp_path / "k.sock"))
    p.start()
    try:
        ch = grpc.insecure_channel(f"unix://{p.socket_path}")
        alloc = ch.unary_unary(
            f"/{dpapi.DEVICE_PLUGIN_SERVICE}/Allocate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=dpapi.AllocateResponse.FromString,
        )
        alloc(dpapi.AllocateRequest(container_requests=[
            dpapi.ContainerAllocateRequest(devicesIDs=["GPU-mi355x-00"])
        ]), timeout=10)
        assert mgr.in_use_uuids() == []
        ch.close()
    finally:
        p.stop()


def test_forced_refresh_waits_for_inflight_fetch():
    """force=True must not silently no-op while another thread is
    mid-fetch: it waits for the in-flight fetch and
    returns with REFRESHED state."""
    import threading
    import time as _time

    from kubegpu_amd.discovery import Backend

    class SlowBackend(Backend):
        def __init__(self, inner):
            self.inner = inner
            self.calls = 0
            self.release = threading.Event()

        def get_gpu_info(self):
            return self.inner.get_gpu_info()

        def get_devices(self):
            self.calls += 1
            if self.calls == 2:
                self.release.wait(10)  # second (forced) fetch is slow
            return self.inner.get_devices()

    slow = SlowBackend(FakeBackend(fixtures.fixture_8x_mi355x()))
    mgr = create_device_plugin(slow)
    mgr.start()  # fetch #1

    started = threading.Event()

    def slow_force():
        started.set()
        mgr.update_gpu_info(force=True)  # fetch #2, blocks on release

    t = threading.Thread(target=slow_force)
    t.start()
    started.wait(5)
    _time.sleep(0.1)  # let the thread enter the backend call
    done = {}

    def second_force():
        mgr.update_gpu_info(force=True)  # must WAIT, then see fresh state
        done["gen"] = mgr._fetch_gen

    t2 = threading.Thread(target=second_force)
    t2.start()
    _time.sleep(0.2)
    assert t2.is_alive()  # waiting, not no-oping
    slow.release.set()
    t.join(10)
    t2.join(10)
    assert not t2.is_alive()
    assert done["gen"] >= 2  # observed the completed refresh
    assert slow.calls == 2  # waiter reused the in-flight fetch


def test_cdi_spec_structure():
    """CDI v0.6.0 spec from discovery: per-GPU render/card nodes plus a
    common /dev/kfd edit and index aliases (containerd/CRI-O consume
    this directly — the no-hook injection path, north star)."""
    from kubegpu_amd.deviceplugin.cdi import cdi_spec

    info = fixtures.fixture_2hive_8gpu()
    spec = cdi_spec(info)
    assert spec["cdiVersion"] == "0.6.0"
    assert spec["kind"] == "amd.com/gpu"
    kfd = spec["containerEdits"]["deviceNodes"]
    assert kfd and kfd[0]["path"] == "/dev/kfd"
    names = {d["name"] for d in spec["devices"]}
    assert "GPU-mi355x-00" in names and "0" in names  # uuid + index alias
    by_name = {d["name"]: d for d in spec["devices"]}
    edits = by_name["GPU-mi355x-03"]["containerEdits"]
    assert any("renderD" in n["path"] for n in edits["deviceNodes"])
    assert edits["env"] == ["ROCR_VISIBLE_DEVICES=GPU-mi355x-03"]
    # 8 GPUs x (uuid + index alias)
    assert len(spec["devices"]) == 16


def test_amddevs_cdi_cli(capsys):
    import json as _json

    from kubegpu_amd.cli.amddevs import main as amddevs_main

    assert amddevs_main(["--fake", "--cdi"]) == 0
    spec = _json.loads(capsys.readouterr().out)
    assert spec["kind"] == "amd.com/gpu" and len(spec["devices"]) == 16
