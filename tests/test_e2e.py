"""End-to-end scheduling tests.

Analog of the reference's scheduler end-to-end test
(gpuschedulerplugin/gpu_test.go:61-112: 3-GPU pod prefers a dense
gpugrp0; after removing the dense node the request splits 2+1) plus the
BASELINE.json scenarios: same-hive 2-GPU placement (config 3), bin-pack
contention without xGMI fragmentation (config 4), whole-node 8-GPU pod
(config 5), and the container-create allocate outputs.
"""

import pytest

from kubegpu_amd.api.types import ContainerInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU
from kubegpu_amd.scheduler import SchedulingError


def _pod(name, n, containers=1):
    conts = {}
    per = n // containers
    for i in range(containers):
        conts[f"c{i}"] = ContainerInfo(kube_requests={RESOURCE_GPU: per})
    return PodInfo(name=name, running_containers=conts)


def _cluster_with(*node_fixtures):
    cluster = Cluster()
    for name, fix in node_fixtures:
        mgr = create_device_plugin(FakeBackend(fix))
        cluster.add_node_from_manager(name, mgr)
    return cluster


def test_dense_preference_and_split():
    """3-GPU pod: all 3 from one hive while a dense node exists; after
    removing it, the 2-hive node serves it 2+1... no — better-connected
    subsets still pack one hive (4 >= 3).  Removing both dense options
    forces a cross-group split."""
    cluster = _cluster_with(
        ("dense", fixtures.fixture_8x_mi355x()),
        ("twohive", fixtures.fixture_2hive_8gpu()),
    )
    pod = _pod("p3", 3)
    res = cluster.schedule(pod)
    assert res.node_name == "dense"
    reqs = sorted(pod.running_containers["c0"].dev_requests)
    # all three synthesized against a single gpugrp0
    assert all("/gpugrp1/0/gpugrp0/0/" in r for r in reqs)
    # bound to three distinct concrete GPUs
    assert len(set(pod.running_containers["c0"].allocate_from.values())) == 3

    # remove the dense node: the 2-hive node hosts all 3 inside one hive
    cluster.remove_node("dense")
    pod2 = _pod("p3b", 3)
    res2 = cluster.schedule(pod2)
    assert res2.node_name == "twohive"
    uuids = sorted(res2.uuids)
    idx = [int(u.split("-")[-1]) for u in uuids]
    assert idx == [0, 1, 2] or idx == [4, 5, 6]


def test_forced_split_across_groups():
    """5-GPU pod on the 2-hive node must straddle hives: 4+1 split."""
    cluster = _cluster_with(("twohive", fixtures.fixture_2hive_8gpu()))
    pod = _pod("p5", 5)
    res = cluster.schedule(pod)
    idx = sorted(int(u.split("-")[-1]) for u in res.uuids)
    in_h0 = sum(1 for i in idx if i < 4)
    assert in_h0 in (1, 4)  # 4+1 split, never 3+2
    assert len(idx) == 5


def test_same_hive_pair_config3():
    cluster = _cluster_with(("twohive", fixtures.fixture_2hive_8gpu()))
    res = cluster.schedule(_pod("pair", 2))
    idx = sorted(int(u.split("-")[-1]) for u in res.uuids)
    assert idx[0] // 4 == idx[1] // 4  # same hive


def test_binpack_contention_config4():
    """Two 2-GPU pods + one 4-GPU pod on 8 GPUs: the 4-GPU pod must get a
    fully-connected quad (no xGMI fragmentation)."""
    cluster = _cluster_with(("twohive", fixtures.fixture_2hive_8gpu()))
    r1 = cluster.schedule(_pod("a", 2))
    r2 = cluster.schedule(_pod("b", 2))
    r4 = cluster.schedule(_pod("c", 4))
    i1 = {int(u.split("-")[-1]) for u in r1.uuids}
    i2 = {int(u.split("-")[-1]) for u in r2.uuids}
    i4 = {int(u.split("-")[-1]) for u in r4.uuids}
    assert not (i1 & i2) and not (i1 & i4) and not (i2 & i4)
    # the two pair-pods share one hive, leaving the other intact for the quad
    assert i4 in ({0, 1, 2, 3}, {4, 5, 6, 7})


def test_whole_node_pod_config5():
    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    res = cluster.schedule(_pod("all8", 8))
    assert len(res.uuids) == 8
    # node now full
    with pytest.raises(SchedulingError):
        cluster.schedule(_pod("one-more", 1))
    # release frees capacity (ReturnPodResources analog)
    # re-fetch the pod object used: schedule mutated it
    # (release by pod identity)


def test_release_frees_capacity():
    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    pod = _pod("all8", 8)
    cluster.schedule(pod)
    cluster.release(pod)
    res = cluster.schedule(_pod("after", 2))
    assert len(res.uuids) == 2


def test_container_allocate_end_to_end():
    """Scheduled pod -> container create: /dev/kfd + render nodes +
    ROCR_VISIBLE_DEVICES for exactly the bound GPUs."""
    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    pod = _pod("p2", 2)
    res = cluster.schedule(pod)
    mounts, devices, envs = cluster.container_allocate(pod, "c0")
    assert "/dev/kfd" in devices
    renders = [d for d in devices if "renderD" in d]
    assert len(renders) == 2
    vis = envs["ROCR_VISIBLE_DEVICES"].split(",")
    assert sorted(vis) == sorted(res.uuids)


def test_multi_node_prefers_less_fragmented():
    """With one node partially used, a 4-GPU pod goes to the node that
    can give it an intact hive."""
    cluster = _cluster_with(
        ("n1", fixtures.fixture_2hive_8gpu()),
        ("n2", fixtures.fixture_2hive_8gpu()),
    )
    # consume 6 GPUs on n1 via three pair-pods... they may spread over
    # nodes; instead pin usage directly through the core:
    st = cluster.core.nodes["n1"]
    for u in list(sorted(st.gpus))[:6]:
        st.used.add(u)
    res = cluster.schedule(_pod("quad", 4))
    assert res.node_name == "n2"


def test_scheduler_latency_sanity():
    """p50 schedule latency stays well under a millisecond budget on the
    synthetic stream shape used by bench.py (informational bound)."""
    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    lat = []
    for i in range(50):
        pod = _pod(f"p{i}", 2)
        r = cluster.schedule(pod)
        lat.append(r.latency_s)
        cluster.release(pod)
    lat.sort()
    assert lat[len(lat) // 2] < 0.25  # generous CI bound


def test_flat_no_topology_path():
    """gpu-generate-topology=0: wildcard requests bind through the core
    and still land on the xGMI-best subset."""
    from kubegpu_amd.scheduler import GPU_TOPOLOGY_GENERATION

    cluster = _cluster_with(("twohive", fixtures.fixture_2hive_8gpu()))
    pod = PodInfo(
        name="flat",
        requests={GPU_TOPOLOGY_GENERATION: 0},
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    res = cluster.schedule(pod)
    assert len(res.uuids) == 2
    idx = sorted(int(u.split("-")[-1]) for u in res.uuids)
    assert idx[0] // 4 == idx[1] // 4  # same hive even via the flat path
    reqs = list(pod.running_containers["c"].dev_requests)
    assert all("/gpugrp1/*/gpugrp0/*/" in r for r in reqs)


def test_agent_oneshot_runs():
    from kubegpu_amd.server import agent

    rc = agent.main([
        "--fake", "--no-register", "--metrics-port", "0",
        "--socket", "/tmp/kubegpu-agent-test.sock", "--oneshot",
    ])
    assert rc == 0


def test_pod_fits_with_group_scheduler_dry_run():
    """pod_fits_device(run_group_scheduler=True) exercises the concrete
    binder: a full node stops fitting even though translation exists."""
    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    cluster.scheduler.group_core = cluster.core
    ni = cluster.node_infos["dense"]
    fits, _, _ = cluster.scheduler.pod_fits_device(
        ni, _pod("f2", 2), fill_allocate_from=False, run_group_scheduler=True
    )
    assert fits
    cluster.schedule(_pod("all8", 8))  # fill the node
    fits, reasons, _ = cluster.scheduler.pod_fits_device(
        ni, _pod("f2b", 2), fill_allocate_from=False, run_group_scheduler=True
    )
    assert not fits and reasons


def test_node_readvertise_preserves_allocations():
    """Re-adding a node (watch update / re-discovery) keeps live pods'
    GPUs allocated; a vanished GPU drops out cleanly."""
    from kubegpu_amd.api.types import NodeInfo
    from kubegpu_amd.discovery import GpusInfo

    cluster = _cluster_with(("dense", fixtures.fixture_8x_mi355x()))
    pod = _pod("live", 2)
    res = cluster.schedule(pod)
    assert cluster.core.free_count("dense") == 6

    # re-advertise the same node (e.g. periodic UpdateNodeInfo)
    mgr = cluster.managers["dense"]
    ni = NodeInfo(name="dense")
    mgr.update_node_info(ni)
    cluster.scheduler.add_node("dense", ni, mgr._last_info)
    cluster.core.register_node(ni, mgr._last_info)
    cluster.node_infos["dense"] = ni
    assert cluster.core.free_count("dense") == 6  # allocations survive

    # shrink the node: one held GPU vanishes
    smaller = GpusInfo.from_json(mgr._last_info.to_json())
    held = sorted(res.uuids)[0]
    smaller.devices = [d for d in smaller.devices if d.uuid != held]
    from kubegpu_amd.discovery import FakeBackend
    mgr2 = create_device_plugin(FakeBackend(smaller))
    ni2 = NodeInfo(name="dense")
    mgr2.start()
    mgr2.update_node_info(ni2)
    cluster.core.register_node(ni2, smaller)
    st = cluster.core.nodes["dense"]
    assert held not in st.gpus
    assert len(st.used) == 1  # the other held GPU is still tracked


def test_state_signature_equivalence_classes():
    """Nodes with identical topology+free state share a signature; a
    pod landing on one, or a different topology, splits the class."""
    from kubegpu_amd.discovery import FakeBackend

    cluster = Cluster()
    for n in range(3):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"eq{n}", mgr)
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_2hive_8gpu()))
    cluster.add_node_from_manager("other", mgr)
    sigs = {n: cluster.core.state_signature(n) for n in cluster.node_infos}
    assert sigs["eq0"] == sigs["eq1"] == sigs["eq2"]
    assert sigs["other"] != sigs["eq0"]
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    res = cluster.schedule(pod)
    after = {n: cluster.core.state_signature(n) for n in cluster.node_infos}
    assert after[res.node_name] != sigs[res.node_name]
    untouched = [n for n in ("eq0", "eq1", "eq2") if n != res.node_name]
    assert after[untouched[0]] == after[untouched[1]] == sigs["eq0"]
    # release restores the class
    cluster.release(pod)
    assert cluster.core.state_signature(res.node_name) == sigs[res.node_name]


def test_schedule_latency_flat_at_scale():
    """Equivalence-class dedup keeps p95 schedule latency bounded on a
    256-node cluster (was O(nodes) bind attempts per pod)."""
    import time as _t
    from kubegpu_amd.discovery import FakeBackend

    cluster = Cluster()
    for n in range(256):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"node{n:04d}", mgr)
    lat = []
    live = []
    for i in range(300):
        pod = PodInfo(
            name=f"p{i}",
            running_containers={
                "c": ContainerInfo(kube_requests={RESOURCE_GPU: [1, 2, 4, 8][i % 4]})
            },
        )
        t0 = _t.perf_counter()
        cluster.schedule(pod)
        lat.append(_t.perf_counter() - t0)
        live.append(pod)
        while len(live) > 64:
            cluster.release(live.pop(0))
    lat.sort()
    # very generous CI bound (~0.5-1 ms measured): guards against a
    # return to O(nodes) behavior, not absolute speed — shared CI boxes
    # can stall any single schedule call by tens of ms
    assert lat[int(0.95 * len(lat))] < 0.200


def test_amddevs_cli_fake_modes(capsys):
    """CLI smoke on CPU via --fake: health view and schedule mode."""
    import json as _json

    from kubegpu_amd.cli.amddevs import main as cli_main

    assert cli_main(["--fake", "--health"]) == 0
    rows = _json.loads(capsys.readouterr().out)
    assert len(rows) == 8 and all(r["healthy"] for r in rows.values())

    assert cli_main(["--fake", "--schedule", "4"]) == 0
    out = _json.loads(capsys.readouterr().out)
    assert len(out["gpus"]) == 4 and "/dev/kfd" in out["devices"]
    assert out["envs"]["ROCR_VISIBLE_DEVICES"].count(",") == 3


def test_bind_plan_cache_equivalence():
    """Plan-cache hits produce bit-identical placements to cold binds."""
    from kubegpu_amd.discovery import FakeBackend

    def build():
        c = Cluster()
        for n in range(2):
            mgr = create_device_plugin(FakeBackend(fixtures.fixture_2hive_8gpu()))
            c.add_node_from_manager(f"n{n}", mgr)
        return c

    cached, cold = build(), build()
    seq = [2, 4, 2, 1, 8, 2, 4, 1, 2, 2, 4]
    placements = [[], []]
    live = [[], []]
    for pi, (cl, out, lv, flush) in enumerate(
        ((cached, placements[0], live[0], False), (cold, placements[1], live[1], True))
    ):
        for i, k in enumerate(seq * 3):
            if flush:
                cl.core._plan_cache.clear()  # force cold path every bind
            pod = PodInfo(
                name=f"p{i}",
                running_containers={
                    "c": ContainerInfo(kube_requests={RESOURCE_GPU: k})
                },
            )
            try:
                res = cl.schedule(pod)
                out.append((res.node_name, tuple(sorted(res.uuids))))
                lv.append(pod)
            except Exception:
                out.append(None)
            while len(lv) > 6:
                cl.release(lv.pop(0))
    assert placements[0] == placements[1]
    assert len(cached.core._plan_cache) > 0  # the cache actually engaged


def test_agent_startup_probe_stub(tmp_path, monkeypatch, capsys):
    """--startup-probe runs the probe binary and exports busbw; verified
    hermetically with a stub probe."""
    import stat

    stub = tmp_path / "rcclprobe"
    stub.write_text(
        "#!/bin/sh\n"
        'echo \'{"busbw_gbps": 321.5, "algbw_gbps": 321.5, "ndev": 8, "check": "pass"}\'\n'
    )
    stub.chmod(stub.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("KUBEGPU_RCCLPROBE", str(stub))

    from kubegpu_amd.metrics import METRICS
    from kubegpu_amd.server.agent import main as agent_main

    rc = agent_main([
        "--fake", "--no-register", "--oneshot", "--startup-probe",
        "--socket", str(tmp_path / "a.sock"), "--metrics-port", "0",
    ])
    assert rc == 0
    if hasattr(METRICS, "_xgmi"):
        assert METRICS._xgmi._value.get() == 321.5


def test_tools_smoke():
    """Scenario scorecard + policy comparison tools stay green."""
    import subprocess, sys, json as _json

    r = subprocess.run([sys.executable, "tools/scenario_report.py"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout[-500:] + r.stderr[-500:]
    rep = _json.loads(r.stdout)
    assert rep["all_passed"]

    r = subprocess.run(
        [sys.executable, "tools/compare_policies.py", "--pods", "200",
         "--nodes", "2", "--topology", "mixedfleet"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-500:]
    out = _json.loads(r.stdout)
    naive, ours = out["results"]
    assert ours["pcie_bound_small_pods"] <= naive["pcie_bound_small_pods"]


def test_multi_container_pod_with_init():
    """A pod with two GPU containers + an init container binds distinct
    GPUs per app container; the init container reuses the pod's set
    (reference: pod demand = max(Σ running, max init), gpu.go:295-303)."""
    from kubegpu_amd.discovery import FakeBackend

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_2hive_8gpu()))
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(
        name="multi",
        running_containers={
            "trainer": ContainerInfo(kube_requests={RESOURCE_GPU: 2}),
            "loader": ContainerInfo(kube_requests={RESOURCE_GPU: 1}),
        },
        init_containers={"warm": ContainerInfo(kube_requests={RESOURCE_GPU: 1})},
    )
    res = cluster.schedule(pod)
    assert len(res.uuids) == 3  # 2 + 1 running; init reuses
    m_t, d_t, e_t = cluster.container_allocate(pod, "trainer")
    m_l, d_l, e_l = cluster.container_allocate(pod, "loader")
    m_w, d_w, e_w = cluster.container_allocate(pod, "warm")
    t = set(e_t["ROCR_VISIBLE_DEVICES"].split(","))
    l = set(e_l["ROCR_VISIBLE_DEVICES"].split(","))
    w = set(e_w["ROCR_VISIBLE_DEVICES"].split(","))
    assert len(t) == 2 and len(l) == 1 and not (t & l)
    assert w <= (t | l)  # init ran on the pod's own set
    # all three on one node, all with /dev/kfd
    for d in (d_t, d_l, d_w):
        assert d[0] == "/dev/kfd"


def test_event_trace_records_schedule_and_release(tmp_path, monkeypatch):
    """EVENTS ring captures placements; KUBEGPU_EVENT_LOG appends JSONL."""
    import json as _json

    from kubegpu_amd import events
    from kubegpu_amd.discovery import FakeBackend

    log = tmp_path / "events.jsonl"
    trace = events.EventTrace(capacity=16, path=str(log))
    monkeypatch.setattr(events, "EVENTS", trace)
    import kubegpu_amd.core.cluster as cluster_mod

    monkeypatch.setattr(cluster_mod, "EVENTS", trace)

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(
        name="traced",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    cluster.schedule(pod)
    cluster.release(pod)
    recent = trace.recent()
    assert [e["event"] for e in recent] == ["schedule", "release"]
    assert recent[0]["pod"] == "traced" and len(recent[0]["gpus"]) == 2
    assert recent[0]["predicted_ring_gbps"] >= 100
    trace.flush()  # file writes are async (off the scheduling lock)
    lines = [_json.loads(l) for l in log.read_text().splitlines()]
    assert len(lines) == 2 and lines[0]["event"] == "schedule"


def test_cpx_partitioned_node_prefers_same_oam():
    """CPX mode: partitions of one OAM (INTERNAL links) outrank
    cross-OAM xGMI — a 4-partition pod gets one whole OAM, a 2-partition
    pod never straddles OAMs."""
    from kubegpu_amd.discovery import FakeBackend

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_cpx_2oam_8part()))
    cluster.add_node_from_manager("cpx0", mgr)
    st = cluster.core.nodes["cpx0"]
    # bandwidth model: INTERNAL > XGMI
    assert st.bw[0][1] > st.bw[0][4] > 100.0

    pod4 = PodInfo(
        name="oam",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 4})},
    )
    res = cluster.schedule(pod4)
    idxs = sorted(st.gpus[u].index for u in res.uuids)
    assert idxs in ([0, 1, 2, 3], [4, 5, 6, 7])  # one intact OAM

    pod2 = PodInfo(
        name="pair",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})},
    )
    res2 = cluster.schedule(pod2)
    i2 = sorted(st.gpus[u].index for u in res2.uuids)
    assert i2[0] // 4 == i2[1] // 4  # same OAM
    assert set(i2).isdisjoint(idxs)


def test_bind_unknown_node_and_cached_infeasible():
    """Group core error paths: unknown node; the cached-infeasible plan
    raises identically on repeat (no silent success from the cache)."""
    import pytest as _pytest

    from kubegpu_amd.core import GroupScheduler
    from kubegpu_amd.discovery import FakeBackend
    from kubegpu_amd.scheduler import SchedulingError

    core = GroupScheduler()
    with _pytest.raises(SchedulingError, match="unknown node"):
        core.bind_pod("ghost", PodInfo(name="p"))

    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_4x_no_xgmi()))
    cluster.add_node_from_manager("n0", mgr)
    big = PodInfo(
        name="big",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 8})},
    )
    with _pytest.raises(SchedulingError):
        cluster.schedule(big)  # cold path
    with _pytest.raises(SchedulingError):
        cluster.schedule(big)  # cached-infeasible path, same outcome


def test_8run_8init_pod_schedules_on_8gpu_node():
    """The round-1 regression case: an 8-GPU pod with an 8-GPU init
    container schedules on an 8-GPU node (pod demand is max(Σ running,
    max init), /root/reference/gpuschedulerplugin/gpu.go:295-303 — the
    init set reuses the running set entirely)."""
    cluster = _cluster_with(("n0", fixtures.fixture_8x_mi355x()))
    pod = PodInfo(
        name="whole-node",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 8})},
        init_containers={"warm": ContainerInfo(kube_requests={RESOURCE_GPU: 8})},
    )
    res = cluster.schedule(pod)
    assert len(res.uuids) == 8
    _, _, e_c = cluster.container_allocate(pod, "c")
    _, _, e_w = cluster.container_allocate(pod, "warm")
    assert set(e_w["ROCR_VISIBLE_DEVICES"].split(",")) == set(
        e_c["ROCR_VISIBLE_DEVICES"].split(",")
    )


def test_init_bigger_than_running_draws_extras_and_reserves_them():
    """4 running + 6-GPU init on an 8-GPU node: schedules (demand 6),
    the init container gets the running set plus 2 extras, and the
    extras stay reserved for the pod's lifetime (a following 4-GPU pod
    must NOT fit while only 2 GPUs remain unreserved) then free on
    release."""
    cluster = _cluster_with(("n0", fixtures.fixture_8x_mi355x()))
    pod = PodInfo(
        name="init-heavy",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: 4})},
        init_containers={"i": ContainerInfo(kube_requests={RESOURCE_GPU: 6})},
    )
    res = cluster.schedule(pod)
    assert len(res.uuids) == 4
    _, _, e_c = cluster.container_allocate(pod, "c")
    _, _, e_i = cluster.container_allocate(pod, "i")
    run_set = set(e_c["ROCR_VISIBLE_DEVICES"].split(","))
    init_set = set(e_i["ROCR_VISIBLE_DEVICES"].split(","))
    assert len(init_set) == 6 and run_set <= init_set
    # effective reservation: 8 - 6 = 2 free -> a 4-GPU pod cannot fit
    with pytest.raises(SchedulingError):
        cluster.schedule(_pod("too-big", 4))
    # ...but a 2-GPU pod can
    p2 = _pod("small", 2)
    cluster.schedule(p2)
    cluster.release(p2)
    cluster.release(pod)
    # everything freed: whole node schedulable again
    cluster.schedule(_pod("whole", 8))
