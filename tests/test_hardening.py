"""Hardening tests: concurrency, fault injection, degenerate inputs
(SURVEY.md §5 / §7 step 6)."""

import json
import os
import threading

import pytest

from kubegpu_amd.api.types import ContainerInfo, NodeInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, SysfsBackend, DiscoveryError, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU
from kubegpu_amd.probe.xgmi_counters import diff_link_metrics
from kubegpu_amd.scheduler import NodeTreeCache, SchedulingError


def test_concurrent_schedule_release():
    """Many threads scheduling/releasing against one cluster: no
    corruption, no double-allocation (the reference's scheduler cache is
    unsynchronized; ours must not be)."""
    cluster = Cluster()
    for n in range(2):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"n{n}", mgr)
    errors = []
    allocated = []
    lock = threading.Lock()

    def worker(tid):
        try:
            for i in range(30):
                pod = PodInfo(
                    name=f"t{tid}-{i}",
                    running_containers={
                        "c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})
                    },
                )
                try:
                    res = cluster.schedule(pod)
                except SchedulingError:
                    continue
                with lock:
                    allocated.append((res.node_name, tuple(sorted(res.uuids))))
                cluster.release(pod)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors
    # cluster drained back to fully free
    assert cluster.core.free_count("n0") == 8
    assert cluster.core.free_count("n1") == 8


def test_concurrent_cache_mutation():
    cache = NodeTreeCache()

    def res_for(shape_id):
        res = {}
        for g in range(8):
            grp = g // (2 if shape_id % 2 else 4)
            res[f"resource/group/gpugrp1/0/gpugrp0/{grp}/gpu/G{g}/cards"] = 1
        return res

    def worker(tid):
        for i in range(200):
            cache.add_node_resources(f"node{tid}", res_for(i))
        cache.remove_node(f"node{tid}")

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert len(cache) == 0
    assert cache.node_location_map == {}


def test_sysfs_backend_missing_root():
    with pytest.raises(DiscoveryError):
        SysfsBackend(root="/nonexistent/kfd").get_gpu_info()


def test_sysfs_backend_synthetic_tree(tmp_path):
    """Parse a synthetic KFD topology: 1 CPU node + 2 GPU nodes with an
    xGMI io_link between them."""
    def write_node(n, props, links=None, mem=None):
        d = tmp_path / str(n)
        d.mkdir()
        (d / "properties").write_text(
            "".join(f"{k} {v}\n" for k, v in props.items())
        )
        if links:
            for i, lp in enumerate(links):
                ld = d / "io_links" / str(i)
                ld.mkdir(parents=True)
                (ld / "properties").write_text(
                    "".join(f"{k} {v}\n" for k, v in lp.items())
                )
        if mem:
            md = d / "mem_banks" / "0"
            md.mkdir(parents=True)
            (md / "properties").write_text(
                "".join(f"{k} {v}\n" for k, v in mem.items())
            )

    write_node(0, {"simd_count": 0})  # CPU
    write_node(
        1,
        {"simd_count": 1024, "simd_per_cu": 4, "gfx_target_version": 90500,
         "drm_render_minor": 128, "device_id": 30115, "location_id": 2048,
         "domain": 0, "unique_id": 111},
        links=[{"type": 2, "node_to": 2, "weight": 15, "max_bandwidth": 153000}],
        mem={"heap_type": 1, "size_in_bytes": 309237645312},
    )
    write_node(
        2,
        {"simd_count": 1024, "simd_per_cu": 4, "gfx_target_version": 90500,
         "drm_render_minor": 129, "device_id": 30115, "location_id": 2304,
         "domain": 0, "unique_id": 222},
        links=[{"type": 2, "node_to": 1, "weight": 15, "max_bandwidth": 153000}],
        mem={"heap_type": 1, "size_in_bytes": 309237645312},
    )
    info = SysfsBackend(root=str(tmp_path)).get_devices()
    assert len(info.devices) == 2
    g = info.devices[0]
    assert g.gfx_target == "gfx950"
    assert g.render_path == "/dev/dri/renderD128"
    assert g.memory.vram_total_bytes == 309237645312
    assert len(g.links) == 1
    assert g.links[0].type == "XGMI"
    assert g.links[0].bandwidth_gbps == 153.0


def test_link_metrics_diff():
    before = {0: [{"link": 0, "type": "XGMI", "read_kb": 1000, "write_kb": 2000}]}
    after = {0: [{"link": 0, "type": "XGMI", "read_kb": 1024000 + 1000,
                  "write_kb": 2048000 + 2000}]}
    d = diff_link_metrics(before, after)
    assert d[0][0]["read_mb"] == 1000.0
    assert d[0][0]["write_mb"] == 2000.0


def test_metrics_percentiles():
    from kubegpu_amd.metrics import Metrics

    m = Metrics()
    for v in [0.001, 0.002, 0.003, 0.004]:
        m.observe_schedule(v)
    assert m.percentile(0.5) == 0.003
    m.inc_allocation()
    m.inc_failure()
    assert m.allocations == 1 and m.failures == 1


def test_zero_gpu_pod_schedules_nowhere_needed():
    """A pod with no GPU request translates to nothing and needs no bind."""
    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster.add_node_from_manager("n0", mgr)
    pod = PodInfo(name="nogpu", running_containers={"c": ContainerInfo()})
    res = cluster.schedule(pod)
    assert res.uuids == []


def test_no_double_allocation_under_contention():
    """Threads racing for a scarce node must never hold overlapping GPU
    sets simultaneously (TOCTOU between trial bind and commit)."""
    cluster = Cluster()
    mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
    cluster.add_node_from_manager("n0", mgr)
    errors = []
    overlap = []
    held = {}
    hlock = threading.Lock()

    def worker(tid):
        try:
            for i in range(40):
                pod = PodInfo(
                    name=f"t{tid}-{i}",
                    running_containers={
                        "c": ContainerInfo(kube_requests={RESOURCE_GPU: 4})
                    },
                )
                try:
                    res = cluster.schedule(pod)
                except SchedulingError:
                    continue
                mine = set(res.uuids)
                with hlock:
                    for other_tid, other in held.items():
                        if other & mine:
                            overlap.append((tid, other_tid, other & mine))
                    held[tid] = mine
                with hlock:
                    held.pop(tid, None)
                cluster.release(pod)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors
    assert not overlap, f"double-allocated GPUs: {overlap[:3]}"
    assert cluster.core.free_count("n0") == 8


def test_node_churn_during_stream():
    """Nodes joining/leaving mid-stream: placements stay valid, released
    state stays consistent, the tree cache's refcounts don't leak."""
    import random

    rng = random.Random(11)
    cluster = Cluster()
    next_node = [0]

    def add_node():
        name = f"churn{next_node[0]}"
        next_node[0] += 1
        fix = rng.choice(
            [fixtures.fixture_8x_mi355x, fixtures.fixture_2hive_8gpu]
        )()
        mgr = create_device_plugin(FakeBackend(fix))
        cluster.add_node_from_manager(name, mgr)
        return name

    names = [add_node() for _ in range(3)]
    live = []
    placed = 0
    for i in range(300):
        r = rng.random()
        if r < 0.05 and len(names) < 6:
            names.append(add_node())
        elif r < 0.10 and len(names) > 1:
            victim = names.pop(rng.randrange(len(names)))
            # release pods on the victim first (kube drains before delete)
            for pod in [p for p in live if p.node_name == victim]:
                cluster.release(pod)
                live.remove(pod)
            cluster.remove_node(victim)
        pod = PodInfo(
            name=f"p{i}",
            running_containers={
                "c": ContainerInfo(
                    kube_requests={RESOURCE_GPU: rng.choice([1, 2, 4])}
                )
            },
        )
        try:
            res = cluster.schedule(pod)
            assert res.node_name in names
            live.append(pod)
            placed += 1
        except SchedulingError:
            pass
        while len(live) > 10:
            cluster.release(live.pop(0))
    assert placed > 200
    # cache bookkeeping: refcounts must match live node registrations
    cache = cluster.scheduler.cache
    assert set(cache.node_location_map) == set(names)
    assert sum(cache._refcount.values()) == len(names)
    # every remaining tree key refcounted exactly
    assert set(cache._refcount) == set(cache.node_cache_map)


def test_amdsmi_backend_error_paths(tmp_path):
    """Subprocess containment: missing binary, crashing binary, and
    garbage output all surface as DiscoveryError (never tracebacks)."""
    import stat

    from kubegpu_amd.discovery import AmdSmiBackend, DiscoveryError, GpusInfo

    with pytest.raises(DiscoveryError, match="not found"):
        AmdSmiBackend(str(tmp_path / "nope")).get_gpu_info()

    crash = tmp_path / "crash"
    crash.write_text("#!/bin/sh\necho boom >&2\nexit 3\n")
    crash.chmod(crash.stat().st_mode | stat.S_IEXEC)
    with pytest.raises(DiscoveryError, match="rc=3"):
        AmdSmiBackend(str(crash)).get_gpu_info()

    garbage = tmp_path / "garbage"
    garbage.write_text("#!/bin/sh\necho 'not json'\n")
    garbage.chmod(garbage.stat().st_mode | stat.S_IEXEC)
    # corrupt output: DiscoveryError (NOT a raw JSONDecodeError crash)
    b = AmdSmiBackend(str(garbage))
    with pytest.raises(DiscoveryError, match="unparseable"):
        b.get_devices()
    mgr = create_device_plugin(b)
    mgr.start()  # must not raise (reference: Start ignores errors)
    assert len(mgr.gpus) == 0


def test_discovery_single_flight_nonblocking():
    """A slow backend fetch must not block concurrent readers: while one
    thread discovers, others serve the previous state immediately."""
    import time as _t

    from kubegpu_amd.discovery import Backend, fixtures

    class SlowBackend(Backend):
        def __init__(self):
            self.calls = 0

        def get_gpu_info(self):
            self.calls += 1
            if self.calls > 1:
                _t.sleep(1.0)  # simulated slow amdsmi under load
            return fixtures.fixture_8x_mi355x().to_json().encode()

    backend = SlowBackend()
    mgr = create_device_plugin(backend)
    mgr.start()  # first (fast) discovery
    started = threading.Event()

    def slow_refresh():
        started.set()
        mgr.update_gpu_info(force=True)

    th = threading.Thread(target=slow_refresh)
    th.start()
    started.wait()
    _t.sleep(0.1)  # let the slow fetch begin
    t0 = _t.perf_counter()
    # non-forced refresh must NOT wait for the slow fetch: Allocate /
    # GetPreferredAllocation call this and serve stale-while-revalidate
    # (force=True DOES wait now — ADVICE r1 #4; see
    # test_forced_refresh_waits_for_inflight_fetch)
    mgr._last_get_time = 0.0  # expire the cache so the path is exercised
    mgr.update_gpu_info()
    dt = _t.perf_counter() - t0
    assert dt < 0.5, f"concurrent update blocked {dt:.2f}s behind the fetch"
    assert len(mgr.gpus) == 8  # stale state still served
    ni = NodeInfo(name="n")
    mgr.update_node_info(ni)  # also non-blocking path
    th.join()
    assert backend.calls == 2  # single-flight: no duplicate fetch


def test_json_log_mode(tmp_path):
    """KUBEGPU_LOG_JSON=1 emits one parseable JSON object per line."""
    import subprocess, sys

    code = (
        "from kubegpu_amd.api import utils;"
        "utils.logf(0, 'hello %s', 'world');"
        "utils.errorf('bad %d', 7)"
    )
    env = dict(os.environ, KUBEGPU_LOG_JSON="1")
    out = subprocess.run([sys.executable, "-c", code], capture_output=True,
                         text=True, env=env, timeout=60)
    lines = [json.loads(l) for l in out.stderr.strip().splitlines()]
    assert lines[0]["msg"] == "hello world" and lines[0]["level"] == "info"
    assert lines[1]["msg"] == "bad 7" and lines[1]["level"] == "error"


def test_concurrent_schedule_with_node_churn():
    """Schedules racing node add/remove (watch events in production):
    the incremental class index must stay consistent and nothing may
    corrupt — the round-2 index is mutated under the scheduling lock."""
    cluster = Cluster()
    for n in range(4):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"stable{n}", mgr)
    errors = []
    stop = threading.Event()

    def churner():
        try:
            i = 0
            while not stop.is_set():
                mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
                cluster.add_node_from_manager(f"churn{i % 3}", mgr)
                cluster.remove_node(f"churn{i % 3}")
                i += 1
        except Exception as e:  # pragma: no cover
            errors.append(e)

    def scheduler_worker(tid):
        try:
            for i in range(40):
                pod = PodInfo(
                    name=f"c{tid}-{i}",
                    running_containers={
                        "c": ContainerInfo(kube_requests={RESOURCE_GPU: 2})
                    },
                )
                try:
                    cluster.schedule(pod)
                except SchedulingError:
                    continue
                cluster.release(pod)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    ch = threading.Thread(target=churner)
    workers = [threading.Thread(target=scheduler_worker, args=(t,)) for t in range(4)]
    ch.start()
    for t in workers:
        t.start()
    for t in workers:
        t.join()
    stop.set()
    ch.join()
    assert not errors
    # index consistent with live nodes
    assert set(cluster._node_sig) == set(cluster.node_infos)
    for n in range(4):
        assert cluster.core.free_count(f"stable{n}") == 8


def test_event_flusher_under_load(tmp_path):
    """The async event-file flusher loses nothing under a
    burst well past its wake batching, and every line is valid JSON."""
    from kubegpu_amd import events

    log = tmp_path / "burst.jsonl"
    trace = events.EventTrace(capacity=64, path=str(log))
    for i in range(5000):
        trace.record("schedule", pod=f"p{i}", node="n0", gpus=["g"], latency_ms=0.1)
    trace.flush()
    lines = log.read_text().splitlines()
    assert len(lines) == 5000
    recs = [json.loads(l) for l in lines[-10:]]
    assert recs[-1]["pod"] == "p4999"
    # ring stays bounded regardless
    assert len(trace.recent(1000)) == 64
