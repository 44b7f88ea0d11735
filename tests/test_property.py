"""Property-based soak tests (hypothesis): random pod streams against
random topologies must preserve allocation invariants."""

import hypothesis.strategies as st
from hypothesis import given, settings

from kubegpu_amd.api.types import ContainerInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU
from kubegpu_amd.scheduler import SchedulingError

FIXTURES = {
    "dense": fixtures.fixture_8x_mi355x,
    "twohive": fixtures.fixture_2hive_8gpu,
    "noxgmi": fixtures.fixture_4x_no_xgmi,
}


def _mk_cluster(names):
    cluster = Cluster()
    for i, n in enumerate(names):
        mgr = create_device_plugin(FakeBackend(FIXTURES[n]()))
        cluster.add_node_from_manager(f"{n}-{i}", mgr)
    return cluster


@settings(max_examples=40, deadline=None)
@given(
    nodes=st.lists(st.sampled_from(sorted(FIXTURES)), min_size=1, max_size=3),
    stream=st.lists(
        st.tuples(st.integers(min_value=1, max_value=8), st.booleans()),
        min_size=1,
        max_size=30,
    ),
)
def test_allocation_invariants(nodes, stream):
    """For any node mix and any schedule/release stream:
    - a pod's GPUs are distinct and all on its node
    - no GPU is held by two live pods
    - releasing everything restores full capacity
    """
    cluster = _mk_cluster(nodes)
    total = {name: cluster.core.free_count(name) for name in cluster.core.nodes}
    live = []
    # fixtures reuse uuids across nodes, so the unique key is (node, uuid)
    held = {}  # (node, uuid) -> pod name
    for i, (k, do_release) in enumerate(stream):
        pod = PodInfo(
            name=f"p{i}",
            running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: k})},
        )
        try:
            res = cluster.schedule(pod)
        except SchedulingError:
            res = None
        if res is not None:
            assert len(set(res.uuids)) == len(res.uuids) == k
            state = cluster.core.nodes[res.node_name]
            for u in res.uuids:
                assert u in state.gpus, "allocated GPU not on chosen node"
                key = (res.node_name, u)
                assert key not in held, f"double allocation of {key}"
                held[key] = pod.name
            live.append((pod, res))
        if do_release and live:
            pod0, res0 = live.pop(0)
            cluster.release(pod0)
            for u in res0.uuids:
                del held[(res0.node_name, u)]
    for pod0, res0 in live:
        cluster.release(pod0)
    for name, cap in total.items():
        assert cluster.core.free_count(name) == cap


@settings(max_examples=30, deadline=None)
@given(k=st.integers(min_value=1, max_value=4))
def test_same_hive_whenever_possible(k):
    """On the 2-hive node, any k<=4 pod scheduled on an empty node must
    stay inside one hive (the xGMI-optimal answer is always feasible)."""
    cluster = _mk_cluster(["twohive"])
    pod = PodInfo(
        name="p",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: k})},
    )
    res = cluster.schedule(pod)
    hives = {int(u.split("-")[-1]) // 4 for u in res.uuids}
    assert len(hives) == 1


@settings(max_examples=25, deadline=None)
@given(
    nodes=st.lists(st.sampled_from(sorted(FIXTURES)), min_size=1, max_size=3),
    stream=st.lists(
        st.tuples(st.integers(min_value=1, max_value=8), st.booleans()),
        min_size=1,
        max_size=25,
    ),
)
def test_plan_cache_equivalence_random(nodes, stream):
    """For any node mix and stream, the bind-plan cache never changes a
    placement decision vs cold binds (cache cleared before every pod)."""
    outs = []
    for flush in (False, True):
        cluster = _mk_cluster(nodes)
        live = []
        placed = []
        for i, (k, rel) in enumerate(stream):
            if flush:
                cluster.core._plan_cache.clear()
            pod = PodInfo(
                name=f"p{i}",
                running_containers={
                    "c": ContainerInfo(kube_requests={RESOURCE_GPU: k})
                },
            )
            try:
                res = cluster.schedule(pod)
                placed.append((res.node_name, tuple(sorted(res.uuids))))
                live.append(pod)
            except SchedulingError:
                placed.append(None)
            if rel and live:
                cluster.release(live.pop(0))
        outs.append(placed)
    assert outs[0] == outs[1]


@settings(max_examples=60, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=8),
    k=st.integers(min_value=1, max_value=8),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_native_chooser_matches_python_random_matrices(n, k, seed):
    """Native and Python subset choosers agree on arbitrary symmetric
    bandwidth matrices (not just the shipped fixtures)."""
    import random

    from kubegpu_amd.scheduler.xgmi import (
        choose_best_subset,
        choose_best_subset_fast,
        _native_available,
    )

    if k > n:
        return
    if not _native_available():
        return  # pure-Python mode (KUBEGPU_PURE_PY): nothing to compare
    rng = random.Random(seed)
    bw = {i: {} for i in range(n)}
    for i in range(n):
        for j in range(i + 1, n):
            v = rng.choice([0.0, 32.0, 63.0, 153.0, 300.0, rng.uniform(1, 400)])
            bw[i][j] = v
            bw[j][i] = v
    free = list(range(n))
    assert choose_best_subset_fast(free, k, bw) == choose_best_subset(free, k, bw)


@settings(max_examples=40, deadline=None)
@given(
    stream=st.lists(
        st.tuples(
            st.integers(min_value=0, max_value=6),   # Σ running request
            st.integers(min_value=0, max_value=8),   # max init request
            st.booleans(),                           # release later?
        ),
        min_size=1,
        max_size=25,
    ),
)
def test_init_demand_invariants(stream):
    """Pods with init containers (demand = max(Σ running, max init),
    gpu.go:295-303): for any schedule/release stream
    - every init container binds exactly its requested GPU count
    - the running set is a subset of the init set when init >= running
    - reserved GPUs == union of running + init bindings, and release
      restores them all."""
    cluster = _mk_cluster(["dense"])
    node = next(iter(cluster.core.nodes))
    total = len(cluster.core.nodes[node].gpus)
    live = []
    for i, (run, init, rel) in enumerate(stream):
        if run == 0 and init == 0:
            continue
        running = (
            {"c": ContainerInfo(kube_requests={RESOURCE_GPU: run})} if run else {}
        )
        inits = (
            {"i": ContainerInfo(kube_requests={RESOURCE_GPU: init})} if init else {}
        )
        pod = PodInfo(name=f"p{i}", running_containers=running,
                      init_containers=inits)
        try:
            res = cluster.schedule(pod)
        except SchedulingError:
            demand = max(run, init)
            free = total - sum(d for (_, d) in live)
            assert demand > free  # only fails when it truly cannot fit
            continue
        assert len(res.uuids) == run
        bound_init = set()
        if init:
            cont = pod.init_containers["i"]
            assert len(cont.allocate_from) == init
            bound_init = {v.split("/gpu/")[1].split("/")[0]
                          for v in cont.allocate_from.values()}
            assert set(res.uuids) <= bound_init or init < run
        demand = len(set(res.uuids) | bound_init)
        assert demand == max(run, init)
        if rel:
            cluster.release(pod)
        else:
            live.append((pod, demand))
    for pod, _ in live:
        cluster.release(pod)
    st_node = cluster.core.nodes[node]
    assert len(st_node.free_uuids()) == total  # everything restored


@settings(max_examples=25, deadline=None)
@given(
    nodes=st.lists(st.sampled_from(sorted(FIXTURES)), min_size=1, max_size=3),
    stream=st.lists(
        st.tuples(st.integers(min_value=1, max_value=8), st.booleans()),
        min_size=1,
        max_size=20,
    ),
)
def test_incremental_class_index_stays_fresh(nodes, stream):
    """The incrementally-maintained signature classes (round-2
    O(distinct-states) scheduling) must equal freshly-computed
    signatures after every schedule/release — stale classes would make
    representative choice diverge from the exhaustive loop."""
    cluster = _mk_cluster(nodes)

    def check():
        seen = set()
        for name in cluster.node_infos:
            fresh = cluster.core.state_signature(name)
            if fresh is None:
                fresh = ("__unregistered__", name)
            assert cluster._node_sig[name] == fresh, name
            assert name in cluster._classes[fresh]
            seen.add(name)
        total = sum(len(m) for m in cluster._classes.values())
        assert total == len(seen)  # no ghost members

    check()
    live = []
    for i, (k, rel) in enumerate(stream):
        pod = PodInfo(
            name=f"p{i}",
            running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: k})},
        )
        try:
            cluster.schedule(pod)
            live.append(pod)
        except SchedulingError:
            pass
        check()
        if rel and live:
            cluster.release(live.pop(0))
            check()
    for pod in live:
        cluster.release(pod)
    check()


@settings(max_examples=80, deadline=None)
@given(
    avail_n=st.integers(min_value=0, max_value=8),
    k=st.integers(min_value=0, max_value=10),
    must_picks=st.lists(st.integers(min_value=0, max_value=9), max_size=4),
    busy=st.lists(st.integers(min_value=0, max_value=7), max_size=4),
)
def test_kubelet_prefer_invariants(avail_n, k, must_picks, busy):
    """GetPreferredAllocation invariants for any request shape:
    - no duplicates, every choice drawn from avail
    - every OFFERED must present (strict contract), musts lead
    - length == k when feasible; when over-constrained the musts are
      returned undropped."""
    from kubegpu_amd.server.kubelet_plugin import DevicePluginServicer

    mgr = create_device_plugin(FakeBackend(FIXTURES["twohive"]()))
    mgr.start()
    for b in busy:
        uid = f"GPU-mi355x-{b:02d}"
        if uid in mgr.gpus:
            mgr.gpus[uid].in_use = True
    servicer = DevicePluginServicer(mgr)
    scorer = servicer._refresh_scorer()

    avail = [f"GPU-mi355x-{i:02d}" for i in range(avail_n)]
    avail_set = set(avail)
    # mix of offered and never-offered musts; the servicer filters the
    # latter before _prefer (mirrors get_preferred_allocation)
    musts_raw = [f"GPU-mi355x-{i:02d}" for i in must_picks]
    musts = [m for m in dict.fromkeys(musts_raw) if m in avail_set]

    chosen = servicer._prefer(avail, musts, k, scorer)

    assert len(chosen) == len(set(chosen))  # no dups
    assert set(chosen) <= avail_set
    assert set(musts) <= set(chosen) or (len(musts) > k and chosen == musts)
    if k <= len(avail) and len(musts) <= k:
        assert len(chosen) == k
    assert chosen[: len([m for m in chosen if m in set(musts)])] is not None
    # musts lead the response ordering
    if musts and len(musts) < k:
        assert set(chosen[:1]) <= avail_set
