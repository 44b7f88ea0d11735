"""xGMI subset-scoring tests (the MI355X replacement for tree scores)."""

import math

from kubegpu_amd.discovery import fixtures
from kubegpu_amd.scheduler.xgmi import (
    TopologyScorer,
    best_ring,
    choose_best_subset,
    choose_best_subset_fast,
    score_subset,
    xgmi_edges,
)


def _bw(fix):
    return fix.bandwidth_matrix()


def test_best_ring_degenerate():
    bw = _bw(fixtures.fixture_8x_mi355x())
    val, order = best_ring([3], bw)
    assert math.isinf(val) and order == [3]
    val, order = best_ring([0, 1], bw)
    assert val == 153.0


def test_best_ring_full_mesh():
    bw = _bw(fixtures.fixture_8x_mi355x())
    val, order = best_ring(list(range(8)), bw)
    assert val == 153.0
    assert sorted(order) == list(range(8))


def test_best_ring_avoids_weak_links():
    """4 GPUs where 0-1-2-3 ring exists but 0-2 and 1-3 are PCIe: the DP
    must find the xGMI ring, not the index-order one."""
    bw = {i: {} for i in range(4)}

    def link(a, b, v):
        bw[a][b] = v
        bw[b][a] = v

    link(0, 1, 153)
    link(1, 2, 153)
    link(2, 3, 153)
    link(3, 0, 153)
    link(0, 2, 63)
    link(1, 3, 63)
    val, order = best_ring([0, 1, 2, 3], bw)
    assert val == 153


def test_choose_subset_same_hive(fixture_2hive):
    """2-GPU pod on a 2-hive node lands inside one hive (config 3)."""
    bw = _bw(fixture_2hive)
    picked = choose_best_subset(list(range(8)), 2, bw)
    assert picked in ([0, 1], [0, 2], [0, 3])  # any same-hive pair; lexicographic -> [0, 1]
    assert picked == [0, 1]
    # 4-GPU pod: a whole hive, never straddling PCIe
    picked4 = choose_best_subset(list(range(8)), 4, bw)
    assert picked4 == [0, 1, 2, 3]


def test_choose_subset_anti_fragmentation(fixture_2hive):
    """With hive-0 partially used, a 2-GPU pod goes to the subset that
    keeps the remaining free set best-connected."""
    bw = _bw(fixture_2hive)
    free = [2, 3, 4, 5, 6, 7]  # 0,1 already used
    picked = choose_best_subset(free, 2, bw)
    # (2,3) finishes off hive 0 leaving hive 1 fully intact (6 edges);
    # any pair inside hive 1 would leave (2,3)+2 stragglers (1+1=2... fewer edges).
    assert picked == [2, 3]


def test_fragmentation_count(fixture_2hive):
    bw = _bw(fixture_2hive)
    assert xgmi_edges([0, 1, 2, 3], bw) == 6
    assert xgmi_edges([0, 1, 4, 5], bw) == 2
    assert xgmi_edges([], bw) == 0


def test_no_xgmi_degenerate(fixture_no_xgmi):
    bw = _bw(fixture_no_xgmi)
    val, _ = best_ring([0, 1, 2, 3], bw)
    assert val == 63.0  # PCIe-bound
    assert choose_best_subset([0, 1, 2, 3], 2, bw) == [0, 1]


def test_subset_too_large_returns_empty():
    bw = _bw(fixtures.fixture_4x_no_xgmi())
    assert choose_best_subset([0, 1], 3, bw) == []


def test_fast_path_matches_python(fixture_2hive):
    """choose_best_subset_fast (native when built) == pure Python."""
    bw = _bw(fixture_2hive)
    for free, k in [
        (list(range(8)), 2),
        (list(range(8)), 4),
        ([2, 3, 4, 5, 6, 7], 2),
        ([1, 3, 5, 7], 3),
    ]:
        assert choose_best_subset_fast(free, k, bw) == choose_best_subset(free, k, bw)


def test_degraded_mesh_avoids_down_links():
    """On a mesh with GPU0's links 0-1/0-2/0-3 down, a 4-GPU subset must
    avoid routing through GPU0's dead links (naive [0,1,2,3] would be
    PCIe-bound)."""
    fix = fixtures.fixture_degraded_mesh(missing=((0, 1), (0, 2), (0, 3), (5, 6)))
    bw = _bw(fix)
    val, _ = best_ring([0, 1, 2, 3], bw)
    assert val == 63.0  # the naive choice is PCIe-bound
    picked = choose_best_subset(list(range(8)), 4, bw)
    val2, order = best_ring(picked, bw)
    assert val2 == 153.0
    # every consecutive ring hop is a live xGMI link (the subset may
    # contain endpoints of a dead link as long as the ring avoids it)
    down = {(0, 1), (0, 2), (0, 3), (5, 6)}
    hops = {tuple(sorted((order[i], order[(i + 1) % len(order)])))
            for i in range(len(order))}
    assert not (hops & down)


def test_choose_with_must_constraint(fixture_2hive):
    """must_include forces the subset to contain the given GPUs and the
    chooser completes the set with the best-connected peers."""
    bw = _bw(fixture_2hive)
    # must=GPU5 (hive 1): the pair must be inside hive 1, containing 5
    picked = choose_best_subset(list(range(8)), 2, bw, must=[5])
    assert 5 in picked and all(g in (4, 5, 6, 7) for g in picked)
    # must spanning both hives: still honoured, ring is PCIe-bound but
    # the musts are in
    picked = choose_best_subset(list(range(8)), 4, bw, must=[0, 5])
    assert {0, 5}.issubset(picked)
    # unsatisfiable: must larger than k, or must not in free
    assert choose_best_subset(list(range(8)), 1, bw, must=[0, 5]) == []
    assert choose_best_subset([1, 2, 3], 2, bw, must=[7]) == []


def test_choose_with_must_native_matches_python(fixture_2hive):
    """Native and Python must-constrained choosers agree (scorer path)."""
    bw = _bw(fixture_2hive)
    scorer = TopologyScorer(list(range(8)), bw)
    for free, k, must in [
        (list(range(8)), 2, [5]),
        (list(range(8)), 4, [0, 5]),
        ([2, 3, 4, 5, 6, 7], 3, [2]),
        ([1, 3, 5, 7], 2, [3, 7]),
    ]:
        assert scorer.choose(free, k, must=must) == choose_best_subset(
            free, k, bw, must=must
        )


# ---- bounded-time heuristic chooser (large free sets / CPX nodes) -------

def _hive_matrix(n, hive, link=153.6, pcie=64.0, jitter_rng=None):
    """Synthetic hive-structured symmetric bandwidth matrix."""
    bw = {a: {} for a in range(n)}
    for a in range(n):
        for b in range(n):
            if a == b:
                continue
            base = link if a // hive == b // hive else pcie
            bw[a][b] = base * (jitter_rng.uniform(0.9, 1.1) if jitter_rng else 1.0)
    for a in range(n):
        for b in range(a + 1, n):
            v = min(bw[a][b], bw[b][a])
            bw[a][b] = bw[b][a] = v
    return bw


def test_heuristic_exact_parity_uniform_hives():
    """On uniform hive topologies (the real MI355X shape: identical xGMI
    links in-hive) the heuristic matches the exact chooser's score
    exactly, for every n <= 10 and k."""
    from kubegpu_amd.scheduler.xgmi import choose_best_subset_heuristic

    for n, hive in [(8, 4), (8, 8), (9, 3), (10, 5), (10, 2)]:
        bw = _hive_matrix(n, hive)
        for k in range(2, min(n, 8) + 1):
            free = list(range(n))
            ex = choose_best_subset(free, k, bw)
            he = choose_best_subset_heuristic(free, k, bw)
            se = score_subset(ex, free, bw)
            sh = score_subset(he, free, bw)
            assert sh[0] == se[0] and sh[1] == se[1], (n, hive, k, se, sh)


def test_heuristic_near_exact_on_degraded_jittered():
    """Jittered + degraded meshes: heuristic ring bottleneck within 3%
    of the exact optimum (gaps only appear under synthetic ±10% link
    jitter; uniform hardware is exact — see the uniform test above)."""
    import random

    from kubegpu_amd.scheduler.xgmi import choose_best_subset_heuristic

    worst = 1.0
    for trial in range(25):
        rng = random.Random(4200 + trial)
        n = rng.choice([8, 9, 10])
        bw = _hive_matrix(n, rng.choice([4, 5, 8]), jitter_rng=rng)
        for _ in range(rng.randint(0, 4)):
            a, b = rng.sample(range(n), 2)
            v = rng.uniform(10.0, 80.0)
            bw[a][b] = bw[b][a] = v
        for k in (2, 3, 4, 8):
            if k > n:
                continue
            free = list(range(n))
            se = score_subset(choose_best_subset(free, k, bw), free, bw)
            sh = score_subset(
                choose_best_subset_heuristic(free, k, bw), free, bw
            )
            ratio = sh[0] / se[0] if se[0] else 1.0
            worst = min(worst, ratio)
            assert ratio >= 0.97, (trial, n, k, se, sh)
    assert worst <= 1.0 + 1e-9  # heuristic can never beat the exact optimum


def test_heuristic_native_matches_python():
    """The C++ twin (csrc/schedcore.cpp choose_best_subset_heuristic)
    and the Python reference return identical subsets."""
    import random

    import pytest

    from kubegpu_amd.scheduler.xgmi import (
        _sym_bw,
        choose_best_subset_heuristic,
    )

    try:
        from kubegpu_amd import _schedcore
    except ImportError:
        pytest.skip("native _schedcore not built")
    for trial in range(10):
        rng = random.Random(77 + trial)
        n = rng.choice([8, 12, 16])
        bw = _hive_matrix(n, rng.choice([4, 8]), jitter_rng=rng)
        flat = [0.0] * (n * n)
        for a in range(n):
            for b in range(n):
                if a != b:
                    flat[a * n + b] = _sym_bw(bw, a, b)
        for k in (2, 4, 8):
            for must in ([], [1], [0, n - 1]):
                he = choose_best_subset_heuristic(list(range(n)), k, bw, must)
                hn = _schedcore.choose_best_subset_heuristic(n, k, flat, must)
                assert he == hn, (trial, n, k, must, he, hn)


def test_heuristic_must_include_respected():
    from kubegpu_amd.scheduler.xgmi import choose_best_subset_heuristic

    bw = _hive_matrix(16, 8)
    picked = choose_best_subset_heuristic(list(range(16)), 4, bw, must=[9])
    assert 9 in picked
    assert all(8 <= g < 16 for g in picked)  # stays inside hive 1
    # unsatisfiable constraints return []
    assert choose_best_subset_heuristic(list(range(16)), 2, bw, must=[0, 5, 9]) == []
    assert choose_best_subset_heuristic(list(range(8)), 2, bw, must=[12]) == []


def test_choose_64_of_cpx_node_under_10ms():
    """CPX reality check: a 64-processor node
    (8 OAMs x 8 partitions) picks a whole OAM for k=8 in bounded time.
    The 10 ms bound needs the native chooser; Python-only gets a looser
    bound (it is the fallback, not the production path)."""
    import time

    from kubegpu_amd.scheduler.xgmi import (
        HEURISTIC_FREE_THRESHOLD,
        choose_best_subset_fast,
    )

    assert HEURISTIC_FREE_THRESHOLD < 64
    bw = _hive_matrix(64, 8)
    t0 = time.perf_counter()
    picked = choose_best_subset_fast(list(range(64)), 8, bw)
    elapsed_ms = (time.perf_counter() - t0) * 1e3
    assert len(picked) == 8
    assert len({g // 8 for g in picked}) == 1  # one whole OAM
    from kubegpu_amd.scheduler.xgmi import _native_available

    if _native_available():
        # generous CI margin over the measured ~2 ms on hardware; the
        # 10 ms claim is pinned by profiles/choose64_timing_mi355x.json
        assert elapsed_ms < 200, elapsed_ms
    else:
        assert elapsed_ms < 5000, elapsed_ms


def test_scorer_routes_large_free_sets_to_heuristic():
    """TopologyScorer.choose on a 64-device node completes fast and
    picks an in-hive subset (the exact path would enumerate C(64,8))."""
    import time

    bw = _hive_matrix(64, 8)
    scorer = TopologyScorer(list(range(64)), bw)
    t0 = time.perf_counter()
    picked = scorer.choose(list(range(64)), 8)
    elapsed = time.perf_counter() - t0
    assert len({g // 8 for g in picked}) == 1
    assert elapsed < 5.0
    # memoized second call
    assert scorer.choose(list(range(64)), 8) == picked
