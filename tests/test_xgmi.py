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
