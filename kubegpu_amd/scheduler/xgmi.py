"""xGMI subset scoring — the MI355X replacement for NVML level trees.

The reference scores placements purely by tree shape (computeTreeScore,
gpuschedulerplugin/gpu.go:180-190): "same gpugrp0 => fast" is an implicit
proxy.  On MI355X the interconnect is an explicit point-to-point xGMI mesh
(7 links × ≈153 GB/s per GPU on an 8-GPU hive) and ring all-reduce over a
GPU subset is bound by the *thinnest link on the ring*, so we score a
candidate k-subset directly:

    score(S) = bottleneck bandwidth of the best ring over S
               (bottleneck-TSP, exact for k <= 8 via bitmask DP)

with a fragmentation tie-breaker: prefer subsets that keep the remaining
free set maximally xGMI-connected (BASELINE.json config 4 — bin-pack
without xGMI fragmentation).

A C++ twin of this module lives in csrc/schedcore.cpp (pybind11); results
are checked for equivalence in tests and the native path is preferred when
built (p50 schedule latency is a headline metric).
"""

from __future__ import annotations

import itertools
import math
import os
from typing import Dict, Iterable, List, Optional, Sequence, Tuple

# Links at or above this count as xGMI-class for fragmentation purposes.
XGMI_CLASS_GBPS = 100.0

# Above this many free devices the exact C(n, k) enumeration is replaced
# by the bounded-time heuristic (choose_best_subset_heuristic): a CPX
# 8-OAM node enumerates 64 processors and C(64, 8) ≈ 4.4e9 subsets.
HEURISTIC_FREE_THRESHOLD = 10

BwMatrix = Dict[int, Dict[int, float]]


def _sym_bw(bw: BwMatrix, i: int, j: int) -> float:
    a = bw.get(i, {}).get(j, 0.0)
    b = bw.get(j, {}).get(i, 0.0)
    if a and b:
        return min(a, b)
    return a or b


def best_ring(subset: Sequence[int], bw: BwMatrix) -> Tuple[float, List[int]]:
    """Max-bottleneck ring over *subset*.

    Returns (bottleneck GB/s, ring order).  k=1 returns (inf, [g]) — the
    degenerate single-GPU case has no interconnect bound.  Exact for the
    node sizes this scheduler sees (k <= 8): DP over (visited-mask, last)
    maximizing the minimum edge, closing the cycle back to the start.
    """
    sub = list(subset)
    k = len(sub)
    if k == 0:
        return 0.0, []
    if k == 1:
        return math.inf, sub
    if k == 2:
        b = _sym_bw(bw, sub[0], sub[1])
        return b, sub
    start = sub[0]
    rest = sub[1:]
    m = len(rest)
    # dp[(mask, last)] = best achievable min-edge for a path start->...->rest[last]
    dp: Dict[Tuple[int, int], float] = {}
    parent: Dict[Tuple[int, int], Tuple[int, int]] = {}
    for i, g in enumerate(rest):
        dp[(1 << i, i)] = _sym_bw(bw, start, g)
    for mask in range(1, 1 << m):
        for last in range(m):
            if not (mask >> last) & 1:
                continue
            cur = dp.get((mask, last))
            if cur is None:
                continue
            for nxt in range(m):
                if (mask >> nxt) & 1:
                    continue
                edge = _sym_bw(bw, rest[last], rest[nxt])
                val = min(cur, edge)
                key = (mask | (1 << nxt), nxt)
                if val > dp.get(key, -1.0):
                    dp[key] = val
                    parent[key] = (mask, last)
    full = (1 << m) - 1
    best_val, best_last = -1.0, -1
    for last in range(m):
        cur = dp.get((full, last))
        if cur is None:
            continue
        closed = min(cur, _sym_bw(bw, rest[last], start))
        if closed > best_val:
            best_val, best_last = closed, last
    # reconstruct
    order = []
    key = (full, best_last)
    while key in parent:
        order.append(rest[key[1]])
        key = parent[key]
    order.append(rest[key[1]])
    order.append(start)
    order.reverse()
    return max(best_val, 0.0), order


def xgmi_edges(gpus: Iterable[int], bw: BwMatrix) -> int:
    """Number of xGMI-class edges inside *gpus*."""
    g = sorted(gpus)
    return sum(
        1
        for a, b in itertools.combinations(g, 2)
        if _sym_bw(bw, a, b) >= XGMI_CLASS_GBPS
    )


def score_subset(
    subset: Sequence[int], free: Iterable[int], bw: BwMatrix
) -> Tuple[float, int, float]:
    """(ring bottleneck GB/s, remaining xGMI edges, aggregate ring bw)."""
    ring_bw, order = best_ring(subset, bw)
    remaining = [g for g in free if g not in set(subset)]
    frag = xgmi_edges(remaining, bw)
    if len(order) >= 3:
        agg = sum(_sym_bw(bw, order[i], order[(i + 1) % len(order)]) for i in range(len(order)))
    elif len(order) == 2:
        agg = _sym_bw(bw, order[0], order[1])
    else:
        agg = 0.0
    cap = 1e9 if math.isinf(ring_bw) else ring_bw
    return cap, frag, agg


def choose_best_subset(
    free: Sequence[int], k: int, bw: BwMatrix, must: Sequence[int] = ()
) -> List[int]:
    """Best k-subset of *free*: max ring bandwidth, then least
    fragmentation of the remainder, then max aggregate ring bandwidth,
    then lexicographically smallest (determinism).  *must* constrains
    the search to supersets of that set (kubelet's
    must_include_deviceIDs contract).

    Returns [] when k > len(free) or the constraint is unsatisfiable.
    """
    free_sorted = sorted(free)
    must_set = set(must)
    if k <= 0 or k > len(free_sorted) or len(must_set) > k:
        return []
    if not must_set.issubset(free_sorted):
        return []
    if k == len(free_sorted):
        return free_sorted
    best: Tuple[float, int, float] = (-1.0, -1, -1.0)
    best_sub: List[int] = []
    for sub in itertools.combinations(free_sorted, k):
        if not must_set.issubset(sub):
            continue
        s = score_subset(sub, free_sorted, bw)
        if s > best:
            best = s
            best_sub = list(sub)
    return best_sub


def choose_best_subset_heuristic(
    free: Sequence[int],
    k: int,
    bw: BwMatrix,
    must: Sequence[int] = (),
    max_passes: int = 4,
) -> List[int]:
    """Bounded-time subset chooser for free sets too large to enumerate.

    Multi-seed greedy max-min growth followed by steepest-descent 1-swap
    local search on the full (ring bottleneck, fragmentation, aggregate)
    objective — the same ordering the exact chooser maximizes.  Python
    twin of _schedcore.choose_best_subset_heuristic (csrc/schedcore.cpp);
    property-tested against the exact chooser for n <= 10 and used above
    HEURISTIC_FREE_THRESHOLD (CPX nodes enumerate up to 64 processors;
    the reference's analog, computeTreeScore at gpu.go:180-190, never
    faced this because NVML trees cap at 8 leaves per level).
    """
    free_sorted = sorted(set(free))
    must_set = set(must)
    if k <= 0 or k > len(free_sorted) or len(must_set) > k:
        return []
    if not must_set.issubset(free_sorted):
        return []
    if k == len(free_sorted):
        return free_sorted

    deg = {v: 0 for v in free_sorted}
    e_free = 0
    for a, b in itertools.combinations(free_sorted, 2):
        if _sym_bw(bw, a, b) >= XGMI_CLASS_GBPS:
            deg[a] += 1
            deg[b] += 1
            e_free += 1

    def frag_of(sub: Sequence[int]) -> int:
        # frag(S) = E_free - Σ_{v∈S} deg_free(v) + E_within(S): O(k²)
        within = sum(
            1
            for a, b in itertools.combinations(sub, 2)
            if _sym_bw(bw, a, b) >= XGMI_CLASS_GBPS
        )
        return e_free - sum(deg[v] for v in sub) + within

    def eval_sub(sub: Sequence[int]) -> Tuple[float, int, float]:
        sub = sorted(sub)
        ring, order = best_ring(sub, bw)
        if len(order) >= 3:
            agg = sum(
                _sym_bw(bw, order[i], order[(i + 1) % len(order)])
                for i in range(len(order))
            )
        elif len(order) == 2:
            agg = _sym_bw(bw, order[0], order[1])
        else:
            agg = 0.0
        cap = 1e9 if math.isinf(ring) else ring
        return (cap, frag_of(sub), agg)

    def grow(seed: Sequence[int]) -> List[int]:
        """Ring-insertion growth: insert the (vertex, position) pair that
        maximizes the new ring's bottleneck.  Unlike max-min-to-set
        growth this tolerates weak intra-set edges the ring can bypass
        (a degraded-mesh hive is still the right subset if a Hamiltonian
        cycle avoids its weak links)."""
        ring = list(seed)
        in_s = set(ring)
        while len(ring) < k:
            m = len(ring)
            edges = [
                _sym_bw(bw, ring[i], ring[(i + 1) % m]) for i in range(m)
            ] if m >= 2 else []
            # min of all ring edges except position p (prefix/suffix mins)
            if edges:
                pre = [math.inf] * (m + 1)
                suf = [math.inf] * (m + 1)
                for i in range(m):
                    pre[i + 1] = min(pre[i], edges[i])
                for i in range(m - 1, -1, -1):
                    suf[i] = min(suf[i + 1], edges[i])
            best_key, best_v, best_pos = None, None, 0
            for v in free_sorted:  # ascending: lowest index wins ties
                if v in in_s:
                    continue
                if m == 1:
                    b = _sym_bw(bw, ring[0], v)
                    key = (b, b)
                    if best_key is None or key > best_key:
                        best_key, best_v, best_pos = key, v, 0
                    continue
                for p in range(m):
                    u, w = ring[p], ring[(p + 1) % m]
                    others = min(pre[p], suf[p + 1])
                    nb = min(others, _sym_bw(bw, u, v), _sym_bw(bw, v, w))
                    key = (nb, _sym_bw(bw, u, v) + _sym_bw(bw, v, w))
                    if best_key is None or key > best_key:
                        best_key, best_v, best_pos = key, v, p
            ring.insert(best_pos + 1, best_v)
            in_s.add(best_v)
        return ring

    seed_base = sorted(must_set)
    seeds: List[List[int]] = []
    if len(seed_base) == k:
        seeds.append(seed_base)
    else:
        for v in free_sorted:
            if v not in must_set:
                seeds.append(seed_base + [v])
        if seed_base:
            seeds.append(seed_base)

    best_score: Optional[Tuple[float, int, float]] = None
    best_sub: List[int] = []
    for s in seeds:
        cand = sorted(grow(s))
        sc = eval_sub(cand)
        if best_score is None or sc > best_score:
            best_score, best_sub = sc, cand
        elif sc == best_score and cand < best_sub:
            best_sub = cand

    for _ in range(max_passes):
        improved = False
        pass_best, pass_sub = best_score, best_sub
        in_best = set(best_sub)
        for out_i, out in enumerate(best_sub):
            if out in must_set:
                continue
            for v in free_sorted:
                if v in in_best:
                    continue
                cand = sorted(best_sub[:out_i] + [v] + best_sub[out_i + 1:])
                sc = eval_sub(cand)
                if sc > pass_best or (sc == pass_best and cand < pass_sub):
                    pass_best, pass_sub = sc, cand
                    improved = True
        if not improved:
            break
        best_score, best_sub = pass_best, pass_sub
    return best_sub


class TopologyScorer:
    """Per-node memoized subset scorer.

    Holds the node's symmetrized bandwidth matrix flattened once (the
    per-call rebuild dominated the schedule path at ~0.75 ms/pod) and
    memoizes subset choices and ring bottlenecks by (free-set, k) — the
    topology is static per node, so steady-state pod streams hit the
    memo almost always.  Uses the native _schedcore twin when built.
    """

    def __init__(self, indices: Sequence[int], bw: BwMatrix):
        self.idx = sorted(set(indices))
        self.n = len(self.idx)
        self.pos = {g: i for i, g in enumerate(self.idx)}
        self.flat = [0.0] * (self.n * self.n)
        for a in range(self.n):
            for b in range(self.n):
                if a != b:
                    self.flat[a * self.n + b] = _sym_bw(bw, self.idx[a], self.idx[b])
        self._native = None
        if not os.environ.get("KUBEGPU_PURE_PY"):
            try:
                from .. import _schedcore

                self._native = _schedcore
            except ImportError:
                pass
        self._bw = bw
        self._choose_memo: Dict[Tuple, List[int]] = {}
        self._ring_memo: Dict[Tuple, float] = {}
        self._edges_memo: Dict[Tuple, int] = {}

    def choose(
        self, free: Sequence[int], k: int, must: Sequence[int] = ()
    ) -> List[int]:
        must_t = tuple(sorted(set(must)))
        key = (tuple(sorted(free)), k, must_t)
        hit = self._choose_memo.get(key)
        if hit is not None:
            return list(hit)
        free_sorted = sorted(free)
        if k <= 0 or k > len(free_sorted) or len(must_t) > k:
            out: List[int] = []
        elif not set(must_t).issubset(free_sorted):
            out = []
        elif len(free_sorted) > HEURISTIC_FREE_THRESHOLD and k < len(free_sorted):
            out = self._choose_heuristic(free_sorted, k, must_t)
        elif self._native is not None:
            if len(free_sorted) == self.n:
                picked = self._native.choose_best_subset(
                    self.n, k, self.flat, [self.pos[g] for g in must_t]
                )
                out = [self.idx[p] for p in picked]
            else:
                # sub-matrix for the current free set
                m = len(free_sorted)
                sub = [0.0] * (m * m)
                for a in range(m):
                    pa = self.pos[free_sorted[a]]
                    for b in range(m):
                        if a != b:
                            sub[a * m + b] = self.flat[pa * self.n + self.pos[free_sorted[b]]]
                subpos = {g: i for i, g in enumerate(free_sorted)}
                picked = self._native.choose_best_subset(
                    m, k, sub, [subpos[g] for g in must_t]
                )
                out = [free_sorted[p] for p in picked]
        else:
            out = choose_best_subset(free_sorted, k, self._bw, must_t)
        self._choose_memo[key] = out
        return list(out)

    def _choose_heuristic(
        self, free_sorted: List[int], k: int, must_t: Tuple[int, ...]
    ) -> List[int]:
        """Bounded-time path for large free sets (CPX: up to 64)."""
        if self._native is not None and hasattr(
            self._native, "choose_best_subset_heuristic"
        ):
            m = len(free_sorted)
            if m == self.n:
                picked = self._native.choose_best_subset_heuristic(
                    self.n, k, self.flat, [self.pos[g] for g in must_t]
                )
                return [self.idx[p] for p in picked]
            sub = [0.0] * (m * m)
            for a in range(m):
                pa = self.pos[free_sorted[a]]
                for b in range(m):
                    if a != b:
                        sub[a * m + b] = self.flat[pa * self.n + self.pos[free_sorted[b]]]
            subpos = {g: i for i, g in enumerate(free_sorted)}
            picked = self._native.choose_best_subset_heuristic(
                m, k, sub, [subpos[g] for g in must_t]
            )
            return [free_sorted[p] for p in picked]
        return choose_best_subset_heuristic(free_sorted, k, self._bw, must_t)

    def ring_bw(self, subset: Sequence[int]) -> float:
        key = tuple(sorted(subset))
        hit = self._ring_memo.get(key)
        if hit is not None:
            return hit
        if self._native is not None and all(g in self.pos for g in key):
            val, _ = self._native.best_ring(
                self.n, [self.pos[g] for g in key], self.flat
            )
        else:
            val, _ = best_ring(sorted(subset), self._bw)
        val = min(val, 1e9)
        self._ring_memo[key] = val
        return val

    def edges(self, gpus: Sequence[int]) -> int:
        key = tuple(sorted(gpus))
        hit = self._edges_memo.get(key)
        if hit is None:
            hit = xgmi_edges(gpus, self._bw)
            self._edges_memo[key] = hit
        return hit


def _native_available() -> bool:
    if os.environ.get("KUBEGPU_PURE_PY"):
        return False
    try:
        from .. import _schedcore  # noqa: F401
        return True
    except ImportError:
        return False


def choose_best_subset_fast(
    free: Sequence[int], k: int, bw: BwMatrix, must: Sequence[int] = ()
) -> List[int]:
    """Native (C++) subset chooser when built; Python fallback otherwise.
    Routes to the bounded-time heuristic above HEURISTIC_FREE_THRESHOLD."""
    if len(set(free)) > HEURISTIC_FREE_THRESHOLD and k < len(set(free)):
        if _native_available():
            from .. import _schedcore

            idx = sorted(set(free))
            must_set = set(must)
            if not must_set.issubset(idx) or len(must_set) > k:
                return []
            n = len(idx)
            pos = {g: i for i, g in enumerate(idx)}
            flat = [0.0] * (n * n)
            for a in range(n):
                for b in range(n):
                    if a != b:
                        flat[a * n + b] = _sym_bw(bw, idx[a], idx[b])
            picked = _schedcore.choose_best_subset_heuristic(
                n, k, flat, [pos[g] for g in must_set]
            )
            return [idx[p] for p in picked]
        return choose_best_subset_heuristic(free, k, bw, must)
    if _native_available():
        from .. import _schedcore

        idx = sorted(set(free))
        must_set = set(must)
        if not must_set.issubset(idx) or len(must_set) > k:
            return []
        n = len(idx)
        pos = {g: i for i, g in enumerate(idx)}
        flat = [0.0] * (n * n)
        for a in range(n):
            for b in range(n):
                if a != b:
                    flat[a * n + b] = _sym_bw(bw, idx[a], idx[b])
        picked = _schedcore.choose_best_subset(n, k, flat, [pos[g] for g in must_set])
        return [idx[p] for p in picked]
    return choose_best_subset(free, k, bw, must)
