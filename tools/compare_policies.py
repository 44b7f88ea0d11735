#!/usr/bin/env python3
"""Head-to-head: xGMI-aware placement vs the reference's policy.

Replays identical random pod streams through two clusters:

* ``naive``  — the reference's observable policy: densest-group packing
  with first-fit group matching, lowest-index GPU choice inside groups,
  no bandwidth model, arbitrary (first-fit) node choice
  (cf. gpuschedulerplugin/gpu.go:247-271 + score-0 fit,
  gpu_scheduler.go:43);
* ``xgmi``   — this build: max-bottleneck-ring subset scoring with
  anti-fragmentation bin-packing and bandwidth-ranked node choice.

For every placed multi-GPU pod we record the model ring bottleneck
bandwidth of its GPU set (the quantity the RCCL probe measures), plus
hive-straddle counts and schedule failures.  Topology: 2-hive nodes
(4+4 xGMI islands bridged by PCIe) — the regime where placement matters
(a healthy full-mesh node makes every subset equal; partitioned /
degraded meshes do not).

Usage: python tools/compare_policies.py [--pods 2000] [--nodes 4] [--seed 7]
"""

from __future__ import annotations

import argparse
import json
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubegpu_amd.api.types import ContainerInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU
from kubegpu_amd.scheduler import SchedulingError


def build_cluster(policy: str, n_nodes: int, topology: str, rng) -> Cluster:
    cluster = Cluster(policy=policy)
    for i in range(n_nodes):
        if topology == "2hive":
            fix = fixtures.fixture_2hive_8gpu()
        elif topology == "mixedfleet":
            # heterogeneous fleet: healthy full-mesh, partitioned 2-hive
            # and link-degraded nodes side by side — the regime where
            # cross-node choice (which the reference lacks entirely:
            # first-fit, score 0.0) dominates placement quality
            kind = i % 3
            if kind == 0:
                fix = fixtures.fixture_8x_mi355x()
            elif kind == 1:
                fix = fixtures.fixture_2hive_8gpu()
            else:
                all_pairs = [(a, b) for a in range(8) for b in range(a + 1, 8)]
                fix = fixtures.fixture_degraded_mesh(rng.sample(all_pairs, k=8))
        else:  # degraded: random subset of links down per node
            all_pairs = [(a, b) for a in range(8) for b in range(a + 1, 8)]
            missing = rng.sample(all_pairs, k=6)
            fix = fixtures.fixture_degraded_mesh(missing)
        mgr = create_device_plugin(FakeBackend(fix))
        cluster.add_node_from_manager(f"node{i}", mgr)
    return cluster


def run(policy: str, stream, n_nodes: int, topology: str, seed: int):
    import random as _random

    cluster = build_cluster(policy, n_nodes, topology, _random.Random(seed + 1))
    live = []
    ring_bws = []
    straddles = 0
    failures = 0
    placed_multi = 0
    for i, (k, release_prob) in enumerate(stream):
        pod = PodInfo(
            name=f"p{i}",
            running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: k})},
        )
        try:
            res = cluster.schedule(pod)
        except SchedulingError:
            failures += 1
            res = None
        if res is not None:
            live.append((pod, res))
            if k >= 2:
                placed_multi += 1
                st = cluster.core.nodes[res.node_name]
                idxs = [st.gpus[u].index for u in res.uuids]
                ring_bws.append(st.scorer.ring_bw(idxs))
                if ring_bws[-1] < 100.0 and k <= 4:
                    straddles += 1  # landed on a PCIe-bound subset
        while live and release_prob:
            pod0, _ = live.pop(0)
            cluster.release(pod0)
            break
    mean_bw = sum(ring_bws) / len(ring_bws) if ring_bws else 0.0
    return {
        "policy": policy,
        "placed_multi_gpu_pods": placed_multi,
        "mean_ring_bottleneck_gbps": round(mean_bw, 1),
        "min_ring_bottleneck_gbps": round(min(ring_bws), 1) if ring_bws else 0.0,
        "pcie_bound_small_pods": straddles,
        "schedule_failures": failures,
    }


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--pods", type=int, default=2000)
    ap.add_argument("--nodes", type=int, default=4)
    ap.add_argument("--seed", type=int, default=7)
    ap.add_argument("--topology", choices=("2hive", "degraded", "mixedfleet"),
                    default="degraded")
    args = ap.parse_args()

    rng = random.Random(args.seed)
    sizes = [1, 2, 2, 2, 4, 4, 1, 8, 2, 4]
    stream = [
        (rng.choice(sizes), rng.random() < 0.55) for _ in range(args.pods)
    ]
    out = {
        "workload": {"pods": args.pods, "nodes": args.nodes,
                     "topology": args.topology, "seed": args.seed},
        "results": [run("naive", stream, args.nodes, args.topology, args.seed),
                    run("xgmi", stream, args.nodes, args.topology, args.seed)],
    }
    naive, ours = out["results"]
    if naive["mean_ring_bottleneck_gbps"] > 0:
        out["mean_bw_improvement"] = round(
            ours["mean_ring_bottleneck_gbps"] / naive["mean_ring_bottleneck_gbps"], 3
        )
    print(json.dumps(out, indent=1))
    return 0


if __name__ == "__main__":
    sys.exit(main())
