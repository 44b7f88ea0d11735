"""amddevs — operator CLI (analog of the reference's nvidiadevs,
/root/reference/nvidiagpuplugin/cmd/main.go:13-45).

Modes:
  amddevs                 raw discovery dump (GpusInfo JSON)
  amddevs --plugin        full device-plugin path: New/Start/UpdateNodeInfo,
                          print the advertised NodeInfo
  amddevs --schedule K    schedule a synthetic K-GPU pod against the local
                          node and print the chosen GPU set + allocation
  amddevs --probe K       same, then run the in-pod RCCL probe over the set
  amddevs --health        per-GPU health view (ECC totals, tombstones)
"""

from __future__ import annotations

import argparse
import json
import sys

from ..api.types import ContainerInfo, NodeInfo, PodInfo
from ..core import Cluster
from ..deviceplugin import create_device_plugin
from ..discovery import default_backend
from ..plugintypes import RESOURCE_GPU


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="amddevs")
    p.add_argument("--plugin", action="store_true", help="run the device-plugin path")
    p.add_argument("--schedule", type=int, metavar="K", help="schedule a K-GPU pod")
    p.add_argument("--probe", type=int, metavar="K", help="schedule + RCCL probe")
    p.add_argument("--bytes", type=int, default=256 << 20)
    p.add_argument("--health", action="store_true",
                   help="per-GPU health view (ECC totals, tombstones)")
    p.add_argument("--cdi", action="store_true",
                   help="emit a CDI v0.6.0 spec for the node's GPUs "
                        "(write to /etc/cdi/amd.com-gpu.json to activate)")
    p.add_argument("--fake", action="store_true",
                   help="use the 8xMI355X fixture backend (no GPU needed)")
    from .. import __version__

    p.add_argument("--version", action="version",
                   version=f"kubegpu-amd {__version__}")
    args = p.parse_args(argv)

    if args.fake:
        from ..discovery import FakeBackend, fixtures

        backend = FakeBackend(fixtures.fixture_8x_mi355x())
    else:
        backend = default_backend()
    if args.cdi:
        from ..deviceplugin.cdi import cdi_json

        info = None
        mgr = create_device_plugin(backend)
        mgr.start()
        if mgr._last_info is None:
            print("no GPUs discovered", file=sys.stderr)
            return 1
        print(cdi_json(mgr._last_info))
        return 0
    if args.health:
        mgr = create_device_plugin(backend)
        mgr.start()
        health = mgr.device_health()
        rows = {}
        for uuid in sorted(health):
            g = mgr.gpu_or_tombstone(uuid)
            rows[uuid] = {
                "healthy": health[uuid],
                "present": uuid in mgr.gpus,
                "in_use": bool(g.in_use) if g else None,
                "ecc_correctable": g.ecc_correctable if g else None,
                "ecc_uncorrectable": g.ecc_uncorrectable if g else None,
                "process_count": g.process_count if g else None,
            }
        print(json.dumps(rows, indent=1))
        return 0
    if not (args.plugin or args.schedule or args.probe):
        print(backend.get_gpu_info().decode())
        return 0

    mgr = create_device_plugin(backend)
    mgr.start()
    ni = NodeInfo(name="local")
    mgr.update_node_info(ni)
    if args.plugin:
        print(json.dumps({
            "capacity": ni.capacity,
            "allocatable": ni.allocatable,
            "kube_cap": ni.kube_cap,
            "kube_alloc": ni.kube_alloc,
        }, indent=1))
        return 0

    k = args.probe or args.schedule
    cluster = Cluster()
    cluster.add_node(ni, mgr._last_info, mgr)
    pod = PodInfo(
        name=f"cli-{k}gpu",
        running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: k})},
    )
    res = cluster.schedule(pod)
    mounts, devices, envs = cluster.container_allocate(pod, "c")
    from ..events import EVENTS

    print(json.dumps({
        "node": res.node_name,
        "gpus": res.uuids,
        "devices": devices,
        "envs": envs,
        "schedule_latency_ms": res.latency_s * 1e3,
        "event": (EVENTS.recent(1) or [None])[-1],
    }, indent=1))

    if args.probe:
        from ..probe import probe_with_link_utilization, run_rccl_probe

        idxs = sorted(mgr.gpus[u].index for u in res.uuids)
        # model prediction for the scheduled subset (the quantity the
        # probe verifies — SURVEY.md hard part (b))
        st = cluster.core.nodes[res.node_name]
        pred = st.scorer.ring_bw(idxs)
        pred = None if pred >= 1e9 else pred
        out, links = probe_with_link_utilization(
            run_rccl_probe, devices=idxs, nbytes=args.bytes
        )
        measured = out.get("busbw_gbps")
        print(json.dumps({
            "probe": out,
            "predicted_ring_bottleneck_gbps": pred,
            "model_vs_measured_ratio": (
                round(measured / pred, 3) if pred and measured else None
            ),
            "xgmi_link_traffic": links,
        }, indent=1))
    return 0


if __name__ == "__main__":
    sys.exit(main())
