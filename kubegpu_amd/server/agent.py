"""Node agent — the long-running production entry point.

Runs the device-plugin gRPC server, registers with kubelet (with retry —
kubelet restarts wipe the plugin registry, so the agent watches its own
socket and re-registers when the plugin dir is recreated), keeps
discovery fresh, and serves Prometheus metrics.

Usage:
    python -m kubegpu_amd.server.agent [--socket PATH] [--plugin-dir DIR]
        [--kubelet-socket PATH] [--no-register] [--metrics-port 9400]
        [--fake]  (fixture backend, for plumbing tests without a GPU)
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import sys
import time

from ..api import utils
from ..deviceplugin import create_device_plugin
from ..discovery import FakeBackend, default_backend, fixtures
from ..metrics import METRICS
from . import dpapi
from .kubelet_plugin import KubeletDevicePlugin


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="kubegpu-amd-agent")
    ap.add_argument("--socket", default=None)
    ap.add_argument("--plugin-dir", default=dpapi.DEVICE_PLUGIN_PATH)
    ap.add_argument("--kubelet-socket", default=dpapi.KUBELET_SOCKET)
    ap.add_argument("--pod-resources-socket", default=None,
                    help="kubelet pod-resources socket for in_use "
                         "reconciliation (default: the standard path "
                         "when it exists)")
    ap.add_argument("--no-register", action="store_true")
    ap.add_argument("--extender-url", default=None,
                    help="scheduler-extender base URL (e.g. "
                         "http://gpu-extender:9109); the agent POSTs its "
                         "inventory to /v1/nodes/<node-name> every health "
                         "tick so the extender scores this node")
    ap.add_argument("--node-name", default=None,
                    help="node name for extender registration "
                         "(default: hostname)")
    ap.add_argument("--metrics-port", type=int, default=9400)
    ap.add_argument("--health-interval", type=float, default=30.0)
    ap.add_argument("--fake", action="store_true",
                    help="use the 8xMI355X fixture backend (no GPU needed)")
    ap.add_argument("--startup-probe", action="store_true",
                    help="run the RCCL all-reduce probe over the node's "
                         "GPUs at start and export the measured busbw "
                         "(the verification loop, BASELINE.json)")
    ap.add_argument("--probe-bytes", type=int, default=256 << 20)
    ap.add_argument("--oneshot", action="store_true",
                    help="start, print state, exit (plumbing check)")
    args = ap.parse_args(argv)

    backend = (
        FakeBackend(fixtures.fixture_8x_mi355x()) if args.fake else default_backend()
    )
    manager = create_device_plugin(backend)
    manager.start()
    utils.logf(0, "agent: discovered %d GPU(s)", len(manager.gpus))

    if args.startup_probe:
        try:
            from ..probe import run_rccl_probe

            idxs = sorted(g.index for g in manager.gpus.values())
            rec = run_rccl_probe(devices=idxs, nbytes=args.probe_bytes,
                                 iters=10, warmup=3)
            METRICS.set_xgmi_gbps(rec.get("busbw_gbps", 0.0))
            utils.logf(
                0, "startup probe: %d GPU(s) busbw %.1f GB/s (check=%s)",
                len(idxs), rec.get("busbw_gbps", 0.0), rec.get("check"),
            )
        except Exception as e:
            utils.errorf("startup probe failed (agent continues): %s", e)

    plugin = KubeletDevicePlugin(
        manager,
        socket_path=args.socket,
        plugin_dir=args.plugin_dir,
    )
    plugin.servicer.health_interval_s = args.health_interval
    plugin.start()

    if args.metrics_port:
        if METRICS.serve(args.metrics_port):
            utils.logf(1, "agent: metrics on :%d", args.metrics_port)

    stop = {"flag": False}

    def _sig(_signo, _frame):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)

    registered = False
    if args.oneshot:
        print(f"agent ok: {len(manager.gpus)} GPUs, socket {plugin.socket_path}")
        plugin.stop()
        return 0

    watcher_started = False
    while not stop["flag"]:
        if not args.no_register and not registered:
            try:
                plugin.register_with_kubelet(args.kubelet_socket)
                registered = True
                if not watcher_started:
                    # kubelet-side restart detection: its socket is
                    # recreated on restart; the watcher re-registers
                    plugin.watch_kubelet(args.kubelet_socket)
                    watcher_started = True
            except Exception as e:
                utils.logf(2, "agent: kubelet registration pending: %s", e)
        # kubelet restart detection: our socket vanishes when the plugin
        # dir is recreated -> re-serve + re-register
        if not os.path.exists(plugin.socket_path):
            utils.logf(0, "agent: socket vanished (kubelet restart?); re-serving")
            plugin.stop()
            plugin.start()
            registered = False
        plugin.servicer.notify()  # wake ListAndWatch to refresh health
        if args.extender_url and manager._last_info is not None:
            try:
                import socket as _socket
                import urllib.request as _ur

                node = args.node_name or _socket.gethostname()
                payload = json.loads(manager._last_info.to_json())
                payload["in_use"] = manager.in_use_uuids()
                req = _ur.Request(
                    f"{args.extender_url.rstrip('/')}/v1/nodes/{node}",
                    data=json.dumps(payload).encode(),
                    headers={"Content-Type": "application/json"},
                    method="POST",
                )
                _ur.urlopen(req, timeout=10).read()
            except Exception as e:
                utils.logf(2, "agent: extender registration failed: %s", e)
        # kubelet is the allocation source of truth on the stock path:
        # reconcile in_use from its pod-resources API when available
        try:
            from .podresources import PodResourcesClient, reconcile_in_use

            client = (PodResourcesClient(args.pod_resources_socket)
                      if args.pod_resources_socket else PodResourcesClient())
            reconcile_in_use(manager, client)
        except Exception as e:  # never let reconcile kill the agent
            utils.logf(2, "agent: pod-resources reconcile error: %s", e)
        for uuid, ok in manager.device_health().items():
            g = manager.gpu_or_tombstone(uuid)
            METRICS.set_gpu_health(
                uuid, ok, g.ecc_uncorrectable if g is not None else 0
            )
            METRICS.set_gpu_in_use(
                uuid, bool(g.in_use) if g is not None else False
            )
        time.sleep(args.health_interval if registered else 5.0)

    plugin.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
