"""kube-scheduler extender webhook — cross-node xGMI awareness for
STOCK Kubernetes.

Three integration paths now cover every deployment style:

1. KubeDevice-style custom core (`kubegpu_amd.core.Cluster`) — full
   grouped-request translation + binding (the reference's model);
2. kubelet `GetPreferredAllocation` (`server/kubelet_plugin.py`) —
   within-node subset choice for vanilla kubelet;
3. THIS: the scheduler-extender webhook (kube-scheduler's
   `--config` `extenders:` stanza, protocol
   k8s.io/kube-scheduler/extender/v1) — node-level filtering and
   scoring for vanilla kube-scheduler, so a 4-GPU pod lands on the
   node that can give it an intact hive even without the custom core.

Endpoints (JSON over HTTP, stdlib server — no extra deps):

* ``POST /v1/filter``      ExtenderArgs -> ExtenderFilterResult
  (nodes that can satisfy the pod's ``amd.com/gpu`` demand with a
  concrete bind; failures carry a reason per node)
* ``POST /v1/prioritize``  ExtenderArgs -> HostPriorityList
  (score 0..10 by the ring-bottleneck bandwidth the pod would get,
  with anti-fragmentation as tie-break — the same objective the
  custom core optimizes)
* ``POST /v1/nodes/<name>``   register/refresh a node's inventory
  (GpusInfo JSON, e.g. piped from ``amdsmiinfo json`` by the node
  agent); ``DELETE`` removes it
* ``GET /healthz``

Pod GPU demand uses the reference semantics: max(Σ running containers,
max init container) of ``amd.com/gpu`` limits (gpu.go:295-303).
"""

from __future__ import annotations

import json
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Dict, List, Optional, Tuple

from ..api import utils
from ..api.types import ContainerInfo, NodeInfo, PodInfo
from ..core import Cluster
from ..discovery import GpusInfo
from ..plugintypes import RESOURCE_GPU
from ..scheduler.translate import SchedulingError

_NODE_RE = re.compile(r"^/v1/nodes/([^/]+)$")


def pod_gpu_demand(pod_spec: Dict) -> int:
    """max(Σ containers, max initContainers) of amd.com/gpu limits."""

    def req(c: Dict) -> int:
        res = c.get("resources", {}) or {}
        for key in ("limits", "requests"):
            v = (res.get(key) or {}).get(RESOURCE_GPU)
            if v is not None:
                return int(v)
        return 0

    running = sum(req(c) for c in pod_spec.get("containers", []) or [])
    init = max((req(c) for c in pod_spec.get("initContainers", []) or []), default=0)
    return max(running, init)


class ExtenderCore:
    """State + filter/prioritize logic (HTTP-free, unit-testable)."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self.cluster = Cluster()

    # -- node registry -----------------------------------------------------

    def register_node(
        self, name: str, gpus_info: GpusInfo, in_use: List[str] = ()
    ) -> None:
        """Register/refresh a node.

        Occupancy comes WITH the registration: the agent reports which
        uuids hold live allocations (its in_use view, reconciled from
        kubelet's pod-resources API) as a top-level ``in_use`` list —
        the extender itself never observes bindings (kube-scheduler
        does not call back after scheduling), so each refresh carries
        the node's current truth.  Raw process_count is deliberately
        NOT treated as occupancy here: system daemons register on KFD
        (a known-idle box reports process_count 2), so it stays a soft
        idle-preference signal in GetPreferredAllocation only.
        """
        with self._lock:
            ni = NodeInfo(name=name)
            # advertise exactly like the device plugin would
            from ..deviceplugin import create_device_plugin
            from ..discovery import FakeBackend

            mgr = create_device_plugin(FakeBackend(gpus_info))
            mgr.start()
            mgr.update_node_info(ni)
            if name in self.cluster.node_infos:
                self.cluster.remove_node(name)
            self.cluster.add_node(ni, mgr._last_info, mgr)
            state = self.cluster.core.nodes[name]
            for uuid in set(in_use):
                if uuid in state.gpus:
                    state.mark_used(uuid)
            self.cluster.reindex_node(name)

    def remove_node(self, name: str) -> None:
        with self._lock:
            self.cluster.remove_node(name)

    # -- extender verbs ----------------------------------------------------

    def _trial(self, pod_name: str, demand: int, node: str) -> Optional[float]:
        """Ring-bottleneck GB/s the pod would get on *node*, or None.

        Uses the flat (wildcard) translation, NOT the cluster-wide best
        canonical tree: kube-scheduler asks about each node on its own
        merits, and the binder's subset scorer then picks the xGMI-best
        free GPUs of THAT node (the grouped grammar only matters on the
        custom-core path where the core owns cross-node choice)."""
        from ..scheduler.translate import GPU_TOPOLOGY_GENERATION

        pod = PodInfo(
            name=pod_name,
            requests={GPU_TOPOLOGY_GENERATION: 0},  # flat per-node trial
            running_containers={"c": ContainerInfo(kube_requests={RESOURCE_GPU: demand})},
        )
        try:
            ni = self.cluster.node_infos[node]
            self.cluster.scheduler.pod_allocate(ni, pod)
            uuids = self.cluster.core.bind_pod(node, pod, commit=False)
        except (SchedulingError, KeyError):
            return None
        state = self.cluster.core.nodes[node]
        idxs = [state.gpus[u].index for u in uuids]
        bw = state.scorer.ring_bw(idxs) if idxs else 0.0
        return min(bw, 1e9)

    def filter(self, args: Dict) -> Dict:
        pod = args.get("Pod") or {}
        pod_name = ((pod.get("metadata") or {}).get("name")) or "pod"
        demand = pod_gpu_demand(pod.get("spec") or {})
        names = self._candidate_names(args)
        if demand == 0:
            return {"NodeNames": names, "FailedNodes": {}, "Error": ""}
        ok, failed = [], {}
        with self._lock:
            for n in names:
                if n not in self.cluster.node_infos:
                    failed[n] = "node not registered with GPU extender"
                    continue
                if self._trial(pod_name, demand, n) is None:
                    failed[n] = f"cannot bind {demand} x {RESOURCE_GPU}"
                else:
                    ok.append(n)
        return {"NodeNames": ok, "FailedNodes": failed, "Error": ""}

    def prioritize(self, args: Dict) -> List[Dict]:
        pod = args.get("Pod") or {}
        pod_name = ((pod.get("metadata") or {}).get("name")) or "pod"
        demand = pod_gpu_demand(pod.get("spec") or {})
        names = self._candidate_names(args)
        out = []
        with self._lock:
            scores: Dict[str, float] = {}
            for n in names:
                bw = (
                    self._trial(pod_name, demand, n)
                    if demand and n in self.cluster.node_infos
                    else None
                )
                scores[n] = bw if bw is not None else 0.0
            top = max(scores.values(), default=0.0)
            for n in names:
                # kube-scheduler extender scores are 0..10
                score = int(round(10.0 * scores[n] / top)) if top > 0 else 0
                out.append({"Host": n, "Score": score})
        return out

    @staticmethod
    def _candidate_names(args: Dict) -> List[str]:
        if args.get("NodeNames"):
            return list(args["NodeNames"])
        nodes = (args.get("Nodes") or {}).get("Items") or []
        return [((n.get("metadata") or {}).get("name")) for n in nodes]


class _Handler(BaseHTTPRequestHandler):
    core: ExtenderCore  # set by serve()

    def _reply(self, code: int, payload) -> None:
        body = json.dumps(payload).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _body(self) -> Dict:
        n = int(self.headers.get("Content-Length") or 0)
        raw = self.rfile.read(n) if n else b"{}"
        return json.loads(raw.decode() or "{}")

    def do_GET(self):  # noqa: N802  (http.server API)
        if self.path == "/healthz":
            self._reply(200, {"ok": True, "nodes": len(self.core.cluster.node_infos)})
        else:
            self._reply(404, {"error": "not found"})

    def do_POST(self):  # noqa: N802
        try:
            if self.path == "/v1/filter":
                self._reply(200, self.core.filter(self._body()))
            elif self.path == "/v1/prioritize":
                self._reply(200, self.core.prioritize(self._body()))
            else:
                m = _NODE_RE.match(self.path)
                if m:
                    body = self._body()
                    in_use = body.get("in_use") or []
                    info = GpusInfo.from_json(json.dumps(body))
                    self.core.register_node(m.group(1), info, in_use)
                    self._reply(200, {"registered": m.group(1),
                                      "gpus": len(info.devices),
                                      "in_use": len(in_use)})
                else:
                    self._reply(404, {"error": "not found"})
        except Exception as e:  # malformed input must not kill the server
            self._reply(400, {"error": str(e)[:300]})

    def do_DELETE(self):  # noqa: N802
        m = _NODE_RE.match(self.path)
        if m:
            self.core.remove_node(m.group(1))
            self._reply(200, {"removed": m.group(1)})
        else:
            self._reply(404, {"error": "not found"})

    def log_message(self, fmt, *fmt_args):  # quiet; route to our logger
        utils.logf(4, "extender: " + fmt, *fmt_args)


def serve(
    host: str = "0.0.0.0",
    port: int = 9109,
    core: Optional[ExtenderCore] = None,
) -> Tuple[ThreadingHTTPServer, ExtenderCore]:
    """Start the extender HTTP server (returns (server, core); call
    server.shutdown() to stop).  Wire into kube-scheduler with:

        extenders:
        - urlPrefix: http://<host>:9109/v1
          filterVerb: filter
          prioritizeVerb: prioritize
          weight: 5
          managedResources:
          - name: amd.com/gpu
            ignoredByScheduler: false
    """
    core = core or ExtenderCore()
    handler = type("BoundHandler", (_Handler,), {"core": core})
    server = ThreadingHTTPServer((host, port), handler)
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    utils.logf(1, "scheduler extender serving on %s:%d", host, port)
    return server, core


def main(argv=None) -> int:
    import argparse
    import signal

    ap = argparse.ArgumentParser(prog="kubegpu-amd-extender")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9109)
    args = ap.parse_args(argv)
    server, _core = serve(host=args.host, port=args.port)
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    server.shutdown()
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(main())
