"""kube-scheduler extender webhook tests (extender/v1 protocol over
real HTTP): node-level xGMI awareness for vanilla kube-scheduler."""

import json
import urllib.request

import pytest

from kubegpu_amd.discovery import fixtures
from kubegpu_amd.server.extender import ExtenderCore, pod_gpu_demand, serve


def _pod(n, init=0):
    spec = {"containers": [
        {"name": "c", "resources": {"limits": {"amd.com/gpu": str(n)}}}
    ]}
    if init:
        spec["initContainers"] = [
            {"name": "i", "resources": {"limits": {"amd.com/gpu": str(init)}}}
        ]
    return {"metadata": {"name": "p"}, "spec": spec}


def test_pod_gpu_demand_reference_semantics():
    """Demand = max(Σ containers, max init) — gpu.go:295-303."""
    assert pod_gpu_demand(_pod(4)["spec"]) == 4
    assert pod_gpu_demand(_pod(4, init=8)["spec"]) == 8
    assert pod_gpu_demand(_pod(8, init=8)["spec"]) == 8
    assert pod_gpu_demand({"containers": [{"name": "c"}]}) == 0


def test_filter_and_prioritize_core():
    core = ExtenderCore()
    core.register_node("full", fixtures.fixture_8x_mi355x())
    core.register_node("twohive", fixtures.fixture_2hive_8gpu())
    core.register_node("small", fixtures.fixture_4x_no_xgmi())

    args = {"Pod": _pod(8), "NodeNames": ["full", "twohive", "small", "ghost"]}
    res = core.filter(args)
    assert set(res["NodeNames"]) == {"full", "twohive"}
    assert "small" in res["FailedNodes"] and "ghost" in res["FailedNodes"]

    # prioritize: the full mesh gives an 8-GPU pod a better ring than
    # the 2-hive node (which must cross the PCIe bridge)
    pri = {p["Host"]: p["Score"] for p in core.prioritize(args)}
    assert pri["full"] == 10
    assert 0 <= pri["twohive"] < pri["full"]
    assert pri["small"] == 0  # cannot fit at all

    # zero-GPU pod passes every node through untouched
    res0 = core.filter({"Pod": _pod(0), "NodeNames": ["full", "ghost"]})
    assert res0["NodeNames"] == ["full", "ghost"]


def test_filter_respects_live_allocations():
    """A node whose hive is partly taken loses to an idle one."""
    from kubegpu_amd.api.types import ContainerInfo, PodInfo
    from kubegpu_amd.plugintypes import RESOURCE_GPU

    core = ExtenderCore()
    core.register_node("busy", fixtures.fixture_8x_mi355x())
    core.register_node("idle", fixtures.fixture_8x_mi355x())
    pod = PodInfo(name="taker", running_containers={
        "c": ContainerInfo(kube_requests={RESOURCE_GPU: 6})})
    # consume 6 GPUs of "busy" through the same cluster the extender uses
    core.cluster.core.bind_pod("busy", _translated(core, pod), commit=True)

    args = {"Pod": _pod(4), "NodeNames": ["busy", "idle"]}
    res = core.filter(args)
    assert res["NodeNames"] == ["idle"]
    assert "busy" in res["FailedNodes"]


def _translated(core, pod):
    ni = core.cluster.node_infos["busy"]
    core.cluster.scheduler.pod_allocate(ni, pod)
    return pod


def test_extender_http_roundtrip():
    """Full HTTP protocol: register nodes via POST /v1/nodes/<n>, then
    filter + prioritize + healthz over the wire."""
    server, core = serve(host="127.0.0.1", port=0)
    port = server.server_address[1]
    base = f"http://127.0.0.1:{port}"

    def post(path, payload):
        req = urllib.request.Request(
            base + path, data=json.dumps(payload).encode(),
            headers={"Content-Type": "application/json"}, method="POST")
        with urllib.request.urlopen(req, timeout=10) as r:
            return json.loads(r.read().decode())

    try:
        fix = json.loads(fixtures.fixture_2hive_8gpu().to_json())
        out = post("/v1/nodes/n0", fix)
        assert out == {"registered": "n0", "gpus": 8, "in_use": 0}

        res = post("/v1/filter", {"Pod": _pod(4), "NodeNames": ["n0", "nope"]})
        assert res["NodeNames"] == ["n0"]
        assert "nope" in res["FailedNodes"]

        pri = post("/v1/prioritize", {"Pod": _pod(4), "NodeNames": ["n0"]})
        assert pri == [{"Host": "n0", "Score": 10}]

        with urllib.request.urlopen(base + "/healthz", timeout=10) as r:
            h = json.loads(r.read().decode())
        assert h["ok"] and h["nodes"] == 1

        # malformed body -> 400, server stays up
        req = urllib.request.Request(
            base + "/v1/nodes/bad", data=b"{not json",
            headers={"Content-Type": "application/json"}, method="POST")
        with pytest.raises(urllib.error.HTTPError) as err:
            urllib.request.urlopen(req, timeout=10)
        assert err.value.code == 400

        # DELETE removes the node
        req = urllib.request.Request(base + "/v1/nodes/n0", method="DELETE")
        with urllib.request.urlopen(req, timeout=10) as r:
            assert json.loads(r.read().decode()) == {"removed": "n0"}
        res = post("/v1/filter", {"Pod": _pod(4), "NodeNames": ["n0"]})
        assert res["NodeNames"] == []
    finally:
        server.shutdown()


@pytest.mark.timeout(120)
def test_agent_feeds_extender(tmp_path):
    """Deployment loop end to end: the node agent POSTs its inventory to
    the extender every health tick; the extender then scores the node
    for kube-scheduler."""
    import signal
    import subprocess
    import sys
    import time as _time
    import os

    server, core = serve(host="127.0.0.1", port=0)
    port = server.server_address[1]
    try:
        proc = subprocess.Popen(
            [sys.executable, "-m", "kubegpu_amd.server.agent",
             "--fake", "--no-register",
             "--socket", str(tmp_path / "agent.sock"),
             "--metrics-port", "0",
             "--health-interval", "0.2",
             "--extender-url", f"http://127.0.0.1:{port}",
             "--node-name", "fed-node"],
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        deadline = _time.time() + 20
        while "fed-node" not in core.cluster.node_infos and _time.time() < deadline:
            _time.sleep(0.1)
        assert "fed-node" in core.cluster.node_infos, "agent never registered"
        res = core.filter({"Pod": _pod(4), "NodeNames": ["fed-node"]})
        assert res["NodeNames"] == ["fed-node"]
        proc.send_signal(signal.SIGTERM)
        out, _ = proc.communicate(timeout=30)
        assert proc.returncode == 0, out[-1000:]
    finally:
        server.shutdown()
        if proc.poll() is None:
            proc.kill()


def test_registration_carries_occupancy():
    """The agent's in_use list (reconciled from kubelet pod-resources)
    and live process counts flow into the extender's occupancy: an
    8-GPU pod no longer fits a node with 2 GPUs held, and a 4-GPU pod
    avoids the held hive; a later refresh with everything free clears
    the state."""
    core = ExtenderCore()
    fix = fixtures.fixture_2hive_8gpu()
    core.register_node("n0", fix,
                       in_use=["GPU-mi355x-00", "GPU-mi355x-01"])

    res = core.filter({"Pod": _pod(8), "NodeNames": ["n0"]})
    assert res["NodeNames"] == []  # only 6 free

    res = core.filter({"Pod": _pod(4), "NodeNames": ["n0"]})
    assert res["NodeNames"] == ["n0"]
    # the chosen subset for 4 must be the intact hive 1
    bw = core._trial("p", 4, "n0")
    assert bw and bw > 100  # xGMI-class, not PCIe-bound

    # raw process_count is NOT occupancy (system daemons register on
    # KFD; a known-idle box reports 2) — only the agent's in_use list is
    fix2 = fixtures.fixture_2hive_8gpu()
    fix2.devices[4].process_count = 3
    core.register_node("n0", fix2)
    res = core.filter({"Pod": _pod(8), "NodeNames": ["n0"]})
    assert res["NodeNames"] == ["n0"]

    # refresh with a clean inventory frees everything
    core.register_node("n0", fixtures.fixture_2hive_8gpu())
    res = core.filter({"Pod": _pod(8), "NodeNames": ["n0"]})
    assert res["NodeNames"] == ["n0"]
