#!/bin/bash
# Round-2 GPU session I: mid-round validation — full gpu suite, smoke,
# bench, and the flattened fleet-scale curve out to 16,384 nodes.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_i.log 2>&1
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_i.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu_i.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke_i.log 2>&1
timeout 300 python bench.py > gpurun_out/bench_default_i.json 2> gpurun_out/bench_default_i.err
timeout 900 python - > gpurun_out/schedule_scale_curve2.json 2>&1 <<'PY'
import json, time
from kubegpu_amd.api.types import ContainerInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU

out = {"workload": "mixed 1/2/4/8-GPU stream, 500 pods, 256 resident",
       "points": []}
for n_nodes in (256, 1024, 4096, 16384):
    cluster = Cluster()
    for n in range(n_nodes):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"node{n:05d}", mgr)
    lat, live = [], []
    for i in range(500):
        pod = PodInfo(name=f"p{i}", running_containers={
            "c": ContainerInfo(kube_requests={RESOURCE_GPU: [1, 2, 4, 8][i % 4]})})
        t0 = time.perf_counter()
        try:
            cluster.schedule(pod)
            live.append(pod)
        except Exception:
            pass
        lat.append(time.perf_counter() - t0)
        while len(live) > 256:
            cluster.release(live.pop(0))
    lat.sort()
    out["points"].append({"nodes": n_nodes,
                          "p50_ms": round(lat[250] * 1e3, 3),
                          "p95_ms": round(lat[475] * 1e3, 3)})
    print(json.dumps(out["points"][-1]))
print(json.dumps(out, indent=1))
PY
echo done
