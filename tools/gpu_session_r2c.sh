#!/bin/bash
# Round-2 GPU session C: full gpu suite (incl. the 2 new record-contract
# tests), fleet-scale schedule curve on box hardware, bench variance point.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

python -m kubegpu_amd.build_native > gpurun_out/build_c.log 2>&1

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_c.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu_c.log

timeout 300 python bench.py > gpurun_out/bench_default_c.json 2> gpurun_out/bench_default_c.err

# fleet-scale schedule latency on the box CPU (the p50 headline metric)
timeout 600 python - > gpurun_out/schedule_scale_curve.json 2>&1 <<'EOF'
import json, time
from kubegpu_amd.api.types import ContainerInfo, PodInfo
from kubegpu_amd.core import Cluster
from kubegpu_amd.deviceplugin import create_device_plugin
from kubegpu_amd.discovery import FakeBackend, fixtures
from kubegpu_amd.plugintypes import RESOURCE_GPU

out = {"workload": "mixed 1/2/4/8-GPU stream, 500 pods, 64 resident",
       "points": []}
for n_nodes in (256, 1024, 4096):
    cluster = Cluster()
    for n in range(n_nodes):
        mgr = create_device_plugin(FakeBackend(fixtures.fixture_8x_mi355x()))
        cluster.add_node_from_manager(f"node{n:04d}", mgr)
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
        while len(live) > 64:
            cluster.release(live.pop(0))
    lat.sort()
    out["points"].append({"nodes": n_nodes,
                          "p50_ms": round(lat[250] * 1e3, 3),
                          "p95_ms": round(lat[475] * 1e3, 3)})
print(json.dumps(out, indent=1))
EOF

echo done
