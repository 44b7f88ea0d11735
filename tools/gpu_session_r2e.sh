#!/bin/bash
# Round-2 GPU session E: head-to-head confirmation of the workgroup-size
# finding (alternating, 50 iters, one box) + read/write threads check.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_e.log 2>&1

timeout 600 python - > gpurun_out/copy_threads_headtohead.json 2>&1 <<'PY'
import json
from kubegpu_amd.probe.bandwidth import load_ext
ext = load_ext(required=True)
GIB = 1 << 30
shapes = [(256, 1024), (512, 1024), (1024, 768), (1024, 1024), (1024, 1280)]
rounds = []
for rep in range(3):  # alternate to cancel drift
    for threads, blocks in shapes:
        bw = ext.copy_bw_gbps(GIB, 50, blocks, True, 0, threads)
        rounds.append({"rep": rep, "threads": threads, "blocks": blocks,
                       "gbps": round(bw, 1)})
best = {}
for r in rounds:
    k = (r["threads"], r["blocks"])
    best.setdefault(k, []).append(r["gbps"])
summary = sorted(
    ({"threads": k[0], "blocks": k[1],
      "median_gbps": sorted(v)[len(v)//2], "runs": v} for k, v in best.items()),
    key=lambda r: -r["median_gbps"])
print(json.dumps({"buffer": "1 GiB", "iters": 50, "summary": summary}, indent=1))
PY

timeout 300 python - > gpurun_out/rw_threads_check.json 2>&1 <<'PY'
import json
from kubegpu_amd.probe.bandwidth import load_ext
ext = load_ext(required=True)
GIB = 1 << 30
out = {"read": [], "write": []}
for blocks in (640, 1024):
    out["read"].append({"blocks": blocks,
                        "gbps": round(ext.read_bw_gbps(GIB, 20, blocks), 1)})
    out["write"].append({"blocks": blocks,
                         "gbps": round(ext.write_bw_gbps(GIB, 20, blocks), 1)})
print(json.dumps(out, indent=1))
PY

timeout 300 python bench.py > gpurun_out/bench_default_e.json 2> gpurun_out/bench_default_e.err
echo done
