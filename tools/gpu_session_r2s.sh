#!/bin/bash
# Round-2 session S: 300 s sustained copy load, value sampled per ~10 s
# window — plateau/thermal dynamics behind the within-box spread.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_s.log 2>&1
timeout 420 python - > gpurun_out/thermal_marathon_r2.json 2>&1 <<'PY'
import json, time
from kubegpu_amd.probe.bandwidth import load_ext
ext = load_ext(required=True)
GIB = 1 << 30
rows = []
t_end = time.monotonic() + 300
while time.monotonic() < t_end:
    bw = ext.copy_bw_gbps(GIB, 25)  # ~10 ms timed per call
    rows.append({"t_s": round(300 - (t_end - time.monotonic()), 1),
                 "gbps": round(bw, 1)})
    time.sleep(9.0)
vals = sorted(r["gbps"] for r in rows)
print(json.dumps({
    "duration_s": 300, "samples": len(rows),
    "min": vals[0], "max": vals[-1], "median": vals[len(vals)//2],
    "series": rows}, indent=1))
PY
timeout 200 python bench.py --pods 200 > gpurun_out/bench_default_s.json 2>/dev/null
echo done
