#!/bin/bash
# Round-2 GPU session J: within-box repeatability (10x default bench,
# one box) + 420 s agent endurance.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_j.log 2>&1

for i in $(seq 1 10); do
  timeout 200 python bench.py --pods 200 2>/dev/null
done > gpurun_out/bench_repeat10.jsonl

python - > gpurun_out/bench_repeat10_summary.json 2>&1 <<'PY'
import json
vals = []
for line in open('gpurun_out/bench_repeat10.jsonl'):
    line = line.strip()
    if line.startswith('{') and '"metric"' in line:
        vals.append(json.loads(line)["value"])
vals.sort()
print(json.dumps({
    "runs": len(vals), "min": vals[0], "max": vals[-1],
    "median": vals[len(vals)//2],
    "spread_pct": round(100*(vals[-1]-vals[0])/vals[len(vals)//2], 2),
    "values": vals}, indent=1))
PY

timeout 600 python tools/agent_soak.py --seconds 420 > gpurun_out/agent_soak_420.json 2> gpurun_out/agent_soak_420.err
echo done
