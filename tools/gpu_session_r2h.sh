#!/bin/bash
# Round-2 GPU session H: self-evidencing utilization — run the bench
# with a ~2 s timed region while sampling GPU busy % concurrently.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_h.log 2>&1

( for i in $(seq 1 40); do
    rocm-smi --showuse --json 2>/dev/null | python -c "
import json,sys,time
try:
    d=json.load(sys.stdin)
    for k,v in d.items():
        print(time.time(), v.get('GPU use (%)'))
except Exception: pass"
    sleep 0.25
  done ) > gpurun_out/busy_series.txt &
SAMPLER=$!
timeout 300 python bench.py --steps 5000 --warmup 50 > gpurun_out/bench_long.json 2> gpurun_out/bench_long.err
wait $SAMPLER
python - > gpurun_out/bench_busy_evidence.json 2>&1 <<'PY'
import json
rec = json.loads(open('gpurun_out/bench_long.json').read().strip().splitlines()[-1])
series = []
for line in open('gpurun_out/busy_series.txt'):
    parts = line.split()
    if len(parts) == 2 and parts[1].isdigit():
        series.append(int(parts[1]))
busy = [s for s in series if s > 50]
print(json.dumps({
    "bench_value_gbps": rec["value"],
    "ms_per_step": rec["ms_per_step"],
    "timed_region_s": rec["ms_per_step"] * rec["steps"] / 1e3,
    "busy_samples": len(series),
    "samples_over_50pct": len(busy),
    "max_busy_pct": max(series) if series else None,
    "series": series,
}, indent=1))
PY
echo done
