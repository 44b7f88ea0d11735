#!/bin/bash
# Round-2 GPU session K: does pinning the perf level stabilize the bench
# at the top DPM plateau?  (Operator-documented knob, not a trick: the
# kernel does identical work; we only stop the clock governor dithering.)
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_k.log 2>&1

echo "== baseline clocks ==" > gpurun_out/perflevel_session.log
rocm-smi --showgpuclocks >> gpurun_out/perflevel_session.log 2>&1 || true

for i in 1 2 3; do timeout 200 python bench.py --pods 100 2>/dev/null; done \
  > gpurun_out/bench_perfauto.jsonl

rocm-smi --setperflevel high >> gpurun_out/perflevel_session.log 2>&1 || \
  amd-smi set -g 0 --perf-level HIGH >> gpurun_out/perflevel_session.log 2>&1 || true
rocm-smi --showgpuclocks >> gpurun_out/perflevel_session.log 2>&1 || true

for i in 1 2 3 4 5; do timeout 200 python bench.py --pods 100 2>/dev/null; done \
  > gpurun_out/bench_perfhigh.jsonl

rocm-smi --setperflevel auto >> gpurun_out/perflevel_session.log 2>&1 || true

python - > gpurun_out/perflevel_summary.json 2>&1 <<'PY'
import json
def vals(p):
    out = []
    for line in open(p):
        line = line.strip()
        if line.startswith('{') and '"metric"' in line:
            out.append(json.loads(line)["value"])
    return sorted(out)
a, h = vals('gpurun_out/bench_perfauto.jsonl'), vals('gpurun_out/bench_perfhigh.jsonl')
print(json.dumps({"auto": a, "high": h,
                  "auto_median": a[len(a)//2] if a else None,
                  "high_median": h[len(h)//2] if h else None}, indent=1))
PY
echo done
