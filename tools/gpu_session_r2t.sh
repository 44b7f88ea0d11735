#!/bin/bash
# Round-2 session T: stability batch — 3 consecutive full GPU suites +
# 3 benches + a 240 s agent soak on one box.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_t.log 2>&1
for i in 1 2 3; do
  timeout 300 python -m pytest tests -m gpu -q 2>&1 | tail -1
done > gpurun_out/pytest_gpu_x3.log
for i in 1 2 3; do timeout 200 python bench.py --pods 100 2>/dev/null; done \
  > gpurun_out/bench_x3.jsonl
timeout 300 python tools/agent_soak.py --seconds 240 > gpurun_out/agent_soak_240_t.json 2>/dev/null
echo done
