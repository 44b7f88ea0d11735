#!/bin/bash
# Round-2 GPU session G: agent endurance on hardware with round-2 code
# (strict must_include, idle preference, watch_kubelet wired) + another
# bench envelope point.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
python -m kubegpu_amd.build_native > gpurun_out/build_g.log 2>&1
timeout 240 python tools/agent_soak.py --seconds 150 > gpurun_out/agent_soak_r2.json 2> gpurun_out/agent_soak_r2.err
timeout 300 python bench.py > gpurun_out/bench_default_g.json 2> gpurun_out/bench_default_g.err
echo done
