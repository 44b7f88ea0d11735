#!/bin/bash
# Round-2 GPU session B: BDF cross-check on hardware + rocprof evidence
# of the round-2 bench + in_use surfacing on a real node.
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out

# 1) bench with pinning + BDF verification (the new record keys)
timeout 300 python bench.py --steps 200 --pods 500 \
  > gpurun_out/bench_bdfcheck.json 2> gpurun_out/bench_bdfcheck.err

# 2) rocprof kernel stats of the timed copy kernel (fresh round-2 evidence)
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof_r2" -o bench_r2 \
  -- python "$REPO/bench.py" --steps 50 --pods 50 \
  > "$REPO/gpurun_out/rocprof_bench_r2.log" 2>&1
cd "$REPO"

# 3) amddevs --health on the real node (shows the new in_use field)
timeout 120 python -m kubegpu_amd.cli.amddevs --health \
  > gpurun_out/amddevs_health_r2.json 2>&1

echo done
