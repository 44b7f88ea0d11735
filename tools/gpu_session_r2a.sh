#!/bin/bash
# Round-2 GPU session A: validate the round-2 changes on a real MI355X
# and collect the VERDICT #8 measurements.  Everything lands in
# gpurun_out/ (merged back); key artifacts are then copied to profiles/.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 0) rebuild natives on the box (mtime-skip makes this cheap)
python -m kubegpu_amd.build_native > gpurun_out/build.log 2>&1

# 1) GPU test suite
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu.log

# 2) default bench (new pinned + honest record)
timeout 300 python bench.py > gpurun_out/bench_default_r2.json 2> gpurun_out/bench_default_r2.err

# 3) amdsmiinfo version block (runtime-queried rocm/amdsmi versions)
kubegpu_amd/csrc/bin/amdsmiinfo json 2>/dev/null | head -2 > gpurun_out/amdsmiinfo_version.txt
# partition mode fields for the CPX story
python - > gpurun_out/partition_modes.json 2>&1 <<'EOF'
import json, subprocess
out = subprocess.run(["kubegpu_amd/csrc/bin/amdsmiinfo", "json"],
                     capture_output=True, timeout=120)
d = json.loads(out.stdout)
print(json.dumps({
    "version": d.get("version"),
    "partitions": [
        {"uuid": g["uuid"], "compute_partition": g.get("compute_partition"),
         "memory_partition": g.get("memory_partition"),
         "compute_units": g.get("compute_units")}
        for g in d.get("devices", [])
    ],
}, indent=1))
EOF

# 4) rcclprobe size sweep (k=1 degenerate; informs the k>=2 probe size)
python - > gpurun_out/rcclprobe_size_sweep_r2.json 2> gpurun_out/rcclprobe_sweep.err <<'EOF'
import json
from kubegpu_amd.probe.rccl_probe import run_rccl_probe
rows = []
for mb in (16, 64, 256, 1024, 2048, 4096):
    try:
        r = run_rccl_probe(ndev=1, nbytes=mb << 20, iters=10, warmup=3,
                           timeout_s=240)
        rows.append({"mb": mb, "algbw_gbps": r.get("algbw_gbps"),
                     "busbw_gbps": r.get("busbw_gbps"),
                     "time_ms_per_iter": r.get("time_ms_per_iter")})
    except Exception as e:
        rows.append({"mb": mb, "error": str(e)[:200]})
print(json.dumps({"note": "k=1 self all-reduce (degenerate); larger is "
                  "closer to the k>=2 saturating size", "rows": rows}, indent=1))
EOF

# 5) choose(64,8) timing on a quiet box (native heuristic chooser)
python - > gpurun_out/choose64_timing.json 2>&1 <<'EOF'
import json, time
from kubegpu_amd.scheduler.xgmi import _sym_bw, choose_best_subset_fast
bw = {}
for a in range(64):
    bw[a] = {}
    for b in range(64):
        if a != b:
            bw[a][b] = 153.6 if a // 8 == b // 8 else 64.0
times = []
for _ in range(20):
    t0 = time.perf_counter()
    picked = choose_best_subset_fast(list(range(64)), 8, bw)
    times.append((time.perf_counter() - t0) * 1e3)
times.sort()
print(json.dumps({
    "picked": picked, "one_oam": len({g // 8 for g in picked}) == 1,
    "p50_ms": round(times[10], 3), "max_ms": round(times[-1], 3),
    "runs": 20, "n": 64, "k": 8,
    "native": True,
}, indent=1))
EOF

# 6) smoke
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke.log 2>&1
echo done
