#!/bin/bash
# Round-2 GPU session D: workgroup-SIZE sweep for the copy kernel (an
# axis round 1 never swept — blockDim is a runtime launch dim) + another
# default-bench variance point.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

python -m kubegpu_amd.build_native > gpurun_out/build_d.log 2>&1

timeout 600 python - > gpurun_out/copy_threads_sweep.json 2>&1 <<'PY'
import json
from kubegpu_amd.probe.bandwidth import load_ext
ext = load_ext(required=True)
GIB = 1 << 30
rows = []
# grid-stride NT (variant 0 nontemporal) across (threads, blocks);
# round-1 best: threads=256, blocks=1024 -> ~5.9 TB/s
for threads in (256, 512, 1024):
    for blocks in (256, 384, 512, 640, 768, 1024, 1280, 2048):
        if threads * blocks > (1 << 22):  # cap total threads sanely
            continue
        bw = ext.copy_bw_gbps(GIB, 20, blocks, True, 0, threads)
        rows.append({"threads": threads, "blocks": blocks,
                     "gbps": round(bw, 1)})
rows.sort(key=lambda r: -r["gbps"])
print(json.dumps({"buffer": "1 GiB", "iters": 20, "kernel": "copy_kernel_v4_nt",
                  "top": rows[:10], "all": sorted(rows, key=lambda r: (r["threads"], r["blocks"]))},
                 indent=1))
PY

# chunked variant at the best few shapes (DRAM-page locality interacts
# with wave width)
timeout 300 python - > gpurun_out/copy_threads_sweep_chunk.json 2>&1 <<'PY'
import json
from kubegpu_amd.probe.bandwidth import load_ext
ext = load_ext(required=True)
GIB = 1 << 30
rows = []
for threads in (256, 512, 1024):
    for blocks in (512, 1024, 2048):
        bw = ext.copy_bw_gbps(GIB, 20, blocks, True, 2, threads)
        rows.append({"threads": threads, "blocks": blocks, "gbps": round(bw, 1)})
rows.sort(key=lambda r: -r["gbps"])
print(json.dumps({"variant": "chunk", "rows": rows}, indent=1))
PY

timeout 300 python bench.py > gpurun_out/bench_default_d.json 2> gpurun_out/bench_default_d.err
echo done
