"""RCCL all-reduce probe wrappers.

Two paths to the same measurement:

* ``run_rccl_probe`` — exec the native single-process ``rcclprobe``
  binary (csrc/rcclprobe.cpp, librccl over xGMI) inside the pod's GPU
  set.  This is the in-pod verification probe named by BASELINE.json.
* ``torch_allreduce_busbw`` — in-process bucketed all-reduce through an
  already-initialized torch.distributed process group (backend "nccl"
  IS RCCL on ROCm); used by bench.py's one-rank-per-GPU mode.
"""

from __future__ import annotations

import json
import os
import subprocess
import time
from typing import Dict, List, Optional


def _default_binary() -> str:
    env = os.environ.get("KUBEGPU_RCCLPROBE")
    if env:
        return env
    here = os.path.dirname(os.path.abspath(__file__))
    return os.path.normpath(os.path.join(here, "..", "csrc", "bin", "rcclprobe"))


def run_rccl_probe(
    ndev: Optional[int] = None,
    devices: Optional[List[int]] = None,
    nbytes: int = 256 << 20,
    iters: int = 20,
    warmup: int = 5,
    binary: Optional[str] = None,
    timeout_s: float = 300.0,
) -> Dict:
    """Run the native probe; returns its JSON record (busbw_gbps etc.)."""
    binary = binary or _default_binary()
    if not os.path.exists(binary):
        raise FileNotFoundError(
            f"rcclprobe binary not built at {binary}; run python -m kubegpu_amd.build_native"
        )
    cmd = [binary, "--bytes", str(nbytes), "--iters", str(iters), "--warmup", str(warmup)]
    if devices is not None:
        cmd += ["--devices", ",".join(str(d) for d in devices)]
    elif ndev is not None:
        cmd += ["--ndev", str(ndev)]
    out = subprocess.run(cmd, capture_output=True, timeout=timeout_s, check=True)
    return json.loads(out.stdout.decode().strip().splitlines()[-1])


def torch_allreduce_busbw(
    nbytes: int = 256 << 20,
    iters: int = 20,
    warmup: int = 5,
    device=None,
) -> Dict:
    """Bucketed all-reduce bandwidth through torch.distributed.

    Requires an initialized process group; each rank calls this
    collectively.  Returns {'busbw_gbps', 'algbw_gbps',
    'time_ms_per_iter'} computed from the MAX per-iteration time across
    ranks (the job is as slow as its slowest rank).
    """
    import torch
    import torch.distributed as dist

    world = dist.get_world_size()
    on_gpu = torch.cuda.is_available()
    if device is None:
        device = torch.device("cuda", torch.cuda.current_device()) if on_gpu else torch.device("cpu")
    count = nbytes // 2
    buf = torch.ones(count, dtype=torch.bfloat16, device=device)

    def _sync():
        if on_gpu:
            torch.cuda.synchronize(device)

    for _ in range(warmup):
        dist.all_reduce(buf)
    dist.barrier()
    _sync()
    t0 = time.perf_counter()
    for _ in range(iters):
        dist.all_reduce(buf)
    _sync()
    dist.barrier()
    _sync()  # the barrier itself is a device op under NCCL
    t1 = time.perf_counter()

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if on_gpu:
        elapsed = elapsed.to(device)
    dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
    sec = float(elapsed.item()) / iters
    algbw = nbytes / sec / 1e9
    factor = 2.0 * (world - 1) / world if world > 1 else 1.0
    return {
        "busbw_gbps": algbw * factor,
        "algbw_gbps": algbw,
        "time_ms_per_iter": sec * 1e3,
        "world": world,
        "bytes": nbytes,
    }
