"""Multi-process distributed-path tests (gloo backend, CPU).

The 8-GPU scaling bench is driver-run; these tests keep the distributed
code path (torch.distributed init, bucketed all-reduce probe, MAX-over-
ranks timing) correct by construction on CPU with world_size=2.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    """A fresh ephemeral port per launch: a fixed port flakes with
    EADDRINUSE when a previous torchrun's TCPStore lingers in TIME_WAIT."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_dist(script: str, nproc: int = 2, timeout: int = 240, extra_env=None):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    if extra_env:
        env.update(extra_env)
    return subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={nproc}",
            "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
            script, *(["--gpus", str(nproc), "--steps", "3", "--warmup", "1",
                       "--pods", "50"] if script.endswith("bench.py") else []),
        ],
        capture_output=True,
        timeout=timeout,
        env=env,
        cwd=REPO,
        text=True,
    )


@pytest.mark.timeout(300)
def test_bench_world2_gloo_contract():
    """bench.py over 2 gloo ranks prints one valid JSON contract line."""
    res = _run_dist(os.path.join(REPO, "bench.py"))
    assert res.returncode == 0, res.stderr[-2000:]
    json_lines = [
        l for l in res.stdout.splitlines() if l.startswith("{") and '"metric"' in l
    ]
    assert len(json_lines) == 1  # rank 0 only
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 3
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["dtype"] == "bf16"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    assert "schedule_p50_ms" in rec
    assert rec["config"]["mode"] == "rccl_allreduce"


HELPER = """
import os, sys
sys.path.insert(0, {repo!r})
import torch.distributed as dist
from kubegpu_amd.probe.rccl_probe import torch_allreduce_busbw
dist.init_process_group(backend="gloo")
out = torch_allreduce_busbw(nbytes=1 << 20, iters=4, warmup=1)
if dist.get_rank() == 0:
    assert out["world"] == 2 and out["busbw_gbps"] > 0
    print("PROBE_OK", out["busbw_gbps"])
dist.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_torch_allreduce_probe_world2(tmp_path):
    script = tmp_path / "probe_helper.py"
    script.write_text(HELPER.format(repo=REPO))
    res = _run_dist(str(script))
    assert res.returncode == 0, res.stderr[-2000:]
    assert "PROBE_OK" in res.stdout


@pytest.mark.timeout(180)
def test_bench_single_process_contract():
    """Default (N=1) bench prints one valid contract line on CPU."""
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "2", "--warmup", "1", "--pods", "20"],
        capture_output=True, timeout=150, cwd=REPO, text=True,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    rec = json.loads(res.stdout.strip().splitlines()[-1])
    assert rec["n_gpus"] == 1 and rec["value"] > 0
    assert rec["metric"] == "scheduled_set_allreduce_busbw_GBps"


def test_bench_gpus_flag_without_ranks_errors():
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "4",
         "--steps", "1", "--warmup", "0", "--pods", "5"],
        capture_output=True, timeout=150, cwd=REPO, text=True,
    )
    assert res.returncode == 2
    assert "torch.distributed.run" in res.stderr


@pytest.mark.timeout(300)
def test_bench_world2_pins_ranks_to_scheduled_set():
    """Subset-pinning plumbing: with a fake 8-GPU
    topology, every rank independently schedules the k=2 pod, agrees on
    the same subset, and the record carries the prediction AND the
    per-rank device map — on a real box the same path exports
    ROCR_VISIBLE_DEVICES before HIP init."""
    res = _run_dist(
        os.path.join(REPO, "bench.py"),
        extra_env={"KUBEGPU_BENCH_FAKE_TOPO": "1"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    rec = json.loads(
        [l for l in res.stdout.splitlines() if l.startswith("{") and '"metric"' in l][0]
    )
    cfg = rec["config"]
    assert cfg["scheduled_devices"] is not None
    assert len(cfg["scheduled_devices"]) == 2
    assert cfg["predicted_ring_bottleneck_gbps"] is not None
    assert cfg["rank_pinning"] == "simulated"  # no real devices on CPU
    # both ranks computed the same schedule and took distinct devices,
    # in rank order == sorted scheduled set
    assert cfg["rank_devices"] == cfg["scheduled_devices"]


@pytest.mark.timeout(300)
def test_bench_world4_pinning_distinct_devices():
    """world=4: four ranks, four distinct scheduled devices."""
    res = _run_dist(
        os.path.join(REPO, "bench.py"),
        nproc=4,
        extra_env={"KUBEGPU_BENCH_FAKE_TOPO": "1"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    rec = json.loads(
        [l for l in res.stdout.splitlines() if l.startswith("{") and '"metric"' in l][0]
    )
    cfg = rec["config"]
    assert len(set(cfg["rank_devices"])) == 4
    assert cfg["rank_devices"] == cfg["scheduled_devices"]
