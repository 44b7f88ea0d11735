#!/usr/bin/env python3
"""Agent soak: run the kubelet device-plugin server and hammer it.

Starts KubeletDevicePlugin (real backend by default, --fake for the
fixture), then for --seconds drives a ListAndWatch stream plus a loop of
GetPreferredAllocation + Allocate RPCs from a real gRPC client, and
reports RPC counts, health frames, and RSS growth (leak watch).

Usage: python tools/agent_soak.py [--seconds 120] [--fake]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def rss_mb() -> float:
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return 0.0


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=120)
    ap.add_argument("--fake", action="store_true")
    ap.add_argument("--socket", default="/tmp/amdgpu-soak.sock")
    args = ap.parse_args()

    import grpc

    from kubegpu_amd.deviceplugin import create_device_plugin
    from kubegpu_amd.discovery import FakeBackend, default_backend, fixtures
    from kubegpu_amd.server import KubeletDevicePlugin, dpapi

    backend = (
        FakeBackend(fixtures.fixture_8x_mi355x()) if args.fake else default_backend()
    )
    mgr = create_device_plugin(backend)
    mgr.start()
    plugin = KubeletDevicePlugin(mgr, socket_path=args.socket)
    plugin.servicer.health_interval_s = 2.0
    plugin.start()

    rss0 = rss_mb()
    stop = time.monotonic() + args.seconds
    frames = {"n": 0}

    def watch():
        ch = grpc.insecure_channel(f"unix://{args.socket}")
        stream = ch.unary_stream(
            f"/{dpapi.DEVICE_PLUGIN_SERVICE}/ListAndWatch",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=dpapi.ListAndWatchResponse.FromString,
        )(dpapi.Empty(), timeout=args.seconds + 30)
        try:
            for frame in stream:
                frames["n"] += 1
                if time.monotonic() > stop:
                    break
        except Exception:
            pass

    t = threading.Thread(target=watch, daemon=True)
    t.start()

    ch = grpc.insecure_channel(f"unix://{args.socket}")
    pref = ch.unary_unary(
        f"/{dpapi.DEVICE_PLUGIN_SERVICE}/GetPreferredAllocation",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=dpapi.PreferredAllocationResponse.FromString,
    )
    alloc = ch.unary_unary(
        f"/{dpapi.DEVICE_PLUGIN_SERVICE}/Allocate",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=dpapi.AllocateResponse.FromString,
    )
    ids = sorted(mgr.gpus)
    calls = allocs = errors = 0
    # bounded reservoir: client-side sample storage must not read as a
    # server "leak" in the RSS numbers (both live in this process)
    import random

    reservoir: list = []
    RESERVOIR = 20000
    rng = random.Random(0)
    while time.monotonic() < stop:
        k = [1, 2, 4, min(8, len(ids))][calls % 4]
        t0 = time.perf_counter()
        try:
            resp = pref(
                dpapi.PreferredAllocationRequest(container_requests=[
                    dpapi.ContainerPreferredAllocationRequest(
                        available_deviceIDs=ids, allocation_size=k)
                ]),
                timeout=10,
            )
            chosen = list(resp.container_responses[0].deviceIDs)
            alloc(
                dpapi.AllocateRequest(container_requests=[
                    dpapi.ContainerAllocateRequest(devicesIDs=chosen)
                ]),
                timeout=10,
            )
            allocs += 1
        except Exception:
            errors += 1
        dt = time.perf_counter() - t0
        if len(reservoir) < RESERVOIR:
            reservoir.append(dt)
        else:
            j = rng.randrange(calls + 1)
            if j < RESERVOIR:
                reservoir[j] = dt
        calls += 1
    ch.close()
    plugin.stop()
    lat = sorted(reservoir)
    out = {
        "duration_s": args.seconds,
        "backend": "fake" if args.fake else "real",
        "frames": frames["n"],
        "preferred_plus_allocate_pairs": calls,
        "alloc_ok": allocs,
        "errors": errors,
        "latency_sample": len(lat),
        "rpc_pair_p50_ms": round(lat[len(lat) // 2] * 1e3, 3) if lat else None,
        "rpc_pair_p99_ms": round(lat[int(len(lat) * 0.99)] * 1e3, 3) if lat else None,
        "rss_start_mb": round(rss0, 1),
        "rss_end_mb": round(rss_mb(), 1),
        "rss_growth_mb": round(rss_mb() - rss0, 1),
    }
    print(json.dumps(out, indent=1))
    return 0 if not errors else 1


if __name__ == "__main__":
    sys.exit(main())
