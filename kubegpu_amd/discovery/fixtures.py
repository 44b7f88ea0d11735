"""Synthetic inventory fixtures for the fake backend and tests.

Mirrors the reference's fixture strategy (two captured JSON payloads:
an 8-GPU 2-level topology node and a degenerate "Topology":null node,
nvidia_gpu_manager_test.go:16-17) with MI355X-shaped data:

* ``fixture_8x_mi355x()``   — one full 8-GPU xGMI hive (7 direct links per
  GPU at ≈153 GB/s): the production MI355X node shape.
* ``fixture_2hive_8gpu()``  — two 4-GPU hives bridged by PCIe: exercises a
  genuinely 2-level tree (gpugrp0 = hive, gpugrp1 = node).
* ``fixture_4x_no_xgmi()``  — degenerate: no xGMI at all (each GPU its own
  group; the analog of the reference's "Topology":null K80 node).
* ``fixture_degraded_mesh()`` — full hive with selected links DOWN
  (pairs fall back to the host path).
* ``fixture_cpx_2oam_8part()`` — CPX partitioning: 2 OAMs × 4 compute
  partitions, INTERNAL links inside an OAM, xGMI across.
"""

from __future__ import annotations

from typing import List, Optional, Set, Tuple

from .types import (
    LINK_PCIE,
    LINK_XGMI,
    PCIE_GBPS_DEFAULT,
    XGMI_LINK_GBPS_DEFAULT,
    GpuInfo,
    GpusInfo,
    LinkInfo,
    MemoryInfo,
    VersionInfo,
)

MI355X_MODEL = "AMD Instinct MI355X"
MI355X_DEVICE_ID = "0x75a0"
MI355X_GFX = "gfx950"
MI355X_VRAM_BYTES = 288 * 1024**3  # 288 GiB HBM3E
MI355X_VRAM_BW_GBPS = 8000.0
MI355X_CUS = 256


def _mk_gpu(index: int, numa: int, uuid: Optional[str] = None) -> GpuInfo:
    return GpuInfo(
        uuid=uuid or f"GPU-mi355x-{index:02d}",
        model=MI355X_MODEL,
        device_id=MI355X_DEVICE_ID,
        gfx_target=MI355X_GFX,
        index=index,
        bdf=f"0000:{0x10 + index:02x}:00.0",
        render_path=f"/dev/dri/renderD{128 + index}",
        card_path=f"/dev/dri/card{index}",
        numa_node=numa,
        compute_units=MI355X_CUS,
        memory=MemoryInfo(
            vram_total_bytes=MI355X_VRAM_BYTES,
            vram_type="HBM3E",
            vram_bandwidth_gbps=MI355X_VRAM_BW_GBPS,
        ),
    )


def _connect(
    gpus: List[GpuInfo],
    xgmi_pairs: Set[Tuple[int, int]],
    xgmi_gbps: float = XGMI_LINK_GBPS_DEFAULT,
) -> None:
    """Wire every pair: direct xGMI for pairs listed, PCIe otherwise."""
    for a in gpus:
        for b in gpus:
            if a.index == b.index:
                continue
            pair = (min(a.index, b.index), max(a.index, b.index))
            if pair in xgmi_pairs:
                a.links.append(
                    LinkInfo(
                        peer_index=b.index,
                        type=LINK_XGMI,
                        hops=1,
                        weight=15,
                        bandwidth_gbps=xgmi_gbps,
                        p2p=True,
                    )
                )
            else:
                a.links.append(
                    LinkInfo(
                        peer_index=b.index,
                        type=LINK_PCIE,
                        hops=2,
                        weight=40,
                        bandwidth_gbps=PCIE_GBPS_DEFAULT,
                        p2p=False,
                    )
                )


def _version() -> VersionInfo:
    return VersionInfo(driver="6.14.14", rocm="7.2.0", amdsmi="26.2.1")


def fixture_8x_mi355x() -> GpusInfo:
    """Full 8-GPU xGMI hive: all 28 pairs direct (7 links per GPU)."""
    gpus = [_mk_gpu(i, numa=i // 4) for i in range(8)]
    pairs = {(i, j) for i in range(8) for j in range(i + 1, 8)}
    _connect(gpus, pairs)
    return GpusInfo(version=_version(), devices=gpus)


def fixture_2hive_8gpu() -> GpusInfo:
    """Two 4-GPU hives (0-3 and 4-7), full mesh inside, PCIe across."""
    gpus = [_mk_gpu(i, numa=i // 4) for i in range(8)]
    pairs = set()
    for base in (0, 4):
        for i in range(base, base + 4):
            for j in range(i + 1, base + 4):
                pairs.add((i, j))
    _connect(gpus, pairs)
    return GpusInfo(version=_version(), devices=gpus)


def fixture_degraded_mesh(missing=((0, 1), (0, 2), (0, 3), (5, 6))) -> GpusInfo:
    """8-GPU mesh with some xGMI links DOWN (pairs fall back to PCIe).

    Models a production node with failed links (amdsmi exposes
    xgmi_link_status up/down); the regime where per-link-aware subset
    choice beats group-shape policies.
    """
    gpus = [_mk_gpu(i, numa=i // 4) for i in range(8)]
    down = {(min(a, b), max(a, b)) for a, b in missing}
    pairs = {
        (i, j)
        for i in range(8)
        for j in range(i + 1, 8)
        if (i, j) not in down
    }
    _connect(gpus, pairs)
    return GpusInfo(version=_version(), devices=gpus)


def fixture_4x_no_xgmi() -> GpusInfo:
    """Degenerate 4-GPU node without any xGMI (PCIe only)."""
    gpus = [_mk_gpu(i, numa=0) for i in range(4)]
    _connect(gpus, set())
    return GpusInfo(version=_version(), devices=gpus)


def fixture_json(info: GpusInfo) -> str:
    return info.to_json()


def fixture_cpx_2oam_8part() -> GpusInfo:
    """Two OAMs in CPX-style partitioning, 4 compute partitions each.

    Partitions of one OAM see each other over same-package INTERNAL
    links (amdsmi link type INTERNAL); partitions on different OAMs see
    single-hop xGMI.  Placement must prefer same-OAM subsets over
    cross-OAM ones (INTERNAL > XGMI > PCIE).
    """
    from .types import INTERNAL_GBPS_DEFAULT, LINK_INTERNAL

    gpus = []
    for i in range(8):
        g = _mk_gpu(i, numa=i // 4)
        g.compute_partition = "CPX"
        g.memory.vram_total_bytes = MI355X_VRAM_BYTES // 4  # per partition
        gpus.append(g)
    for a in gpus:
        for b in gpus:
            if a.index == b.index:
                continue
            same_oam = a.index // 4 == b.index // 4
            if same_oam:
                a.links.append(LinkInfo(
                    peer_index=b.index, type=LINK_INTERNAL, hops=1,
                    weight=5, bandwidth_gbps=INTERNAL_GBPS_DEFAULT, p2p=True,
                ))
            else:
                a.links.append(LinkInfo(
                    peer_index=b.index, type=LINK_XGMI, hops=1,
                    weight=15, bandwidth_gbps=XGMI_LINK_GBPS_DEFAULT, p2p=True,
                ))
    return GpusInfo(version=_version(), devices=gpus)
