"""GPU inventory schema — the MI355X-native analog of nvgputypes.

The reference's discovery JSON (nvidiagpuplugin/gpu/nvgputypes/types.go:22-43)
carries UUID, /dev path, memory, PCI bus id + bandwidth and a pairwise
"link level" topology.  The MI355X schema replaces the NVML level scale
with the explicit per-link xGMI graph: every device lists its peers with
link type (XGMI / PCIE), hop count, link weight and per-link bandwidth
(xGMI on an 8-GPU MI355X hive: 7 point-to-point links × ≈153 GB/s/GPU),
plus gfx950 identity, 288 GB HBM3E size and the /dev/dri render node used
for container injection.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List

LINK_XGMI = "XGMI"
LINK_PCIE = "PCIE"
# Same-package fabric between compute partitions of one OAM (CPX/DPX
# modes): amdsmi reports link type INTERNAL for them.
LINK_INTERNAL = "INTERNAL"

# Single-hop xGMI link bandwidth on MI355X (GB/s per link, spec ≈153).
XGMI_LINK_GBPS_DEFAULT = 153.0
# Host PCIe Gen5 x16 bandwidth (GB/s) used when two GPUs only reach each
# other through the host bridge.
PCIE_GBPS_DEFAULT = 63.0
# Same-package (INTERNAL) default: partitions of one OAM share the
# on-package Infinity Fabric / HBM path — far faster than any xGMI hop.
# Conservative placeholder (no public per-partition figure); what
# matters for placement is INTERNAL > XGMI > PCIE ordering.  A measured
# replacement needs a CPX-partitioned lease: the round-2 box reported
# SPX/NPS1 only (profiles/partition_modes_mi355x.json), so this stays a
# documented placeholder, never reported as a measurement.
INTERNAL_GBPS_DEFAULT = 300.0


@dataclass
class LinkInfo:
    """One edge of the interconnect graph, from a device to a peer."""

    peer_index: int
    type: str = LINK_PCIE
    hops: int = 1
    weight: int = 0  # amdsmi link weight (lower = closer)
    bandwidth_gbps: float = 0.0
    p2p: bool = False

    def to_dict(self) -> dict:
        return {
            "peer_index": self.peer_index,
            "type": self.type,
            "hops": self.hops,
            "weight": self.weight,
            "bandwidth_gbps": self.bandwidth_gbps,
            "p2p": self.p2p,
        }

    @staticmethod
    def from_dict(d: dict) -> "LinkInfo":
        return LinkInfo(
            peer_index=int(d["peer_index"]),
            type=str(d.get("type", LINK_PCIE)),
            hops=int(d.get("hops", 1)),
            weight=int(d.get("weight", 0)),
            bandwidth_gbps=float(d.get("bandwidth_gbps", 0.0)),
            p2p=bool(d.get("p2p", False)),
        )


@dataclass
class MemoryInfo:
    vram_total_bytes: int = 0
    vram_type: str = "HBM3E"
    vram_bandwidth_gbps: float = 0.0

    def to_dict(self) -> dict:
        return {
            "vram_total_bytes": self.vram_total_bytes,
            "vram_type": self.vram_type,
            "vram_bandwidth_gbps": self.vram_bandwidth_gbps,
        }

    @staticmethod
    def from_dict(d: dict) -> "MemoryInfo":
        return MemoryInfo(
            vram_total_bytes=int(d.get("vram_total_bytes", 0)),
            vram_type=str(d.get("vram_type", "HBM3E")),
            vram_bandwidth_gbps=float(d.get("vram_bandwidth_gbps", 0.0)),
        )


@dataclass
class GpuInfo:
    """One GPU (cf. nvgputypes.GpuInfo, types.go:22-34)."""

    uuid: str = ""
    model: str = ""
    device_id: str = ""
    gfx_target: str = ""
    index: int = 0
    bdf: str = ""
    render_path: str = ""
    card_path: str = ""
    numa_node: int = 0
    compute_units: int = 0
    ecc_correctable: int = 0
    ecc_uncorrectable: int = 0
    process_count: int = 0  # compute processes seen by amdsmi (external in-use)
    compute_partition: str = ""  # MI355X partition mode (SPX/DPX/.../CPX)
    memory_partition: str = ""  # NUMA-per-socket mode (NPS1/NPS4/...)
    memory: MemoryInfo = field(default_factory=MemoryInfo)
    links: List[LinkInfo] = field(default_factory=list)

    # Runtime-only fields (never serialized; cf. the reference's
    # Found/Index/InUse/TopoDone/Name runtime fields, types.go:28-33).
    found: bool = False
    in_use: bool = False
    topo_done: bool = False
    name: str = ""  # topology-prefixed: gpugrp1/H/gpugrp0/G/gpu/<uuid>

    @property
    def healthy(self) -> bool:
        """Any uncorrectable (fatal) ECC error marks the GPU unhealthy;
        correctable errors are informational only."""
        return self.ecc_uncorrectable == 0

    def to_dict(self) -> dict:
        return {
            "uuid": self.uuid,
            "model": self.model,
            "device_id": self.device_id,
            "gfx_target": self.gfx_target,
            "index": self.index,
            "bdf": self.bdf,
            "render_path": self.render_path,
            "card_path": self.card_path,
            "numa_node": self.numa_node,
            "compute_units": self.compute_units,
            "ecc_correctable": self.ecc_correctable,
            "ecc_uncorrectable": self.ecc_uncorrectable,
            "process_count": self.process_count,
            "compute_partition": self.compute_partition,
            "memory_partition": self.memory_partition,
            "memory": self.memory.to_dict(),
            "links": [l.to_dict() for l in self.links],
        }

    @staticmethod
    def from_dict(d: dict) -> "GpuInfo":
        return GpuInfo(
            uuid=str(d.get("uuid", "")),
            model=str(d.get("model", "")),
            device_id=str(d.get("device_id", "")),
            gfx_target=str(d.get("gfx_target", "")),
            index=int(d.get("index", 0)),
            bdf=str(d.get("bdf", "")),
            render_path=str(d.get("render_path", "")),
            card_path=str(d.get("card_path", "")),
            numa_node=int(d.get("numa_node", 0)),
            compute_units=int(d.get("compute_units", 0)),
            ecc_correctable=int(d.get("ecc_correctable", 0)),
            ecc_uncorrectable=int(d.get("ecc_uncorrectable", 0)),
            process_count=int(d.get("process_count", 0)),
            compute_partition=str(d.get("compute_partition", "")),
            memory_partition=str(d.get("memory_partition", "")),
            memory=MemoryInfo.from_dict(d.get("memory", {})),
            links=[LinkInfo.from_dict(x) for x in d.get("links", [])],
        )


@dataclass
class VersionInfo:
    driver: str = ""
    rocm: str = ""
    amdsmi: str = ""

    def to_dict(self) -> dict:
        return {"driver": self.driver, "rocm": self.rocm, "amdsmi": self.amdsmi}

    @staticmethod
    def from_dict(d: dict) -> "VersionInfo":
        return VersionInfo(
            driver=str(d.get("driver", "")),
            rocm=str(d.get("rocm", "")),
            amdsmi=str(d.get("amdsmi", "")),
        )


@dataclass
class GpusInfo:
    """Whole-node inventory (cf. nvgputypes.GpusInfo, types.go:36-43)."""

    version: VersionInfo = field(default_factory=VersionInfo)
    devices: List[GpuInfo] = field(default_factory=list)

    def to_json(self) -> str:
        return json.dumps(
            {
                "version": self.version.to_dict(),
                "devices": [g.to_dict() for g in self.devices],
            },
            indent=1,
        )

    @staticmethod
    def from_json(payload: str) -> "GpusInfo":
        d = json.loads(payload)
        return GpusInfo(
            version=VersionInfo.from_dict(d.get("version", {})),
            devices=[GpuInfo.from_dict(x) for x in d.get("devices", [])],
        )

    def bandwidth_matrix(self) -> Dict[int, Dict[int, float]]:
        """Pairwise effective p2p bandwidth (GB/s) from the link graph.

        Missing edges (no entry for a peer) fall back to the host PCIe
        path.  This is the matrix the scheduler's subset scorer consumes.
        """
        idx = [g.index for g in self.devices]
        bw: Dict[int, Dict[int, float]] = {i: {} for i in idx}
        for g in self.devices:
            for l in g.links:
                if l.peer_index == g.index:
                    continue
                gbps = l.bandwidth_gbps
                if gbps <= 0.0:
                    if l.type == LINK_XGMI:
                        gbps = XGMI_LINK_GBPS_DEFAULT / max(1, l.hops)
                    elif l.type == LINK_INTERNAL:
                        gbps = INTERNAL_GBPS_DEFAULT
                    else:
                        gbps = PCIE_GBPS_DEFAULT
                if l.type in (LINK_XGMI, LINK_INTERNAL) and not l.p2p:
                    # link advertised but peer access disabled/down:
                    # traffic bounces through the host path
                    gbps = min(gbps, PCIE_GBPS_DEFAULT)
                bw[g.index][l.peer_index] = gbps
        for i in idx:
            for j in idx:
                if i != j and j not in bw[i]:
                    bw[i][j] = PCIE_GBPS_DEFAULT
        return bw


def direct_xgmi_pairs(info: GpusInfo) -> List[tuple]:
    """Pairs (i, j) connected by a single-hop xGMI or same-package
    (INTERNAL, CPX-partition) link."""
    out = []
    for g in info.devices:
        for l in g.links:
            if (
                l.type in (LINK_XGMI, LINK_INTERNAL)
                and l.hops <= 1
                and g.index < l.peer_index
            ):
                out.append((g.index, l.peer_index))
    return out
