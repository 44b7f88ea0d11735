"""Discovery backends.

Backend interface parity: the reference's NvidiaPlugin interface exposes
GetGPUInfo() -> JSON bytes (nvidia_plugin.go:7-10); its production path
exec's an out-of-process `nvmlinfo json` binary so a cgo/NVML crash cannot
kill the node agent (nvgputypes/types.go:45-58).  We keep both properties:

* ``AmdSmiBackend``  — runs the in-tree native ``amdsmiinfo json`` binary
  (C++ over libamd_smi) in a subprocess.  Crash containment preserved.
* ``SysfsBackend``   — pure-Python fallback reading the KFD topology under
  /sys/class/kfd/kfd/topology (no amdsmi library needed).
* ``FakeBackend``    — fixture-driven test double (cf.
  nvidia_fake_plugin.go:10-41).

There is deliberately no GetGPUCommandLine analog: ROCm needs no helper
daemon — allocation computes /dev/kfd + render-node paths directly
(north star).
"""

from __future__ import annotations

import abc
import glob
import os
import subprocess
from typing import Dict, List, Optional

from ..api import utils
from .types import (
    LINK_PCIE,
    LINK_XGMI,
    GpuInfo,
    GpusInfo,
    LinkInfo,
    MemoryInfo,
    VersionInfo,
)


class DiscoveryError(RuntimeError):
    pass


class Backend(abc.ABC):
    """GPU inventory source."""

    @abc.abstractmethod
    def get_gpu_info(self) -> bytes:
        """Return the inventory as JSON bytes (raise DiscoveryError)."""

    def get_devices(self) -> GpusInfo:
        payload = self.get_gpu_info()
        try:
            return GpusInfo.from_json(payload.decode())
        except (ValueError, KeyError, TypeError) as e:
            # corrupt subprocess output must surface as a discovery
            # failure (which Start/update tolerate), not a crash — the
            # whole point of the subprocess isolation
            raise DiscoveryError(
                f"unparseable inventory payload ({len(payload)} bytes): {e}"
            ) from e


class FakeBackend(Backend):
    """In-memory fixture backend (cf. NvidiaFakePlugin)."""

    def __init__(self, info: GpusInfo):
        self._info = info

    def get_gpu_info(self) -> bytes:
        return self._info.to_json().encode()

    # test hooks
    def set_info(self, info: GpusInfo) -> None:
        self._info = info


class CrashingBackend(Backend):
    """Fault-injection backend: always fails (subprocess-crash analog)."""

    def get_gpu_info(self) -> bytes:
        raise DiscoveryError("injected discovery failure")


def _default_amdsmiinfo_path() -> str:
    env = os.environ.get("KUBEGPU_AMDSMIINFO")
    if env:
        return env
    here = os.path.dirname(os.path.abspath(__file__))
    candidates = [
        os.path.join(here, "..", "csrc", "bin", "amdsmiinfo"),
        "/usr/local/bin/amdsmiinfo",
    ]
    for c in candidates:
        c = os.path.normpath(c)
        if os.path.exists(c):
            return c
    return "amdsmiinfo"


class AmdSmiBackend(Backend):
    """Subprocess amdsmiinfo backend (production path on MI355X nodes)."""

    def __init__(self, binary: Optional[str] = None, timeout_s: float = 30.0):
        self._binary = binary or _default_amdsmiinfo_path()
        self._timeout_s = timeout_s

    def get_gpu_info(self) -> bytes:
        try:
            out = subprocess.run(
                [self._binary, "json"],
                capture_output=True,
                timeout=self._timeout_s,
                check=True,
            )
        except FileNotFoundError as e:
            raise DiscoveryError(f"amdsmiinfo binary not found: {self._binary}") from e
        except subprocess.TimeoutExpired as e:
            raise DiscoveryError(f"amdsmiinfo timed out after {self._timeout_s}s") from e
        except subprocess.CalledProcessError as e:
            raise DiscoveryError(
                f"amdsmiinfo failed rc={e.returncode}: {e.stderr[-500:] if e.stderr else ''}"
            ) from e
        return out.stdout


KFD_TOPO = "/sys/class/kfd/kfd/topology/nodes"


def _read_props(path: str) -> Dict[str, int]:
    props: Dict[str, int] = {}
    try:
        with open(path) as f:
            for line in f:
                parts = line.split()
                if len(parts) == 2:
                    try:
                        props[parts[0]] = int(parts[1])
                    except ValueError:
                        pass
    except OSError:
        pass
    return props


class SysfsBackend(Backend):
    """KFD-sysfs fallback enumerator (no amdsmi library required).

    Reads /sys/class/kfd/kfd/topology/nodes/<n>/{properties,io_links/*}.
    GPU nodes are those with simd_count > 0.  xGMI peers show up as
    io_links with type 2 (XGMI) in the KFD topology.
    """

    def __init__(self, root: str = KFD_TOPO):
        self._root = root

    def get_gpu_info(self) -> bytes:
        if not os.path.isdir(self._root):
            raise DiscoveryError(f"no KFD topology at {self._root}")
        nodes = sorted(
            (d for d in os.listdir(self._root) if d.isdigit()), key=int
        )
        kfd_to_gpu: Dict[int, int] = {}
        raw: List[dict] = []
        for n in nodes:
            props = _read_props(os.path.join(self._root, n, "properties"))
            if props.get("simd_count", 0) <= 0:
                continue
            raw.append({"kfd_node": int(n), "props": props, "dir": os.path.join(self._root, n)})
        gpus: List[GpuInfo] = []
        for gpu_index, entry in enumerate(raw):
            kfd_to_gpu[entry["kfd_node"]] = gpu_index
        for gpu_index, entry in enumerate(raw):
            props = entry["props"]
            gfx_ver = props.get("gfx_target_version", 0)
            # gfx_target_version encodes e.g. 90500 -> gfx950 family coding
            # (major*10000 + minor*100 + step).
            major, minor, step = gfx_ver // 10000, (gfx_ver // 100) % 100, gfx_ver % 100
            drm_minor = props.get("drm_render_minor", 0)
            mem_bytes = 0
            mem_props = glob.glob(os.path.join(entry["dir"], "mem_banks", "*", "properties"))
            for mp in mem_props:
                p = _read_props(mp)
                # heap types 1/2 = FB public/private (VRAM)
                if p.get("heap_type", 0) in (1, 2):
                    mem_bytes += p.get("size_in_bytes", 0)
            g = GpuInfo(
                uuid=f"GPU-kfd-{props.get('unique_id', entry['kfd_node']):x}"
                if props.get("unique_id")
                else f"GPU-kfdnode-{entry['kfd_node']}",
                model=f"gfx{major}{minor:x}{step:x}" if gfx_ver else "AMD GPU",
                device_id=hex(props.get("device_id", 0)),
                gfx_target=f"gfx{major}{minor:x}{step:x}" if gfx_ver else "",
                index=gpu_index,
                bdf="%04x:%02x:%02x.%x" % (
                    (props.get("domain", 0)),
                    (props.get("location_id", 0) >> 8) & 0xFF,
                    (props.get("location_id", 0) >> 3) & 0x1F,
                    props.get("location_id", 0) & 0x7,
                ),
                render_path=f"/dev/dri/renderD{drm_minor}" if drm_minor else "",
                card_path="",
                numa_node=props.get("numa_node", 0) if "numa_node" in props else 0,
                compute_units=props.get("simd_count", 0) // max(1, props.get("simd_per_cu", 4)),
                memory=MemoryInfo(vram_total_bytes=mem_bytes),
            )
            # io_links: peers by KFD node id
            for lp in glob.glob(os.path.join(entry["dir"], "io_links", "*", "properties")):
                p = _read_props(lp)
                peer_kfd = p.get("node_to", -1)
                if peer_kfd not in kfd_to_gpu:
                    continue
                # KFD iolink type: 2 = XGMI, 11 = PCIe (amdkfd crat.h)
                is_xgmi = p.get("type", 0) == 2
                g.links.append(
                    LinkInfo(
                        peer_index=kfd_to_gpu[peer_kfd],
                        type=LINK_XGMI if is_xgmi else LINK_PCIE,
                        hops=1 if is_xgmi else 2,
                        weight=p.get("weight", 0),
                        bandwidth_gbps=float(p.get("max_bandwidth", 0)) / 1000.0
                        if p.get("max_bandwidth", 0)
                        else 0.0,
                        p2p=is_xgmi,
                    )
                )
            gpus.append(g)
        rocm = "sysfs"
        try:
            with open("/opt/rocm/.info/version") as f:
                v = f.read().strip()
                if v:
                    rocm = v
        except OSError:
            pass
        info = GpusInfo(version=VersionInfo(rocm=rocm), devices=gpus)
        return info.to_json().encode()


def default_backend() -> Backend:
    """amdsmiinfo when the binary exists, else KFD sysfs."""
    path = _default_amdsmiinfo_path()
    if os.path.exists(path):
        return AmdSmiBackend(path)
    utils.logf(2, "amdsmiinfo binary not found, using sysfs backend")
    return SysfsBackend()
