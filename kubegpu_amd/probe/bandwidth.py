"""HBM bandwidth probe — wrapper over the _gpuprobe HIP extension.

The k=1 degenerate point of the BASELINE.md bandwidth curve: a 1-GPU
"scheduled set" has no interconnect, so the sanity number is HBM3E
streaming bandwidth from the hand-written CDNA4 copy kernel
(csrc/gpuprobe.hip; ≈6.3 TB/s achievable of the 8 TB/s spec).

Fails LOUDLY when a GPU is present but the native extension is missing —
GPU tests must never silently fall back to eager PyTorch.
"""

from __future__ import annotations

import importlib
import os
import sys

_EXT_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "_ext")
_ext = None


class NativeExtensionMissing(RuntimeError):
    pass


def load_ext(required: bool = True):
    """Import the _gpuprobe torch extension built by build_native."""
    global _ext
    if _ext is not None:
        return _ext
    import torch  # noqa: F401  (extension needs torch symbols loaded)

    if _EXT_DIR not in sys.path:
        sys.path.insert(0, _EXT_DIR)
    try:
        _ext = importlib.import_module("_gpuprobe")
    except ImportError as e:
        if required or torch.cuda.is_available():
            raise NativeExtensionMissing(
                "_gpuprobe HIP extension is not built; run "
                "`python -m kubegpu_amd.build_native` (gfx950) — refusing "
                f"to fall back to eager PyTorch on a GPU box: {e}"
            ) from e
        return None
    return _ext


def copy(dst, src) -> None:
    """Launch the streaming copy kernel (numerics-testable)."""
    load_ext().copy(dst, src)


def d2d_copy_bw_gbps(
    nbytes: int = 1 << 30,
    iters: int = 20,
    blocks: int = 0,
    nontemporal: bool = True,
    variant: int = 0,
) -> float:
    """Timed device-to-device streaming-copy bandwidth (GB/s, R+W).

    variant: 0 = grid-stride NT, 1 = 4x-unrolled NT, 2 = contiguous-chunk
    NT, 3 = regular loads + NT stores.
    """
    return load_ext().copy_bw_gbps(nbytes, iters, blocks, nontemporal, variant)


def read_bw_gbps(nbytes: int = 1 << 30, iters: int = 20, blocks: int = 0) -> float:
    return load_ext().read_bw_gbps(nbytes, iters, blocks)


def write_bw_gbps(nbytes: int = 1 << 30, iters: int = 20, blocks: int = 0) -> float:
    """Timed write-only (nontemporal fill) bandwidth (GB/s)."""
    return load_ext().write_bw_gbps(nbytes, iters, blocks)
