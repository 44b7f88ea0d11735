"""Build the native components in-tree.

Artifacts (all inside the package so they ship with the repo snapshot):

* csrc/bin/amdsmiinfo   — C++ enumerator over libamd_smi (g++)
* csrc/bin/rcclprobe    — HIP + librccl all-reduce probe (hipcc, gfx950)
* _schedcore*.so        — pybind11 scheduler hot path (g++)
* _ext/_gpuprobe*.so    — torch extension with CDNA4 bandwidth kernels
                          (torch.utils.cpp_extension -> hipcc, gfx950)

Rebuilds are skipped when the artifact is newer than its source.
"""

from __future__ import annotations

import os
import subprocess
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
BIN = os.path.join(CSRC, "bin")
EXT_DIR = os.path.join(PKG_DIR, "_ext")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
HIPCC = os.path.join(ROCM, "bin", "hipcc")
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _needs_build(target: str, *sources: str) -> bool:
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    return any(os.path.getmtime(s) > t for s in sources)


def _run(cmd, **kw):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True, **kw)


def build_amdsmiinfo() -> str:
    os.makedirs(BIN, exist_ok=True)
    src = os.path.join(CSRC, "amdsmiinfo.cpp")
    out = os.path.join(BIN, "amdsmiinfo")
    if _needs_build(out, src):
        _run([
            "g++", "-O2", "-std=c++17", src,
            f"-I{ROCM}/include", f"-L{ROCM}/lib", "-lamd_smi",
            f"-Wl,-rpath,{ROCM}/lib", "-o", out,
        ])
    return out


def build_rcclprobe() -> str:
    os.makedirs(BIN, exist_ok=True)
    src = os.path.join(CSRC, "rcclprobe.cpp")
    out = os.path.join(BIN, "rcclprobe")
    if _needs_build(out, src):
        _run([
            HIPCC, f"--offload-arch={GFX_ARCH}", "-O2", "-std=c++17", src,
            f"-L{ROCM}/lib", "-lrccl", f"-Wl,-rpath,{ROCM}/lib", "-o", out,
        ])
    return out


def build_schedcore() -> str:
    import pybind11

    src = os.path.join(CSRC, "schedcore.cpp")
    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out = os.path.join(PKG_DIR, f"_schedcore{suffix}")
    if _needs_build(out, src):
        py_inc = sysconfig.get_paths()["include"]
        _run([
            "g++", "-O3", "-shared", "-fPIC", "-std=c++17", src,
            f"-I{pybind11.get_include()}", f"-I{py_inc}", "-o", out,
        ])
    return out


def build_gpuprobe(verbose: bool = True) -> str:
    """HIP bandwidth kernels as a torch extension, built into _ext/."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", GFX_ARCH)
    os.makedirs(EXT_DIR, exist_ok=True)
    src = os.path.join(CSRC, "gpuprobe.hip")
    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out = os.path.join(EXT_DIR, "_gpuprobe" + suffix)
    alt = os.path.join(EXT_DIR, "_gpuprobe.so")
    if not (_needs_build(out, src) and _needs_build(alt, src)):
        return out if os.path.exists(out) else alt
    from torch.utils.cpp_extension import load

    load(
        name="_gpuprobe",
        sources=[src],
        build_directory=EXT_DIR,
        extra_cflags=["-O3"],
        verbose=verbose,
        is_python_module=False,  # just build; import happens lazily
    )
    return out if os.path.exists(out) else alt


def build_all(verbose: bool = True) -> None:
    build_amdsmiinfo()
    build_rcclprobe()
    build_schedcore()
    build_gpuprobe(verbose=verbose)


if __name__ == "__main__":
    build_all()
