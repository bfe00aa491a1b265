"""Build driver for the in-tree native libraries.

Two artifacts, both built in-tree so they travel with the repo snapshot:
  * agentfield_amd/libafops.so   — pure-HIP gfx950 kernel library (ctypes C API)
  * agentfield_amd/_native*.so   — CPU-side C++ (pybind11): scheduler/allocator,
                                   Ed25519 via libcrypto (built by native/setup)

hipcc cross-compiles gfx950 without a GPU, so this runs in CPU-only CI too.
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
LIB = PKG_DIR / "libafops.so"

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("AF_OFFLOAD_ARCH", "gfx950")

HIP_SOURCES = [
    "rmsnorm.hip",
    "rope.hip",
    "activation.hip",
    "cache.hip",
    "attn_decode.hip",
    "attn_prefill.hip",
    "gemm.hip",
    "gemm_ring.hip",
    "gemm_q8.hip",
    "gemm_mxfp8.hip",
    "allreduce.hip",
    "gemm_skinny.hip",
    "sampling.hip",
    "probe.hip",
]


def _needs_build() -> bool:
    if not LIB.exists():
        return True
    lib_m = LIB.stat().st_mtime
    deps = [CSRC / s for s in HIP_SOURCES] + [CSRC / "common.h", Path(__file__)]
    return any(d.stat().st_mtime > lib_m for d in deps)


def build_afops(force: bool = False, verbose: bool = True) -> Path:
    """Compile the HIP kernel library for gfx950 (incremental per-object)."""
    if not force and not _needs_build():
        return LIB
    objdir = CSRC / ".obj"
    objdir.mkdir(exist_ok=True)
    objs = []
    for src in HIP_SOURCES:
        sp = CSRC / src
        op = objdir / (src + ".o")
        objs.append(op)
        if (not force and op.exists()
                and op.stat().st_mtime > sp.stat().st_mtime
                and op.stat().st_mtime > (CSRC / "common.h").stat().st_mtime):
            continue
        cmd = [
            HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
            "-fvisibility=hidden", "-c", str(sp), "-o", str(op),
        ]
        if verbose:
            print("[afops]", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)
    cmd = [HIPCC, f"--offload-arch={ARCH}", "-shared", "-fPIC",
           *[str(o) for o in objs], "-o", str(LIB)]
    if verbose:
        print("[afops]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return LIB


def build_native(force: bool = False, verbose: bool = True):
    """Build the CPU-side pybind11 extension (scheduler, allocator, crypto)."""
    import sys as _sys
    _sys.path.insert(0, str(PKG_DIR.parent))
    from agentfield_amd.native_build import build as _build  # lazy import
    return _build(force=force, verbose=verbose)


def build_all(force: bool = False):
    build_afops(force=force)
    build_native(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
