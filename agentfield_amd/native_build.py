"""Build agentfield_amd._native (pybind11 + libcrypto) in-tree."""
from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

PKG = Path(__file__).resolve().parent
SRC = PKG / "native" / "module.cpp"
EXT = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
OUT = PKG / f"_native{EXT}"


ASAN_OUT = PKG / f"_native_asan{EXT}"


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and OUT.exists() and OUT.stat().st_mtime > SRC.stat().st_mtime:
        return OUT
    import pybind11
    cmd = [
        "g++", "-O2", "-shared", "-fPIC", "-std=c++17",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        str(SRC), "-lcrypto", "-o", str(OUT),
    ]
    if verbose:
        print("[native]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


def build_asan(force: bool = False, verbose: bool = True) -> Path:
    """AddressSanitizer build of the native extension (the
    new framework adds the sanitizer tier the reference never had).
    Import it with LD_PRELOAD=libasan.so — see tests/test_sanitizer.py."""
    if not force and ASAN_OUT.exists() and \
            ASAN_OUT.stat().st_mtime > SRC.stat().st_mtime:
        return ASAN_OUT
    import pybind11
    cmd = [
        "g++", "-O1", "-g", "-fsanitize=address", "-fno-omit-frame-pointer",
        "-shared", "-fPIC", "-std=c++17",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        str(SRC), "-lcrypto", "-o", str(ASAN_OUT),
    ]
    if verbose:
        print("[native-asan]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return ASAN_OUT


TSAN_OUT = ASAN_OUT.with_name(ASAN_OUT.name.replace("asan", "tsan")) \
    if "asan" in ASAN_OUT.name else ASAN_OUT.with_suffix(".tsan.so")


def build_tsan(force: bool = False, verbose: bool = True) -> Path:
    """ThreadSanitizer build of the native extension.  The engine calls
    the scheduler from one driver thread, but the Ed25519/AES-GCM
    helpers run under FastAPI worker THREADS concurrently — that is the
    surface TSAN guards (see tests/test_sanitizer.py)."""
    if not force and TSAN_OUT.exists() and \
            TSAN_OUT.stat().st_mtime > SRC.stat().st_mtime:
        return TSAN_OUT
    import pybind11
    cmd = [
        "g++", "-O1", "-g", "-fsanitize=thread", "-fno-omit-frame-pointer",
        "-shared", "-fPIC", "-std=c++17",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        str(SRC), "-lcrypto", "-o", str(TSAN_OUT),
    ]
    if verbose:
        print("[native-tsan]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return TSAN_OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
