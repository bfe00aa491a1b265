"""ctypes loader for the in-tree HIP kernel library (libafops.so).

The kernels are pure HIP (no torch headers); tensors cross the boundary as raw
device pointers + the current HIP stream.  On a GPU box the library MUST load
— ops raise instead of silently falling back to eager torch (the reference
framework has no model math to fall back to; ours must run the native path).
"""
from __future__ import annotations

import ctypes
import functools
from pathlib import Path

import torch

_LIB_PATH = Path(__file__).resolve().parent.parent / "libafops.so"


class AfOpsError(RuntimeError):
    pass


@functools.lru_cache(maxsize=1)
def lib() -> ctypes.CDLL:
    if not _LIB_PATH.exists():
        if torch.cuda.is_available():
            raise AfOpsError(
                f"libafops.so not built at {_LIB_PATH}; run "
                "`python -m agentfield_amd.build` (GPU path never falls back)")
        # CPU-only environment: try to build (hipcc cross-compiles w/o GPU)
        from agentfield_amd.build import build_afops
        build_afops()
    cdll = ctypes.CDLL(str(_LIB_PATH))
    _declare(cdll)
    return cdll


def _declare(l: ctypes.CDLL) -> None:
    p = ctypes.c_void_p
    i = ctypes.c_int
    i64 = ctypes.c_int64
    f = ctypes.c_float
    u32 = ctypes.c_uint32
    l.af_rmsnorm.argtypes = [p, p, p, p, p, f, i, i, p]
    l.af_rope_cache.argtypes = [p, p, p, p, p, p, p, p, p, f, f,
                                i, i, i, i, i64, i64, i64, i, p]
    l.af_silu_mul.argtypes = [p, p, p, f, f, i64, i64, p]
    l.af_add.argtypes = [p, p, p, i64, p]
    l.af_reshape_and_cache.argtypes = [p, p, p, p, p, i, i, i, i, p]
    l.af_embedding.argtypes = [p, p, p, p, i, i, p]
    l.af_attn_decode.argtypes = [p, p, p, p, p, p, p, p, f, i, i, i, i, i, i, i,
                                 i64, i, p]
    l.af_attn_prefill.argtypes = [p, p, p, p, p, p, p, p, p, f,
                                  i, i, i, i, i64, i, i, i, p]
    l.af_gemm_bf16.argtypes = [p, p, p, i, i, i, p]
    l.af_gemm_bf16_ring.argtypes = [p, p, p, i, i, i, p]
    l.af_gemm_mxfp8.argtypes = [p, p, p, p, p, i, i, i, p]
    l.af_gemm_bf16_q8.argtypes = [p, p, p, i, i, i, p]
    l.af_ipc_get_handle.argtypes = [p, p]
    l.af_ipc_open_handle.argtypes = [p, p]
    l.af_ipc_close_handle.argtypes = [p]
    l.af_oneshot_allreduce.argtypes = [p, p, p, i, i, ctypes.c_long,
                                       ctypes.c_long, ctypes.c_ulonglong, p]
    l.af_gemm_skinny.argtypes = [p, p, p, p, p, i, i, i, i, i,
                                 p, f, p, p]
    l.af_sample.argtypes = [p, p, p, p, p, p, u32, i, i, p]
    l.af_sample_topkp.argtypes = [p, p, p, p, p, p, p, p, p, p, p, p,
                                  u32, i, i, p]
    l.af_gather_rows.argtypes = [p, p, p, i, i, p]
    l.af_mfma_probe.argtypes = [p, p, p, p]
    l.af_mfma_mx_probe.argtypes = [p, p, p, p, p, p]
    l.af_lds_stride_probe.argtypes = [p, p, i, p]
    l.af_mfma_mx32_probe.argtypes = [p, p, p, p, p, i, p]
    l.af_axpy.argtypes = [p, p, f, i, p]
    l.af_device_sync.argtypes = []
    for fn in ("af_rmsnorm", "af_rope_cache", "af_silu_mul", "af_add",
               "af_reshape_and_cache", "af_embedding", "af_attn_decode",
               "af_attn_prefill", "af_gemm_bf16", "af_gemm_bf16_ring", "af_gemm_bf16_q8", "af_gemm_mxfp8", "af_oneshot_allreduce",
               "af_gemm_skinny", "af_sample",
               "af_sample_topkp", "af_gather_rows", "af_mfma_probe", "af_mfma_mx_probe", "af_lds_stride_probe", "af_mfma_mx32_probe", "af_axpy",
               "af_device_sync"):
        getattr(l, fn).restype = ctypes.c_int


def cur_stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def ptr(t: torch.Tensor | None) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    return ctypes.c_void_p(t.data_ptr())


def check(rc: int, name: str) -> None:
    if rc != 0:
        raise AfOpsError(f"{name} failed with code {rc} (hipError or 9xxx arg check)")


def native_loaded() -> bool:
    try:
        lib()
        return True
    except Exception:
        return False
