#!/usr/bin/env python3
"""Per-kernel microbenchmarks on MI355X: our HIP kernels vs roofline and
vs hipBLASLt (torch.matmul) where applicable.  Prints one JSON line per case.
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops

DEV = "cuda"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemm():
    for (M, N, K) in [(8192, 8192, 8192), (4096, 4096, 4096),
                      (8192, 14336 * 2, 4096), (16, 4096, 4096)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        tf = 2.0 * M * N * K / 1e12
        t_ours = timeit(lambda: ops.gemm_bf16(a, w))
        t_blas = timeit(lambda: a @ w.t())
        row = {"op": "gemm_bf16", "MNK": [M, N, K],
               "ours_tflops": round(tf / t_ours, 1),
               "hipblaslt_tflops": round(tf / t_blas, 1)}
        if K % 32 == 0 and M >= 256:
            t_ring = timeit(lambda: ops.gemm_bf16_ring(a, w))
            row["ring_tflops"] = round(tf / t_ring, 1)
        if K % 64 == 0 and K >= 128 and M >= 256:
            t_q8 = timeit(lambda: ops.gemm_bf16_q8(a, w))
            row["q8_tflops"] = round(tf / t_q8, 1)
        if K % 128 == 0 and K >= 256 and M >= 256:
            from agentfield_amd.quant import quantize_mx
            a8, sa = quantize_mx(a.float().cpu())
            w8, sw = quantize_mx(w.float().cpu())
            a8, sa, w8, sw = (t.to(DEV) for t in (a8, sa, w8, sw))
            t_mx = timeit(lambda: ops.gemm_mxfp8(a8, sa, w8, sw))
            row["mxfp8_tflops"] = round(tf / t_mx, 1)
            # torch fp8 rowwise-scaled baseline when available
            try:
                af = a.float()
                wf = w.float()
                sa_r = af.abs().amax(1, keepdim=True) / 448.0
                sw_r = wf.abs().amax(1, keepdim=True) / 448.0
                a_f8 = (af / sa_r).to(torch.float8_e4m3fn)
                w_f8 = (wf / sw_r).to(torch.float8_e4m3fn)
                t_s = timeit(lambda: torch._scaled_mm(
                    a_f8, w_f8.t(), scale_a=sa_r, scale_b=sw_r.t(),
                    out_dtype=torch.bfloat16))
                row["scaled_mm_tflops"] = round(tf / t_s, 1)
            except Exception as e:
                row["scaled_mm_tflops"] = str(e)[:60]
        print(json.dumps(row))


def bench_skinny():
    for (M, N, K) in [(16, 6144, 4096), (16, 4096, 4096), (16, 28672, 4096),
                      (16, 4096, 14336), (64, 28672, 4096), (16, 128256, 4096),
                      (64, 4096, 14336), (64, 128256, 4096), (64, 6144, 4096),
                      (128, 6144, 4096), (128, 4096, 4096), (128, 28672, 4096),
                      (128, 4096, 14336)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        gb = N * K * 2 / 1e9
        t_ours = timeit(lambda: ops.linear_skinny(a, w), iters=30)
        t_blas = timeit(lambda: a @ w.t(), iters=30)
        print(json.dumps({"op": "gemm_skinny", "MNK": [M, N, K],
                          "ours_tbps": round(gb / t_ours / 1e3, 2),
                          "hipblaslt_tbps": round(gb / t_blas / 1e3, 2),
                          "ours_us": round(t_ours * 1e6, 1),
                          "blas_us": round(t_blas * 1e6, 1)}))


def bench_prefill(Hq=32, Hk=8, D=128, page=16):
    for (B, S) in [(16, 512), (4, 2048), (1, 8192)]:
        T = B * S
        q = torch.randn(T, Hq * D, dtype=torch.bfloat16, device=DEV)
        maxp = (S + page - 1) // page
        npages = 1 + B * maxp
        kc = torch.randn(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
        vc = torch.randn_like(kc)
        bt = torch.arange(1, npages, dtype=torch.int32,
                          device=DEV).reshape(B, maxp)
        cu = torch.arange(0, T + 1, S, dtype=torch.int32, device=DEV)
        qstart = torch.zeros(B, dtype=torch.int32, device=DEV)
        lens = [S] * B
        # causal flops: 2 * (QK + PV) * 0.5 * S^2 * D * Hq per seq
        tf = 2.0 * 2 * 0.5 * S * S * D * Hq * B / 1e12
        t = timeit(lambda: ops.attn_prefill(q, kc, vc, bt, qstart, cu, lens))
        print(json.dumps({"op": "attn_prefill_paged", "B": B, "S": S,
                          "tflops": round(tf / t, 1), "ms": round(t * 1e3, 3)}))


def bench_decode(Hq=32, Hk=8, D=128, page=None):
    import itertools
    for (B, L, nsplit), page in itertools.product(
            [(16, 576, 4), (128, 576, 1), (64, 1024, 1), (256, 1024, 1)],
            (16, 32, 64)):
        npages = B * ((L + page - 1) // page) + 1
        q = torch.randn(B, Hq * D, dtype=torch.bfloat16, device=DEV)
        kc = torch.randn(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
        vc = torch.randn_like(kc)
        maxp = (L + page - 1) // page
        bt = torch.arange(1, npages, dtype=torch.int32, device=DEV).reshape(B, maxp)
        lens = torch.full((B,), L, dtype=torch.int32, device=DEV)
        gb = 2.0 * B * L * Hk * D * 2 / 1e9  # K+V bytes read
        t = timeit(lambda: ops.attn_decode(q, kc, vc, bt, lens, nsplit=nsplit))
        print(json.dumps({"op": "attn_decode", "B": B, "L": L, "page": page,
                          "nsplit": nsplit, "gbps": round(gb / t, 1),
                          "us": round(t * 1e6, 1)}))


def bench_norm_rope(H=4096):
    for T in (16, 8192):
        x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
        res = torch.randn_like(x)
        gb = 4.0 * T * H * 2 / 1e9  # read x,res; write out,res
        t = timeit(lambda: ops.rmsnorm(x, w, 1e-5, residual=res))
        print(json.dumps({"op": "rmsnorm_fused", "T": T, "gbps": round(gb / t, 1),
                          "us": round(t * 1e6, 1)}))
        gu = torch.randn(T, 28672, dtype=torch.bfloat16, device=DEV)
        gb2 = (T * 28672 + T * 14336) * 2 / 1e9
        t2 = timeit(lambda: ops.silu_and_mul(gu))
        print(json.dumps({"op": "silu_mul", "T": T, "gbps": round(gb2 / t2, 1),
                          "us": round(t2 * 1e6, 1)}))
    x = torch.randn(65536, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    res = torch.randn_like(x)
    gb = 4.0 * 65536 * H * 2 / 1e9
    t = timeit(lambda: ops.rmsnorm(x, w, 1e-5, residual=res), iters=10)
    print(json.dumps({"op": "rmsnorm_fused", "T": 65536,
                      "gbps": round(gb / t, 1), "us": round(t * 1e6, 1)}))


def bench_sample(B=64, V=128256):
    logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV)
    st = ops.SamplerState(B, DEV)
    temps = torch.full((B,), 0.8, device=DEV)
    t = timeit(lambda: ops.sample(logits, temps, st))
    print(json.dumps({"op": "sample", "B": B, "V": V, "us": round(t * 1e6, 1)}))


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--only", default=None)
    args = ap.parse_args()
    torch.manual_seed(0)
    for name, fn in [("gemm", bench_gemm), ("skinny", bench_skinny),
                     ("prefill", bench_prefill),
                     ("decode", bench_decode), ("norm", bench_norm_rope),
                     ("sample", bench_sample)]:
        if args.only in (None, name):
            fn()
