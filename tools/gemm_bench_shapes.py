#!/usr/bin/env python3
"""q8 vs hipBLASLt on the FLAGSHIP PREFILL shapes (tall M)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    for (M, N, K, tag) in [
            (16384, 6144, 4096, "qkv"),
            (16384, 28672, 4096, "gate_up"),
            (16384, 4096, 14336, "down"),
            (16384, 128256, 4096, "lm_head"),
            (65536, 6144, 4096, "qkv-full"),
    ]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        tf = 2.0 * M * N * K / 1e12
        t_blas = timeit(lambda: a @ w.t())
        row = {"shape": tag, "MNK": [M, N, K],
               "blas_tf": round(tf / t_blas, 1)}
        if N % 256 == 0 and K % 64 == 0:
            t_q8 = timeit(lambda: ops.gemm_bf16_q8(a, w))
            row["q8_tf"] = round(tf / t_q8, 1)
        print(row, flush=True)
        del a, w
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
