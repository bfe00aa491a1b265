#!/usr/bin/env python3
"""Minimal PMC profiling target: runs ONLY our HIP kernels a few times.
Device RNG kernels crash rocprofv3's counter sampler, so all data is
generated on host and copied."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops


def h2d(*shape, scale=0.3):
    return (torch.randn(*shape) * scale).bfloat16().cuda()


def main():
    torch.manual_seed(0)
    D, Hq, Hk, page = 128, 32, 8, 16
    # prefill attention B=4 S=2048
    B, S = 4, 2048
    T = B * S
    q = h2d(T, Hq * D)
    maxp = S // page
    npages = 1 + B * maxp
    kc = h2d(npages, Hk, page, D)
    vc = h2d(npages, Hk, page, D)
    bt = torch.arange(1, npages, dtype=torch.int32).reshape(B, maxp).cuda()
    cu = torch.arange(0, T + 1, S, dtype=torch.int32).cuda()
    qs = torch.zeros(B, dtype=torch.int32).cuda()
    for _ in range(3):
        ops.attn_prefill(q, kc, vc, bt, qs, cu, [S] * B)
    torch.cuda.synchronize()
    # skinny GEMM M=16 qkv shape
    a = h2d(16, 4096)
    w = h2d(6144, 4096)
    for _ in range(3):
        ops.linear_skinny(a, w)
    torch.cuda.synchronize()
    # decode attention B=64 L=1024
    Bd, L = 64, 1024
    npages = 1 + Bd * (L // page)
    qd = h2d(Bd, Hq * D)
    kcd = h2d(npages, Hk, page, D)
    vcd = h2d(npages, Hk, page, D)
    btd = torch.arange(1, npages, dtype=torch.int32).reshape(Bd, L // page).cuda()
    lens = torch.full((Bd,), L, dtype=torch.int32).cuda()
    for _ in range(3):
        ops.attn_decode(qd, kcd, vcd, btd, lens, nsplit=2)
    torch.cuda.synchronize()
    # tiled MFMA GEMM 4096^3
    a2 = h2d(4096, 4096)
    w2 = h2d(4096, 4096)
    for _ in range(3):
        ops.gemm_bf16(a2, w2)
    torch.cuda.synchronize()
    print("pmc target done")


if __name__ == "__main__":
    main()
