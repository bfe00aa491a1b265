#!/usr/bin/env python3
"""Map the A<->B hw-k pairing of mfma_scale_f32_16x16x128_f8f6f4:
one-hot A at memory k=p, one-hot B at memory k=q; D[0][0] != 0 iff the
kernel's load positions carry the same hardware k."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops

DEV = "cuda"
ONE = torch.tensor(1.0).to(torch.float8_e4m3fn).view(torch.uint8)
UNIFORM = 0x7F7F7F7F
flat = torch.full((64,), UNIFORM, dtype=torch.int64).to(torch.int32).to(DEV)


def hit(a8, b8):
    d = ops.mfma_mx_probe(a8, b8, flat, flat)
    torch.cuda.synchronize()
    return d[0, 0].item() > 0.5


def onehot_a(p):
    a = torch.zeros(16, 128, dtype=torch.uint8)
    a[0, p] = ONE
    return a.to(DEV)


def onehot_b(q):
    b = torch.zeros(128, 16, dtype=torch.uint8)
    b[q, 0] = ONE
    return b.to(DEV)


def main():
    A = [onehot_a(p) for p in range(128)]
    B = [onehot_b(q) for q in range(128)]
    pair = {}
    missed = []
    for p in range(128):
        if hit(A[p], B[p]):
            pair[p] = p
        else:
            missed.append(p)
    print("identity pairs:", len(pair), "missed:", len(missed))
    # full scan for a sample of missed positions
    for p in missed[:12] + missed[-4:]:
        qs = [q for q in range(128) if hit(A[p], B[q])]
        print(f"A p={p:3d} pairs with B q={qs}")


if __name__ == "__main__":
    main()
