#!/usr/bin/env python3
"""PMC profiling target for the q8 GEMM only (host-generated data)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    torch.manual_seed(0)
    a = (torch.randn(n, n) * 0.3).bfloat16().cuda()
    w = (torch.randn(n, n) * 0.3).bfloat16().cuda()
    for _ in range(5):
        ops.gemm_bf16_q8(a, w)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
