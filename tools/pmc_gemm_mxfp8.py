#!/usr/bin/env python3
"""PMC profiling target for the MX-fp8 GEMM only."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops
from agentfield_amd.quant import quantize_mx


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    torch.manual_seed(0)
    a8, sa = quantize_mx(torch.randn(n, n) * 0.3)
    w8, sw = quantize_mx(torch.randn(n, n) * 0.3)
    a8, sa, w8, sw = (t.cuda() for t in (a8, sa, w8, sw))
    for _ in range(5):
        ops.gemm_mxfp8(a8, sa, w8, sw)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
