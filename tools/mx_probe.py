#!/usr/bin/env python3
"""Validate the MX-fp8 scaled-MFMA (16x16x128) layouts + E8M0 scale
semantics on gfx950: D must equal dequant(A) @ dequant(B) exactly.

Mapped empirically (mx_probe2/3/4.py):
  * scale: byte 0 (op_sel 0) of scale lane (row, s = lane>>4) covers hw
    K-block s; E8M0 (2^(byte-127)).
  * data: lane (row, g) byte j -> hw k = 32*(2*(j>>4)+(g>>1))
    + 16*(g&1) + (j&15): a lane spans TWO hw blocks.  Loading with this
    mapping makes memory-k == hw-k so per-32-block scales line up (the
    naive lane-g=block-g guess cancels under uniform scales only).
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd import ops
from agentfield_amd.quant import dequantize_mx, quantize_mx

DEV = "cuda"


def lanes_from(s8: torch.Tensor) -> torch.Tensor:
    """[16,4] E8M0 bytes -> per-lane i32 (byte0), lane = g*16 + row."""
    out = torch.zeros(64, dtype=torch.int32)
    for g in range(4):
        for r in range(16):
            out[g * 16 + r] = int(s8[r, g])
    return out


def main():
    torch.manual_seed(7)
    A = torch.randn(16, 128) * 2.0
    B = torch.randn(128, 16) * 2.0
    A[3] *= 37.0
    B[:, 5] *= 0.01
    a8, sa = quantize_mx(A)
    b8, sb = quantize_mx(B.t().contiguous())
    want = dequantize_mx(a8, sa) @ dequantize_mx(b8, sb).t()
    d = ops.mfma_mx_probe(a8.to(DEV), b8.t().contiguous().to(DEV),
                          lanes_from(sa).to(DEV), lanes_from(sb).to(DEV))
    torch.cuda.synchronize()
    rel = (d.cpu() - want).abs().max().item() / want.abs().max().item()
    print("full check rel err", rel)
    assert rel < 1e-4, "layout or scale semantics mismatch"
    print("MX PROBE OK")


if __name__ == "__main__":
    main()
