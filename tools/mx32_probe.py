#!/usr/bin/env python3
"""Map the 32x32x64 f8f6f4 scaled-MFMA layouts.

FINDING (measured): both natural data-layout hypotheses produce NaN
even with uniform x1 scales and all-1.0 e4m3 data, so the 32x32x64
operand encoding differs fundamentally from the 16x16x128 one (which
is mapped and shipped in gemm_mxfp8.hip).  Parked: the 16x16 kernel is
correct at 862 TF; revisit with an ISA-level decode if the 32x32 shape
is ever needed (it would halve operand registers per MFMA)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from agentfield_amd.ops import _lib
from agentfield_amd.quant import dequantize_mx, quantize_mx

DEV = "cuda"


def run(a8, b8, sa64, sb64, layout):
    d = torch.empty(32, 32, dtype=torch.float32, device=DEV)
    _lib.check(_lib.lib().af_mfma_mx32_probe(
        _lib.ptr(d), _lib.ptr(a8.to(DEV)), _lib.ptr(b8.to(DEV)),
        _lib.ptr(sa64.to(torch.int32).to(DEV)),
        _lib.ptr(sb64.to(torch.int32).to(DEV)), layout,
        _lib.cur_stream()), "mx32")
    torch.cuda.synchronize()
    return d.cpu()


def lanes_from(s8):
    # scale lane (m = lane&31, s = lane>>5) -> byte for (row m, block s)
    out = torch.zeros(64, dtype=torch.int64)
    for g in range(2):
        for r in range(32):
            out[g * 32 + r] = int(s8[r, g])
    return out


def main():
    torch.manual_seed(11)
    A = torch.randn(32, 64) * 2.0
    B = torch.randn(64, 32) * 2.0
    A[5] *= 31.0
    B[:, 9] *= 0.03
    a8, sa = quantize_mx(A)
    b8, sb = quantize_mx(B.t().contiguous())
    want = dequantize_mx(a8, sa) @ dequantize_mx(b8, sb).t()
    for layout in (0, 1):
        d = run(a8, b8.t().contiguous(), lanes_from(sa), lanes_from(sb),
                layout)
        rel = (d - want).abs().max().item() / want.abs().max().item()
        print(f"layout={layout}: rel err {rel:.6f}")
    # scale perturbation with ones data under layout 1
    ONE = torch.tensor(1.0).to(torch.float8_e4m3fn).view(torch.uint8)
    a1 = ONE.repeat(32, 64)
    b1 = ONE.repeat(64, 32)
    flat = torch.full((64,), 127, dtype=torch.int64)
    for layout in (0, 1):
        d0 = run(a1, b1, flat, flat, layout)
        print(f"layout={layout} uniform D[0,0]={d0[0,0].item()} (64 wanted)",
              "uniform" if bool((d0 == d0[0, 0]).all()) else "NONUNIFORM")
        for L in (0, 32, 17):
            sa2 = flat.clone()
            sa2[L] = 128  # x2 in byte0
            d = run(a1, b1, sa2, flat, layout)
            delta = d - d0
            rows = [int(r) for r in range(32) if delta[r].abs().max() > 0.5]
            mx = delta.max().item()
            print(f"  perturb SA lane {L}: rows {rows} "
                  f"+{mx if mx != mx else round(mx)}")


if __name__ == "__main__":
    main()
