"""MXFP8 quantization (OCP e4m3 + E8M0 per-32-block scales) for the
CDNA4 scaled MFMA path (mfma_scale_*_f8f6f4 — hardware-fused
dequant+matmul at 2x the bf16 rate).

Weight-only or W8A8 quantized serving is opt-in; the bf16 path stays
the default (BASELINE dtype).
"""
from __future__ import annotations

import torch

BLOCK = 32          # MX scale granularity along K
FP8_MAX = 448.0     # e4m3fn max magnitude


def quantize_mx(x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """[R, K] float -> (e4m3 bytes [R, K] u8, E8M0 scales [R, K/32] u8).

    Per-block power-of-two scale: e = ceil(log2(absmax / FP8_MAX)), value
    stored as x / 2^e in e4m3, scale byte = e + 127 (E8M0 bias).  Exact
    powers of two round-trip losslessly; absmax maps inside fp8 range.
    """
    R, K = x.shape
    assert K % BLOCK == 0, "K must be a multiple of the MX block (32)"
    xf = x.float().reshape(R, K // BLOCK, BLOCK)
    amax = xf.abs().amax(dim=-1).clamp_min(1e-30)
    e = torch.ceil(torch.log2(amax / FP8_MAX)).clamp(-127, 127)
    scale = torch.pow(2.0, e)
    q = (xf / scale.unsqueeze(-1)).clamp(-FP8_MAX, FP8_MAX)
    q8 = q.to(torch.float8_e4m3fn).view(torch.uint8).reshape(R, K)
    s8 = (e + 127).to(torch.uint8)
    return q8, s8


def dequantize_mx(q8: torch.Tensor, s8: torch.Tensor) -> torch.Tensor:
    """Inverse of quantize_mx (fp32)."""
    R, K = q8.shape
    v = q8.view(torch.float8_e4m3fn).float().reshape(R, K // BLOCK, BLOCK)
    e = s8.float() - 127.0
    return (v * torch.pow(2.0, e).unsqueeze(-1)).reshape(R, K)
