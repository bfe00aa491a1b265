"""MI355X-native ops: thin Python wrappers over the HIP kernel library.

Every op has two paths:
  * CUDA (= ROCm/HIP) tensors -> the hand-written gfx950 kernel, always.
    There is no eager fallback on GPU; a missing/failed kernel raises.
  * CPU tensors -> a plain fp32 torch reference (used by CPU-only tests and
    as the numerics oracle for the GPU kernels).
"""
from __future__ import annotations

import math

import torch

from . import _lib
from ._lib import AfOpsError, native_loaded
from . import reference as ref


def _on_gpu(*ts: torch.Tensor | None) -> bool:
    return any(t is not None and t.is_cuda for t in ts)


# ---------------------------------------------------------------- rmsnorm
def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5,
            residual: torch.Tensor | None = None):
    """Fused (residual-add) RMSNorm.

    Returns (out, new_residual) when residual is given (residual updated
    in-place to x+residual), else out.
    """
    if not _on_gpu(x):
        return ref.rmsnorm(x, weight, eps, residual)
    T, H = x.shape
    out = torch.empty_like(x)
    rc = _lib.lib().af_rmsnorm(
        _lib.ptr(out), _lib.ptr(residual), _lib.ptr(x), _lib.ptr(residual),
        _lib.ptr(weight), eps, T, H, _lib.cur_stream())
    _lib.check(rc, "af_rmsnorm")
    return (out, residual) if residual is not None else out


# ---------------------------------------------------------------- rope
def rope_table(max_pos: int, head_dim: int, theta: float = 500000.0,
               device="cpu") -> torch.Tensor:
    """Host-precomputed fp32 [max_pos, head_dim] table: [cos | sin] halves."""
    half = head_dim // 2
    inv = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    t = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(t, inv)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float().to(device)


def rope_cache(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
               positions: torch.Tensor, table: torch.Tensor,
               kcache: torch.Tensor, vcache: torch.Tensor,
               slots: torch.Tensor, scale: tuple | None = None) -> None:
    """Fused in-place RoPE on q/k + paged-cache append of k/v.

    q [T, Hq*D], k/v [T, Hk*D] — rows may be strided (views straight out of
    the fused QKV projection); caches [npages, Hk, page, D].
    scale=(ss [T,8], k_dim, eps): rows multiplied by the RMSNorm scalar
    rsqrt(mean+eps) — used when the QKV GEMM ran un-normalized (blas path of
    the fused decode chain; norm weight folded into the projection).
    """
    _, Hk, page, D = kcache.shape
    T = q.shape[0]
    Hq = q.shape[1] // D
    assert q.stride(1) == 1 and k.stride(1) == 1 and v.stride(1) == 1
    if not _on_gpu(q):
        q3 = q.unflatten(-1, (Hq, D))
        k3 = k.unflatten(-1, (Hk, D))
        v3 = v.unflatten(-1, (Hk, D))
        if scale is not None:
            ss, kd, eps = scale
            rstd = torch.rsqrt(ss.view(T, 8).sum(-1) / kd + eps)
            for t3 in (q3, k3, v3):
                t3.mul_(rstd[:, None, None].to(t3.dtype))
        ref.rope(q3, k3, positions, table)
        ref.reshape_and_cache(k3, v3, kcache, vcache, slots)
        return
    if scale is not None:
        ss, kd, eps = scale
        ss_p, inv_k = _lib.ptr(ss), 1.0 / float(kd)
    else:
        ss_p, inv_k, eps = _lib.ptr(None), 0.0, 0.0
    rc = _lib.lib().af_rope_cache(
        _lib.ptr(q), _lib.ptr(k), _lib.ptr(v), _lib.ptr(positions.int()),
        _lib.ptr(table), _lib.ptr(kcache), _lib.ptr(vcache), _lib.ptr(slots),
        ss_p, inv_k, eps,
        T, Hq, Hk, D, q.stride(0), k.stride(0), v.stride(0), page,
        _lib.cur_stream())
    _lib.check(rc, "af_rope_cache")


# ---------------------------------------------------------------- activation
def silu_and_mul(gate_up: torch.Tensor,
                 scale: tuple | None = None) -> torch.Tensor:
    """gate_up [T, 2I] -> silu(gate*r)*(up*r) [T, I] where r is the optional
    per-row RMSNorm scalar from scale=(ss [T,8], k_dim, eps)."""
    T, I2 = gate_up.shape
    if not _on_gpu(gate_up):
        if scale is not None:
            ss, kd, eps = scale
            rstd = torch.rsqrt(ss.view(T, 8).sum(-1) / kd + eps)
            gate_up = (gate_up.float() * rstd[:, None]).to(gate_up.dtype)
        return ref.silu_and_mul(gate_up)
    if scale is not None:
        ss, kd, eps = scale
        ss_p, inv_k = _lib.ptr(ss), 1.0 / float(kd)
    else:
        ss_p, inv_k, eps = _lib.ptr(None), 0.0, 0.0
    out = torch.empty(T, I2 // 2, dtype=gate_up.dtype, device=gate_up.device)
    rc = _lib.lib().af_silu_mul(_lib.ptr(out), _lib.ptr(gate_up), ss_p, inv_k,
                                eps, T, I2 // 2, _lib.cur_stream())
    _lib.check(rc, "af_silu_mul")
    return out


# ---------------------------------------------------------------- cache
def reshape_and_cache(k: torch.Tensor, v: torch.Tensor, kcache: torch.Tensor,
                      vcache: torch.Tensor, slots: torch.Tensor) -> None:
    """Scatter k/v [T,Hk,D] into paged caches [npages,Hk,page,D] at slots [T]."""
    if not _on_gpu(k):
        return ref.reshape_and_cache(k, v, kcache, vcache, slots)
    T = k.shape[0]
    _, Hk, page, D = kcache.shape
    rc = _lib.lib().af_reshape_and_cache(
        _lib.ptr(k), _lib.ptr(v), _lib.ptr(kcache), _lib.ptr(vcache),
        _lib.ptr(slots), T, Hk, D, page, _lib.cur_stream())
    _lib.check(rc, "af_reshape_and_cache")


def embedding(ids: torch.Tensor, table: torch.Tensor,
              ss: torch.Tensor | None = None) -> torch.Tensor:
    if not _on_gpu(table):
        out = table[ids.long()]
        if ss is not None:
            ss.zero_()
            ss.view(-1, 8)[:, 0] = out.float().pow(2).sum(-1)
        return out
    T = ids.numel()
    V, H = table.shape
    out = torch.empty(T, H, dtype=table.dtype, device=table.device)
    rc = _lib.lib().af_embedding(_lib.ptr(out), _lib.ptr(table),
                                 _lib.ptr(ids.int()), _lib.ptr(ss), T, H,
                                 _lib.cur_stream())
    _lib.check(rc, "af_embedding")
    return out


# ---------------------------------------------------------------- attention
def attn_decode(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                block_table: torch.Tensor, seq_lens: torch.Tensor,
                scale: float | None = None, nsplit: int = 1,
                scratch: tuple[torch.Tensor, torch.Tensor] | None = None,
                out: torch.Tensor | None = None,
                window: int = 0) -> torch.Tensor:
    """Paged decode attention. q [B, Hq*D] (strided rows ok) -> out [B, Hq*D].
    window > 0: sliding-window (Mistral) — only the last `window` cached
    tokens are attended."""
    _, Hk, page, D = kcache.shape
    B = q.shape[0]
    Hq = q.shape[1] // D
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if not _on_gpu(q):
        return ref.attn_decode(q.unflatten(-1, (Hq, D)), kcache, vcache,
                               block_table, seq_lens, scale,
                               window=window).flatten(1)
    if out is None:
        out = torch.empty(B, Hq * D, dtype=q.dtype, device=q.device)
    G = Hq // Hk
    if nsplit > 1:
        if scratch is None:
            po = torch.empty(B, Hk, nsplit, G, D, dtype=torch.float32, device=q.device)
            pml = torch.empty(B, Hk, nsplit, G, 2, dtype=torch.float32, device=q.device)
        else:
            po, pml = scratch
    else:
        po = pml = None
    rc = _lib.lib().af_attn_decode(
        _lib.ptr(out), _lib.ptr(po), _lib.ptr(pml), _lib.ptr(q),
        _lib.ptr(kcache), _lib.ptr(vcache), _lib.ptr(block_table),
        _lib.ptr(seq_lens), scale, B, Hq, Hk, D, page, block_table.shape[1],
        nsplit, q.stride(0), window, _lib.cur_stream())
    _lib.check(rc, "af_attn_decode")
    return out


def prefill_tiles(seq_lens: list[int], rows_per_wg: int):
    """Host-side tile map for varlen prefill: (tile_seq[], tile_q0[])."""
    ts, tq = [], []
    for s, ln in enumerate(seq_lens):
        for q0 in range(0, ln, rows_per_wg):
            ts.append(s)
            tq.append(q0)
    return ts, tq


def attn_prefill(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                 block_table: torch.Tensor, q_start: torch.Tensor,
                 cu_seqlens: torch.Tensor, seq_lens: list[int],
                 scale: float | None = None, head_dim: int = 128,
                 window: int = 0) -> torch.Tensor:
    """Paged causal prefill attention: chunk rows q [T, Hq*D] (strided ok)
    attend to cached history + chunk through the block table.  q_start[s] is
    the chunk's absolute start position (0 = full-prompt prefill); the chunk
    itself must already be appended to the cache (rope_cache does this)."""
    D = head_dim
    _, Hk, page, _ = kcache.shape
    T = q.shape[0]
    Hq = q.shape[1] // D
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if not _on_gpu(q):
        return ref.attn_prefill_paged(q.unflatten(-1, (Hq, D)), kcache, vcache,
                                      block_table, q_start, cu_seqlens,
                                      scale, window=window).flatten(1)
    G = Hq // Hk
    rows_per_wg = 16 * max(1, 4 // G)
    ts, tq = prefill_tiles(seq_lens, rows_per_wg)
    dev = q.device
    tile_seq = torch.tensor(ts, dtype=torch.int32, device=dev)
    tile_q0 = torch.tensor(tq, dtype=torch.int32, device=dev)
    out = torch.empty(T, Hq * D, dtype=q.dtype, device=q.device)
    rc = _lib.lib().af_attn_prefill(
        _lib.ptr(out), _lib.ptr(q), _lib.ptr(kcache), _lib.ptr(vcache),
        _lib.ptr(block_table), _lib.ptr(q_start),
        _lib.ptr(cu_seqlens), _lib.ptr(tile_seq), _lib.ptr(tile_q0),
        scale, len(ts), Hq, Hk, D, q.stride(0), page, block_table.shape[1],
        window, _lib.cur_stream())
    _lib.check(rc, "af_attn_prefill")
    return out


# ---------------------------------------------------------------- gemm
def choose_sk(N: int, K: int) -> int:
    """K-split for the skinny GEMM: fill >=1024 workgroups, >=256 K/split."""
    ntiles = N // 64
    sk = max(1, min(16, 1024 // max(1, ntiles)))
    sk = min(sk, max(1, K // 256))
    return sk


def linear_skinny(x: torch.Tensor, w: torch.Tensor, mode: int = 0,
                  residual: torch.Tensor | None = None,
                  sk: int | None = None,
                  scale: tuple | None = None,
                  ss_out: torch.Tensor | None = None) -> torch.Tensor:
    """Decode-path GEMM (M<=256) via the MFMA weight-streaming kernel.

    mode 0: plain; mode 1: +residual (in-place update); mode 2: fused SwiGLU
    (w holds [gate|up] rows, returns [M, N/2]); mode 4: +residual AND
    per-row sum-of-squares stats into ss_out [M,8].
    scale=(ss, eps) applies the RMSNorm row scalar rsqrt(mean(x^2)+eps) on
    the OUTPUT (valid when the norm weight is folded into w — see
    LlamaForCausalLM.fold_norm_weights); x itself stays unnormalized so the
    weight stream keeps its async global_load_lds pipeline.
    """
    M, K = x.shape
    N = w.shape[0]
    if sk is None:
        sk = choose_sk(N, K)
    cols = N // 2 if mode == 2 else N
    out = torch.empty(M, cols, dtype=x.dtype, device=x.device)
    partial = torch.empty(sk, M, N, dtype=torch.float32, device=x.device)
    if scale is not None:
        ss, eps = scale
        ss_p = _lib.ptr(ss)
    else:
        ss_p = _lib.ptr(None)
        eps = 0.0
    rc = _lib.lib().af_gemm_skinny(
        _lib.ptr(out), _lib.ptr(partial), _lib.ptr(residual), _lib.ptr(x),
        _lib.ptr(w), M, N, K, sk, mode, ss_p, eps, _lib.ptr(ss_out),
        _lib.cur_stream())
    _lib.check(rc, "af_gemm_skinny")
    return out


def linear(x: torch.Tensor, w: torch.Tensor, silu_fuse: bool = False) -> torch.Tensor:
    """Linear layer dispatch: hand-written skinny MFMA kernel for decode
    shapes (M<=64), hipBLASLt (torch.matmul) for prefill GEMMs."""
    M, K = x.shape
    N = w.shape[0]
    # policy from measured MI355X data (profiles/kernel_bench_r01_final.log
    # + M=128 sweep): the hand-written path wins where work/CU is scarce —
    # small-M x small-N (qkv/o at M<=64), fused-SwiGLU gate_up at M<=64, and
    # the K=14336 down-projection at 32<M<=128; hipBLASLt wins larger M and
    # the 128k-vocab lm_head.
    ok_shape = (x.is_cuda and N % 64 == 0 and K % 64 == 0
                and x.is_contiguous() and M <= 128)
    if ok_shape and ((M <= 64 and (silu_fuse or (N <= 8192 and K <= 8192)))
                     or (32 < M and K >= 8192 and N <= 8192)):
        return linear_skinny(x, w, mode=2 if silu_fuse else 0)
    y = x @ w.t()
    if silu_fuse:
        return silu_and_mul(y)
    return y


def gemm_bf16(a: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """C[M,N] = a[M,K] @ w[N,K]^T via the hand-written MFMA kernel."""
    if not _on_gpu(a):
        return (a.float() @ w.float().t()).to(a.dtype)
    M, K = a.shape
    N = w.shape[0]
    c = torch.empty(M, N, dtype=a.dtype, device=a.device)
    rc = _lib.lib().af_gemm_bf16(_lib.ptr(c), _lib.ptr(a), _lib.ptr(w),
                                 M, N, K, _lib.cur_stream())
    _lib.check(rc, "af_gemm_bf16")
    return c


def gemm_bf16_ring(a: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Deep-pipelined 256x256 ring-buffered MFMA GEMM (K % 32 == 0):
    4-slot LDS K-chunk ring, counted vmcnt (never drained in-loop),
    setprio around the MFMA cluster (guide §5 8-phase structure)."""
    if not _on_gpu(a):
        return (a.float() @ w.float().t()).to(a.dtype)
    M, K = a.shape
    N = w.shape[0]
    c = torch.empty(M, N, dtype=a.dtype, device=a.device)
    rc = _lib.lib().af_gemm_bf16_ring(_lib.ptr(c), _lib.ptr(a),
                                      _lib.ptr(w), M, N, K,
                                      _lib.cur_stream())
    _lib.check(rc, "af_gemm_bf16_ring")
    return c


def gemm_bf16_q8(a: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Block-quadrant 8-phase deep-pipelined MFMA GEMM (K % 64 == 0,
    K >= 128): 8-slot LDS granule ring, one granule staged per phase two
    K-tiles ahead, counted vmcnt(8), barrier-paired 16-MFMA clusters
    under s_setprio.  1.04-1.15 PF measured (vs 0.83 for gemm.hip)."""
    if not _on_gpu(a):
        return (a.float() @ w.float().t()).to(a.dtype)
    M, K = a.shape
    N = w.shape[0]
    c = torch.empty(M, N, dtype=a.dtype, device=a.device)
    rc = _lib.lib().af_gemm_bf16_q8(_lib.ptr(c), _lib.ptr(a),
                                    _lib.ptr(w), M, N, K,
                                    _lib.cur_stream())
    _lib.check(rc, "af_gemm_bf16_q8")
    return c


def gemm_mxfp8(a8: torch.Tensor, sa: torch.Tensor, w8: torch.Tensor,
               sw: torch.Tensor, M: int | None = None,
               N: int | None = None, K: int | None = None) -> torch.Tensor:
    """C[M,N] bf16 = dequant(a8,sa) @ dequant(w8,sw)^T on the CDNA4
    block-scaled MFMA (MX-fp8: OCP e4m3 + E8M0 per-32-K-block scales,
    quantize with agentfield_amd.quant.quantize_mx).  K % 128 == 0,
    K >= 256."""
    M = M or a8.shape[0]
    K = K or a8.shape[1]
    N = N or w8.shape[0]
    if not _on_gpu(a8):
        from ..quant import dequantize_mx
        return (dequantize_mx(a8, sa) @ dequantize_mx(w8, sw).t()).to(
            torch.bfloat16)
    c = torch.empty(M, N, dtype=torch.bfloat16, device=a8.device)
    rc = _lib.lib().af_gemm_mxfp8(_lib.ptr(c), _lib.ptr(a8), _lib.ptr(sa),
                                  _lib.ptr(w8), _lib.ptr(sw), M, N, K,
                                  _lib.cur_stream())
    _lib.check(rc, "af_gemm_mxfp8")
    return c


# ---------------------------------------------------------------- sampling
class SamplerState:
    """Device-side scratch for graph-capturable sampling."""

    def __init__(self, max_b: int, device, seed: int = 0x5EED):
        self.pv = torch.empty(max_b, 16, dtype=torch.float32, device=device)
        self.pi = torch.empty(max_b, 16, dtype=torch.int32, device=device)
        self.step = torch.zeros(1, dtype=torch.int32, device=device)
        self.seed = seed
        # top-k/top-p scratch (histogram threshold selection)
        self.rmax = torch.empty(max_b, dtype=torch.float32, device=device)
        self.thresh = torch.empty(max_b, dtype=torch.float32, device=device)
        self.hist_n = torch.empty(max_b, 256, dtype=torch.int32, device=device)
        self.hist_m = torch.empty(max_b, 256, dtype=torch.float32, device=device)


def sample(logits: torch.Tensor, temps: torch.Tensor, state: SamplerState,
           out: torch.Tensor | None = None,
           topk: torch.Tensor | None = None,
           topp: torch.Tensor | None = None) -> torch.Tensor:
    """Greedy (temp==0) or Gumbel-max temperature sampling; optional
    top-k/top-p nucleus restriction (sort-free histogram threshold).
    logits [B,V] bf16; topk [B] i32 (0 = off); topp [B] f32 (>=1 = off)."""
    B, V = logits.shape
    if not _on_gpu(logits):
        res = ref.sample_greedy(logits)
        if out is not None:
            out[:B].copy_(res)
            return out
        return res
    if out is None:
        out = torch.empty(B, dtype=torch.int32, device=logits.device)
    if topk is not None or topp is not None:
        if topk is None:
            topk = torch.zeros(B, dtype=torch.int32, device=logits.device)
        if topp is None:
            topp = torch.ones(B, dtype=torch.float32, device=logits.device)
        rc = _lib.lib().af_sample_topkp(
            _lib.ptr(out), _lib.ptr(state.pv), _lib.ptr(state.pi),
            _lib.ptr(state.rmax), _lib.ptr(state.thresh),
            _lib.ptr(state.hist_n), _lib.ptr(state.hist_m), _lib.ptr(logits),
            _lib.ptr(temps), _lib.ptr(topk), _lib.ptr(topp),
            _lib.ptr(state.step), state.seed, B, V, _lib.cur_stream())
        _lib.check(rc, "af_sample_topkp")
        return out
    rc = _lib.lib().af_sample(
        _lib.ptr(out), _lib.ptr(state.pv), _lib.ptr(state.pi), _lib.ptr(logits),
        _lib.ptr(temps), _lib.ptr(state.step), state.seed, B, V, _lib.cur_stream())
    _lib.check(rc, "af_sample")
    return out


def gather_rows(x: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    if not _on_gpu(x):
        return x[idx.long()]
    B = idx.numel()
    H = x.shape[1]
    out = torch.empty(B, H, dtype=x.dtype, device=x.device)
    rc = _lib.lib().af_gather_rows(_lib.ptr(out), _lib.ptr(x), _lib.ptr(idx.int()),
                                   B, H, _lib.cur_stream())
    _lib.check(rc, "af_gather_rows")
    return out


def mfma_mx_probe(a8: torch.Tensor, b8: torch.Tensor, sa: torch.Tensor,
                  sb: torch.Tensor) -> torch.Tensor:
    """Single-wave 16x16x128 MX-fp8 scaled MFMA with the assumed layouts.
    a8 [16,128] u8 (e4m3 bytes), b8 [128,16] u8, sa/sb [16,4] u8 (E8M0
    per 32-element K-block)."""
    d = torch.empty(16, 16, dtype=torch.float32, device=a8.device)
    rc = _lib.lib().af_mfma_mx_probe(_lib.ptr(d), _lib.ptr(a8), _lib.ptr(b8),
                                     _lib.ptr(sa), _lib.ptr(sb),
                                     _lib.cur_stream())
    _lib.check(rc, "af_mfma_mx_probe")
    return d


def mfma_probe(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Single-wave 16x16x32 MFMA with the assumed layouts (GPU layout test)."""
    d = torch.empty(16, 16, dtype=torch.float32, device=a.device)
    rc = _lib.lib().af_mfma_probe(_lib.ptr(d), _lib.ptr(a), _lib.ptr(b),
                                  _lib.cur_stream())
    _lib.check(rc, "af_mfma_probe")
    return d
