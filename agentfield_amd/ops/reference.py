"""Plain-torch fp32 reference implementations.

These are the numerics oracle for the HIP kernels (GPU tests compare the
kernel against these at fp32) and the execution path for CPU-only tests.
They intentionally mirror the kernel semantics, not any external library.
"""
from __future__ import annotations

import torch


def rmsnorm(x, weight, eps=1e-5, residual=None):
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    out = (xf * rstd * weight.float()).to(x.dtype)
    if residual is not None:
        h = xf.to(x.dtype)
        residual.copy_(h)
        return out, residual
    return out


def rope(q, k, positions, table):
    """In-place rotate-half RoPE. q [T,Hq,D], k [T,Hk,D], table [P, D]=[cos|sin]."""
    D = q.shape[-1]
    half = D // 2
    cos = table[positions.long(), :half].unsqueeze(1).float()  # [T,1,half]
    sin = table[positions.long(), half:].unsqueeze(1).float()
    for t in (q, k):
        x1 = t[..., :half].float()
        x2 = t[..., half:].float()
        t[..., :half] = (x1 * cos - x2 * sin).to(t.dtype)
        t[..., half:] = (x2 * cos + x1 * sin).to(t.dtype)


def silu_and_mul(gate_up):
    I = gate_up.shape[-1] // 2
    g = gate_up[..., :I].float()
    u = gate_up[..., I:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def reshape_and_cache(k, v, kcache, vcache, slots):
    _, Hk, page, D = kcache.shape
    for t in range(k.shape[0]):
        s = int(slots[t])
        if s < 0:
            continue
        p, o = divmod(s, page)
        kcache[p, :, o] = k[t]
        vcache[p, :, o] = v[t]


def attn_decode(q, kcache, vcache, block_table, seq_lens, scale, window=0):
    """q [B,Hq,D] -> out [B,Hq,D], gathering K/V from pages.
    window > 0: only the last `window` tokens are attended."""
    B, Hq, D = q.shape
    _, Hk, page, _ = kcache.shape
    G = Hq // Hk
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        w0 = max(0, L - window) if window > 0 else 0
        pages = block_table[b, : (L + page - 1) // page].long()
        ks = kcache[pages].permute(1, 0, 2, 3).reshape(Hk, -1, D)[:, w0:L].float()
        vs = vcache[pages].permute(1, 0, 2, 3).reshape(Hk, -1, D)[:, w0:L].float()
        for h in range(Hq):
            kh = ks[h // G]
            vh = vs[h // G]
            s = (kh @ (q[b, h].float() * scale)).softmax(-1)
            out[b, h] = (s @ vh).to(q.dtype)
    return out


def attn_prefill(q, k, v, cu_seqlens, scale):
    """Causal varlen attention. q [T,Hq,D], k/v [T,Hk,D]."""
    T, Hq, D = q.shape
    Hk = k.shape[1]
    G = Hq // Hk
    out = torch.empty_like(q)
    cs = cu_seqlens.tolist()
    for i in range(len(cs) - 1):
        s0, s1 = cs[i], cs[i + 1]
        L = s1 - s0
        qf = q[s0:s1].float().permute(1, 0, 2)            # [Hq, L, D]
        kf = k[s0:s1].float().permute(1, 0, 2)            # [Hk, L, D]
        vf = v[s0:s1].float().permute(1, 0, 2)
        kf = kf.repeat_interleave(G, dim=0)
        vf = vf.repeat_interleave(G, dim=0)
        att = (qf @ kf.transpose(-1, -2)) * scale         # [Hq, L, L]
        mask = torch.triu(torch.ones(L, L, dtype=torch.bool, device=q.device), 1)
        att = att.masked_fill(mask, float("-inf")).softmax(-1)
        out[s0:s1] = (att @ vf).permute(1, 0, 2).to(q.dtype)
    return out


def attn_prefill_paged(q, kcache, vcache, block_table, q_start, cu_seqlens,
                       scale, window=0):
    """Causal attention of chunk rows against cached history + chunk.
    q [T,Hq,D]; caches [npages,Hk,page,D]; q_start[s] = absolute position of
    the chunk's first row."""
    T, Hq, D = q.shape
    _, Hk, page, _ = kcache.shape
    G = Hq // Hk
    out = torch.empty_like(q)
    cs = cu_seqlens.tolist()
    for i in range(len(cs) - 1):
        s0, s1 = cs[i], cs[i + 1]
        L = s1 - s0
        hist = int(q_start[i])
        total = hist + L
        npg = (total + page - 1) // page
        pages = block_table[i, :npg].long()
        ks = kcache[pages].permute(1, 0, 2, 3).reshape(Hk, -1, D)[:, :total].float()
        vs = vcache[pages].permute(1, 0, 2, 3).reshape(Hk, -1, D)[:, :total].float()
        qf = q[s0:s1].float().permute(1, 0, 2)            # [Hq, L, D]
        kf = ks.repeat_interleave(G, dim=0)
        vf = vs.repeat_interleave(G, dim=0)
        att = (qf @ kf.transpose(-1, -2)) * scale         # [Hq, L, total]
        qpos = torch.arange(hist, hist + L).unsqueeze(1)
        kpos = torch.arange(total).unsqueeze(0)
        bad = kpos > qpos
        if window > 0:
            bad |= kpos <= qpos - window
        att = att.masked_fill(bad, float("-inf")).softmax(-1)
        out[s0:s1] = (att @ vf).permute(1, 0, 2).to(q.dtype)
    return out


def sample_greedy(logits):
    return logits.float().argmax(-1).int()
