"""GPU numerics tests: every HIP kernel vs the plain-torch fp32 reference.

All inputs are asymmetric random (guide G9: symmetric inputs can't detect
operand/output transposes).
"""
import math

import pytest
import torch

from agentfield_amd import ops
from agentfield_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

DEV = "cuda"


def rnd(*shape, dtype=torch.bfloat16, seed=None, scale=1.0):
    if seed is not None:
        torch.manual_seed(seed)
    return (torch.randn(*shape, dtype=torch.float32, device=DEV) * scale).to(dtype)


def close(a, b, atol=2e-2, rtol=2e-2):
    a, b = a.float().cpu(), b.float().cpu()
    ok = torch.allclose(a, b, atol=atol, rtol=rtol)
    if not ok:
        d = (a - b).abs()
        print("max abs diff", d.max().item(), "mean", d.mean().item())
    return ok


def test_native_lib_loads():
    assert ops.native_loaded(), "libafops.so must load on a GPU box"


def test_axpy_plumbing():
    from agentfield_amd.ops import _lib
    y = torch.ones(100, dtype=torch.float32, device=DEV)
    x = torch.full((100,), 2.0, device=DEV)
    rc = _lib.lib().af_axpy(_lib.ptr(y), _lib.ptr(x), 3.0, 100, _lib.cur_stream())
    _lib.check(rc, "af_axpy")
    torch.cuda.synchronize()
    assert torch.allclose(y.cpu(), torch.full((100,), 7.0))


def test_mfma_probe_layouts():
    """Assumed A/B/C fragment layouts for mfma_f32_16x16x32_bf16."""
    a = rnd(16, 32, seed=1)
    b = rnd(32, 16, seed=2)
    d = ops.mfma_probe(a, b)
    torch.cuda.synchronize()
    want = a.float() @ b.float()
    assert close(d, want, atol=1e-1), "MFMA fragment layout mismatch"


def test_rmsnorm():
    x = rnd(33, 4096, seed=3)
    w = rnd(4096, seed=4)
    out = ops.rmsnorm(x, w, 1e-5)
    torch.cuda.synchronize()
    assert close(out, ref.rmsnorm(x.cpu(), w.cpu(), 1e-5))


def test_rmsnorm_fused_residual():
    x = rnd(17, 512, seed=5)
    w = rnd(512, seed=6)
    res = rnd(17, 512, seed=7)
    res_ref = res.cpu().clone()
    out, new_res = ops.rmsnorm(x, w, 1e-5, residual=res)
    torch.cuda.synchronize()
    out_ref, res_ref = ref.rmsnorm(x.cpu(), w.cpu(), 1e-5, residual=res_ref)
    assert close(out, out_ref)
    assert close(new_res, res_ref)


def test_rope_cache_fused_strided():
    """Fused RoPE+cache on strided rows (views of a fused QKV buffer)."""
    T, Hq, Hk, D, page, npages = 9, 4, 2, 128, 4, 8
    qkv = rnd(T, (Hq + 2 * Hk) * D, seed=8)
    qkv_c = qkv.cpu().clone()
    q = qkv[:, :Hq * D]
    k = qkv[:, Hq * D:(Hq + Hk) * D]
    v = qkv[:, (Hq + Hk) * D:]
    pos = torch.randint(0, 100, (T,), dtype=torch.int32, device=DEV)
    tab = ops.rope_table(128, D, device=DEV)
    kc = torch.zeros(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    slots = torch.randperm(npages * page, device=DEV)[:T].to(torch.int64)
    ops.rope_cache(q, k, v, pos, tab, kc, vc, slots)
    torch.cuda.synchronize()
    # CPU reference on the same strided layout
    qr = qkv_c[:, :Hq * D]
    kr = qkv_c[:, Hq * D:(Hq + Hk) * D]
    vr = qkv_c[:, (Hq + Hk) * D:]
    kcr = torch.zeros_like(kc, device="cpu")
    vcr = torch.zeros_like(vc, device="cpu")
    ops.rope_cache(qr, kr, vr, pos.cpu(), tab.cpu(), kcr, vcr, slots.cpu())
    assert close(q, qr) and close(k, kr)
    assert close(kc, kcr, atol=0) and close(vc, vcr, atol=0)


def test_silu_mul():
    x = rnd(31, 1024, seed=10)
    out = ops.silu_and_mul(x)
    torch.cuda.synchronize()
    assert close(out, ref.silu_and_mul(x.cpu()))


def test_reshape_and_cache():
    T, Hk, D, page, npages = 11, 2, 128, 4, 8
    k, v = rnd(T, Hk, D, seed=11), rnd(T, Hk, D, seed=12)
    kc = torch.zeros(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    slots = torch.randperm(npages * page, device=DEV)[:T].to(torch.int64)
    ops.reshape_and_cache(k, v, kc, vc, slots)
    torch.cuda.synchronize()
    kc_ref = torch.zeros_like(kc, device="cpu")
    vc_ref = torch.zeros_like(vc, device="cpu")
    ref.reshape_and_cache(k.cpu(), v.cpu(), kc_ref, vc_ref, slots.cpu())
    assert close(kc, kc_ref, atol=0) and close(vc, vc_ref, atol=0)


def _decode_case(B, Hq, Hk, lens, page=16, nsplit=1, seed=13, window=0):
    D = 128
    torch.manual_seed(seed)
    maxp = (max(lens) + page - 1) // page
    npages = 1 + sum((l + page - 1) // page for l in lens)
    q = rnd(B, Hq * D, seed=seed)
    kc = rnd(npages, Hk, page, D, seed=seed + 1)
    vc = rnd(npages, Hk, page, D, seed=seed + 2)
    perm = torch.randperm(npages - 1) + 1
    bt = torch.zeros(B, maxp, dtype=torch.int32)
    at = 0
    for b, l in enumerate(lens):
        n = (l + page - 1) // page
        bt[b, :n] = perm[at:at + n]
        at += n
    bt = bt.to(DEV)
    lens_t = torch.tensor(lens, dtype=torch.int32, device=DEV)
    out = ops.attn_decode(q, kc, vc, bt, lens_t, nsplit=nsplit, window=window)
    torch.cuda.synchronize()
    want = ref.attn_decode(q.cpu().unflatten(-1, (Hq, D)), kc.cpu(), vc.cpu(),
                           bt.cpu(), lens_t.cpu(), 1.0 / math.sqrt(D),
                           window=window).flatten(1)
    assert close(out, want), \
        f"decode mismatch B={B} G={Hq//Hk} nsplit={nsplit} win={window}"


def test_attn_decode_g4():
    _decode_case(3, 8, 2, [5, 33, 17])


def test_attn_decode_g1_g8():
    _decode_case(2, 4, 4, [20, 64], seed=20)
    _decode_case(2, 8, 1, [31, 7], seed=21)


def test_attn_decode_split():
    _decode_case(2, 8, 2, [100, 230], nsplit=4, seed=22)
    _decode_case(1, 4, 1, [511], nsplit=8, seed=23)


def _prefill_case(Hq, Hk, lens, seed=30, page=16, window=0):
    """Full-prompt paged prefill vs the contiguous fp32 reference."""
    D = 128
    T = sum(lens)
    q = rnd(T, Hq * D, seed=seed)
    k = rnd(T, Hk * D, seed=seed + 1)
    v = rnd(T, Hk * D, seed=seed + 2)
    cu = [0]
    for l in lens:
        cu.append(cu[-1] + l)
    cu_t = torch.tensor(cu, dtype=torch.int32, device=DEV)
    # scatter k/v into a paged cache with per-seq page runs
    maxp = (max(lens) + page - 1) // page
    npages = 1 + len(lens) * maxp
    kc = torch.zeros(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    bt = torch.zeros(len(lens), maxp, dtype=torch.int32, device=DEV)
    slots = []
    nextpage = 1
    for sidx, l in enumerate(lens):
        n = (l + page - 1) // page
        bt[sidx, :n] = torch.arange(nextpage, nextpage + n, dtype=torch.int32)
        slots.extend(nextpage * page + i for i in range(l))
        nextpage += n
    slots_t = torch.tensor(slots, dtype=torch.int64, device=DEV)
    ops.reshape_and_cache(k.unflatten(-1, (Hk, D)), v.unflatten(-1, (Hk, D)),
                          kc, vc, slots_t)
    qstart = torch.zeros(len(lens), dtype=torch.int32, device=DEV)
    out = ops.attn_prefill(q, kc, vc, bt, qstart, cu_t, lens, window=window)
    torch.cuda.synchronize()
    want = ref.attn_prefill_paged(q.cpu().unflatten(-1, (Hq, D)), kc.cpu(),
                                  vc.cpu(), bt.cpu(), qstart.cpu(), cu_t.cpu(),
                                  1.0 / math.sqrt(D),
                                  window=window).flatten(1)
    assert close(out, want), \
        f"prefill mismatch G={Hq//Hk} lens={lens} win={window}"


def test_attn_prefill_g4():
    _prefill_case(8, 2, [48, 17, 100])


def test_attn_prefill_g1():
    _prefill_case(2, 2, [33, 64], seed=40)


def test_attn_prefill_g2_g8():
    _prefill_case(4, 2, [40, 23], seed=50)
    _prefill_case(8, 1, [129], seed=60)


def test_attn_prefill_long():
    _prefill_case(8, 2, [1024], seed=70)


def test_attn_decode_sliding_window():
    # window < L (band active), window >= L (equivalent to full causal)
    _decode_case(3, 8, 2, [5, 33, 200], seed=91, window=48)
    _decode_case(2, 8, 2, [100, 230], nsplit=4, seed=92, window=64)
    _decode_case(2, 4, 4, [20, 64], seed=93, window=128)


def test_attn_prefill_sliding_window():
    _prefill_case(8, 2, [48, 170], seed=94, window=32)
    _prefill_case(4, 2, [300], seed=95, window=96)
    _prefill_case(8, 2, [40], seed=96, window=64)  # win >= len: full causal


def test_attn_prefill_chunked_history():
    """Second chunk attends to cached history: must match the one-shot
    paged reference over the whole prompt."""
    D, Hq, Hk, page = 128, 8, 2, 16
    total, hist = 48, 29
    chunk = total - hist
    q_all = rnd(total, Hq * D, seed=80)
    k_all = rnd(total, Hk * D, seed=81)
    v_all = rnd(total, Hk * D, seed=82)
    npages = 1 + (total + page - 1) // page
    kc = torch.zeros(npages, Hk, page, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    bt = torch.arange(1, npages, dtype=torch.int32, device=DEV)[None, :]
    slots = torch.arange(page, page + total, dtype=torch.int64, device=DEV)
    ops.reshape_and_cache(k_all.unflatten(-1, (Hk, D)),
                          v_all.unflatten(-1, (Hk, D)), kc, vc, slots)
    q2 = q_all[hist:]
    cu = torch.tensor([0, chunk], dtype=torch.int32, device=DEV)
    qstart = torch.tensor([hist], dtype=torch.int32, device=DEV)
    out = ops.attn_prefill(q2, kc, vc, bt, qstart, cu, [chunk])
    torch.cuda.synchronize()
    want = ref.attn_prefill_paged(
        q2.cpu().unflatten(-1, (Hq, D)), kc.cpu(), vc.cpu(), bt.cpu(),
        qstart.cpu(), cu.cpu(), 1.0 / math.sqrt(D)).flatten(1)
    assert close(out, want), "chunked prefill history mismatch"


def test_gemm():
    for (M, N, K) in [(128, 128, 64), (256, 512, 256), (100, 300, 128),
                      (512, 4096, 4096)]:
        a = rnd(M, K, seed=M + N, scale=0.5)
        w = rnd(N, K, seed=M + N + 1, scale=0.5)
        c = ops.gemm_bf16(a, w)
        torch.cuda.synchronize()
        want = (a.float() @ w.float().t())
        # bf16 accumulation tolerance scales with K
        tol = 0.1 + 0.02 * math.sqrt(K)
        assert close(c, want, atol=tol, rtol=5e-2), f"gemm {M}x{N}x{K}"


def test_gemm_ring():
    # deep-pipelined ring variant (K % 32 == 0): exact tiles, ragged M/N,
    # K shorter than the ring depth, and a real model shape
    for (M, N, K) in [(256, 256, 32), (256, 256, 96), (512, 512, 512),
                      (300, 700, 160), (512, 4096, 4096)]:
        a = rnd(M, K, seed=M + N + 11, scale=0.5)
        w = rnd(N, K, seed=M + N + 12, scale=0.5)
        c = ops.gemm_bf16_ring(a, w)
        torch.cuda.synchronize()
        want = (a.float() @ w.float().t())
        tol = 0.1 + 0.02 * math.sqrt(K)
        assert close(c, want, atol=tol, rtol=5e-2), f"gemm_ring {M}x{N}x{K}"
        if K % 64 == 0 and K >= 128:
            c8 = ops.gemm_bf16_q8(a, w)
            torch.cuda.synchronize()
            assert close(c8, want, atol=tol, rtol=5e-2), f"gemm_q8 {M}x{N}x{K}"


def test_gemm_skinny():
    for M in (1, 7, 16, 33, 64, 96, 128, 192, 256):
        for (N, K) in [(6144, 4096), (4096, 14336), (128256, 4096)]:
            a = rnd(M, K, seed=M + N, scale=0.3)
            w = rnd(N, K, seed=M + N + 1, scale=0.3)
            c = ops.linear_skinny(a, w)
            torch.cuda.synchronize()
            want = a.float() @ w.float().t()
            tol = 0.1 + 0.02 * math.sqrt(K)
            assert close(c, want, atol=tol, rtol=5e-2), f"skinny {M}x{N}x{K}"


def test_gemm_skinny_silu_fused():
    M, I, K = 16, 14336, 4096
    a = rnd(M, K, seed=5, scale=0.3)
    w = rnd(2 * I, K, seed=6, scale=0.3)
    c = ops.linear_skinny(a, w, mode=2)
    torch.cuda.synchronize()
    gu = (a.float() @ w.float().t())
    want = torch.nn.functional.silu(gu[:, :I]) * gu[:, I:]
    assert close(c, want, atol=3.0, rtol=8e-2)


def test_gemm_skinny_residual_mode():
    M, N, K = 16, 4096, 4096
    a = rnd(M, K, seed=7, scale=0.3)
    w = rnd(N, K, seed=8, scale=0.3)
    res = rnd(M, N, seed=9)
    res_ref = res.float().cpu().clone()
    c = ops.linear_skinny(a, w, mode=1, residual=res)
    torch.cuda.synchronize()
    want = a.float() @ w.float().t() + res_ref.to(DEV)
    tol = 0.1 + 0.02 * math.sqrt(K)
    assert close(c, want, atol=tol, rtol=5e-2)
    assert close(res, want, atol=tol, rtol=5e-2)  # residual updated in place


def test_sample_greedy_matches_argmax():
    torch.manual_seed(99)
    B, V = 5, 128256
    logits = rnd(B, V, seed=99)
    st = ops.SamplerState(B, DEV)
    temps = torch.zeros(B, dtype=torch.float32, device=DEV)
    toks = ops.sample(logits, temps, st)
    torch.cuda.synchronize()
    assert toks.cpu().tolist() == logits.float().argmax(-1).cpu().tolist()


def test_sample_temperature_varies_with_step():
    B, V = 2, 1000
    logits = torch.zeros(B, V, dtype=torch.bfloat16, device=DEV)
    st = ops.SamplerState(B, DEV)
    temps = torch.full((B,), 1.0, dtype=torch.float32, device=DEV)
    draws = set()
    for _ in range(8):
        t = ops.sample(logits, temps, st)
        torch.cuda.synchronize()
        draws.add(tuple(t.cpu().tolist()))
    assert len(draws) > 1, "uniform logits should sample different tokens per step"


def test_embedding_and_gather():
    tab = rnd(64, 256, seed=101)
    ids = torch.randint(0, 64, (10,), dtype=torch.int32, device=DEV)
    out = ops.embedding(ids, tab)
    torch.cuda.synchronize()
    assert close(out, tab.cpu()[ids.cpu().long()], atol=0)
    rows = torch.tensor([3, 7, 9], dtype=torch.int32, device=DEV)
    g = ops.gather_rows(out, rows)
    torch.cuda.synchronize()
    assert close(g, out.cpu()[rows.cpu().long()], atol=0)


def test_sample_topk_topp():
    torch.manual_seed(7)
    B, V = 4, 50000
    logits = rnd(B, V, seed=7, scale=3.0)
    st = ops.SamplerState(B, DEV)
    temps = torch.full((B,), 1.0, dtype=torch.float32, device=DEV)
    # top_k=1 must equal argmax regardless of temperature
    topk = torch.ones(B, dtype=torch.int32, device=DEV)
    topp = torch.ones(B, dtype=torch.float32, device=DEV)
    t = ops.sample(logits, temps, st, topk=topk, topp=topp)
    torch.cuda.synchronize()
    assert t.cpu().tolist() == logits.float().argmax(-1).cpu().tolist()
    # top_k=8: draws must respect the histogram-threshold contract — at or
    # above the lower edge of the bin holding the 8th-largest logit
    # (span 20, 256 bins => granularity 20/256)
    topk8 = torch.full((B,), 8, dtype=torch.int32, device=DEV)
    lf = logits.float().cpu()
    k8 = lf.topk(8, dim=-1).values[:, -1]
    mx = lf.max(-1).values
    gran = 20.0 / 256
    floor8 = torch.floor((k8 - (mx - 20.0)) / gran) * gran + (mx - 20.0)
    for _ in range(10):
        t = ops.sample(logits, temps, st, topk=topk8, topp=topp)
        torch.cuda.synchronize()
        for b in range(B):
            drawn = lf[b, int(t[b])]
            assert drawn >= floor8[b] - 1e-3, \
                f"draw below top-8 threshold (b={b}): {drawn} < {floor8[b]}"
    # tiny top_p: survivors all come from the top histogram bin
    topp_tiny = torch.full((B,), 1e-6, dtype=torch.float32, device=DEV)
    topk0 = torch.zeros(B, dtype=torch.int32, device=DEV)
    for _ in range(5):
        t = ops.sample(logits, temps, st, topk=topk0, topp=topp_tiny)
        torch.cuda.synchronize()
        for b in range(B):
            assert lf[b, int(t[b])] >= mx[b] - gran - 1e-3


def test_engine_topk_sampling_runs():
    from agentfield_amd.engine import LLMEngine, SamplingParams
    from agentfield_amd.models import CONFIGS
    eng = LLMEngine(CONFIGS["tiny"], device="cuda", page_size=4, num_pages=128,
                    max_num_seqs=4, enable_graphs=True, seed=2)
    outs = eng.generate([[1, 2, 3]], SamplingParams(
        max_tokens=6, temperature=0.9, top_k=10, top_p=0.9, ignore_eos=True))
    assert len(outs[0]) == 6


def test_gemm_mxfp8():
    """MX-fp8 block-scaled GEMM vs the exact dequantized fp32 product
    (hardware dequant is exact for power-of-two E8M0 scales; only fp32
    accumulation order differs)."""
    from agentfield_amd.quant import dequantize_mx, quantize_mx
    for (M, N, K) in [(256, 256, 256), (512, 512, 1024), (300, 256, 512)]:
        torch.manual_seed(M + K)
        A = torch.randn(M, K) * 2.0
        W = torch.randn(N, K) * 2.0
        A[min(3, M - 1)] *= 29.0   # exercise non-uniform scales
        W[min(7, N - 1)] *= 0.02
        a8, sa = quantize_mx(A)
        w8, sw = quantize_mx(W)
        want = (dequantize_mx(a8, sa) @ dequantize_mx(w8, sw).t())
        c = ops.gemm_mxfp8(a8.to(DEV), sa.to(DEV), w8.to(DEV), sw.to(DEV),
                           M=M, N=N, K=K)
        torch.cuda.synchronize()
        rel = (c.float().cpu() - want).abs().max().item() / \
            want.abs().max().item()
        assert rel < 5e-3, (M, N, K, rel)


def test_mxfp8_mlp_layer_accuracy():
    """Quantized-weight MLP block (gate_up + silu_mul + down) through the
    MX-fp8 GEMM: end-to-end relative error stays in the e4m3 range
    (~2^-3 worst-case per-element) vs the bf16 reference."""
    from agentfield_amd.quant import quantize_mx
    torch.manual_seed(4)
    T, H, I = 256, 1024, 2816
    I2 = 2 * I
    x = rnd(T, H, seed=90, scale=0.5)
    gate_up = rnd(I2, H, seed=91, scale=0.1)
    down = rnd(H, I, seed=92, scale=0.1)
    # bf16 reference path
    act = ops.silu_and_mul(x @ gate_up.t())
    want = act @ down.t()
    # mxfp8 path: weights quantized ahead, activations on the fly
    gu8, gus = quantize_mx(gate_up.float().cpu())
    dn8, dns = quantize_mx(down.float().cpu())
    x8, xs = quantize_mx(x.float().cpu())
    gu = ops.gemm_mxfp8(x8.to(DEV), xs.to(DEV), gu8.to(DEV), gus.to(DEV))
    act_q = ops.silu_and_mul(gu)
    a8, as_ = quantize_mx(act_q.float().cpu())
    got = ops.gemm_mxfp8(a8.to(DEV), as_.to(DEV), dn8.to(DEV), dns.to(DEV))
    torch.cuda.synchronize()
    rel = (got.float() - want.float()).norm() / want.float().norm()
    # two chained W8A8 GEMMs with re-quantized activations: each e4m3
    # value carries ~2^-4 relative noise, and the down-projection sums
    # partially-correlated errors -> ~0.1 end-to-end is the expected
    # regime (single-GEMM exactness vs the dequantized product is pinned
    # to 5e-3 in test_gemm_mxfp8)
    assert rel.item() < 0.15, rel.item()
