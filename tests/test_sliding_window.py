"""Sliding-window attention (Mistral): band mask in both attention paths,
reference oracle vs a direct dense band-mask computation, and the engine
end-to-end with a windowed config."""
import math
import sys
from pathlib import Path

import pytest
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent))

from agentfield_amd import ops
from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models.llama import CONFIGS, LlamaConfig
from agentfield_amd.ops import reference as ref


def _paged(T, Hk, D=128, page=4, seed=0):
    torch.manual_seed(seed)
    npages = (T + page - 1) // page + 1
    kc = torch.randn(npages, Hk, page, D) * 0.3
    vc = torch.randn(npages, Hk, page, D) * 0.3
    bt = torch.arange(1, npages, dtype=torch.int32)[None, :]
    return kc, vc, bt


def _dense_band(q, ks, vs, scale, window, causal_upto=None):
    """q [Hq,D] single row attending last-`window` of ks/vs [L,Hk,D]."""
    Hq, D = q.shape
    L, Hk, _ = ks.shape
    G = Hq // Hk
    w0 = max(0, L - window) if window > 0 else 0
    out = torch.empty_like(q)
    for h in range(Hq):
        s = (ks[w0:, h // G] @ (q[h] * scale)).softmax(-1)
        out[h] = s @ vs[w0:, h // G]
    return out


def test_reference_decode_window_matches_dense():
    T, Hq, Hk, D, page = 37, 4, 2, 128, 4
    kc, vc, bt = _paged(T, Hk, page=page, seed=3)
    torch.manual_seed(4)
    q = torch.randn(1, Hq, D)
    lens = torch.tensor([T], dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    for window in (0, 8, 16, 64):
        got = ref.attn_decode(q, kc, vc, bt, lens, scale, window=window)[0]
        # reconstruct the contiguous K/V from pages
        ks = kc[bt[0].long()].permute(0, 2, 1, 3).reshape(-1, Hk, D)[:T]
        vs = vc[bt[0].long()].permute(0, 2, 1, 3).reshape(-1, Hk, D)[:T]
        want = _dense_band(q[0], ks, vs, scale, window)
        assert torch.allclose(got.float(), want.float(), atol=1e-4), window


def test_reference_prefill_window_band():
    """Windowed paged prefill == full-causal on a prompt shorter than the
    window, != (and banded) when longer."""
    T, Hq, Hk, D, page = 24, 4, 2, 128, 4
    torch.manual_seed(9)
    q = torch.randn(T, Hq, D)
    kc, vc, bt = _paged(T, Hk, page=page, seed=5)
    cu = torch.tensor([0, T], dtype=torch.int32)
    qs = torch.zeros(1, dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    full = ref.attn_prefill_paged(q, kc, vc, bt, qs, cu, scale)
    same = ref.attn_prefill_paged(q, kc, vc, bt, qs, cu, scale, window=T + 5)
    band = ref.attn_prefill_paged(q, kc, vc, bt, qs, cu, scale, window=6)
    assert torch.allclose(full, same, atol=1e-5)
    # rows inside the window are unchanged; later rows differ
    assert torch.allclose(full[:6], band[:6], atol=1e-5)
    assert not torch.allclose(full[-1], band[-1], atol=1e-3)
    # row t with window w must equal attending exactly [t-w+1, t]
    t, w = T - 1, 6
    ks = kc[bt[0].long()].permute(0, 2, 1, 3).reshape(-1, Hk, D)[:T]
    vs = vc[bt[0].long()].permute(0, 2, 1, 3).reshape(-1, Hk, D)[:T]
    want = _dense_band(q[t], ks[:t + 1], vs[:t + 1], scale, w)
    assert torch.allclose(band[t].float(), want.float(), atol=1e-4)


def test_engine_sliding_window_end_to_end():
    """Windowed tiny config decodes greedily; beyond the window the output
    diverges from the full-causal config with identical weights."""
    base = CONFIGS["tiny"]
    swa = LlamaConfig(name="tiny-swa", hidden_size=base.hidden_size,
                      intermediate_size=base.intermediate_size,
                      num_layers=base.num_layers, num_heads=base.num_heads,
                      num_kv_heads=base.num_kv_heads,
                      vocab_size=base.vocab_size,
                      max_position=base.max_position, sliding_window=8)

    def run(cfg):
        eng = LLMEngine(cfg, device="cpu", dtype=torch.float32, page_size=4,
                        num_pages=64, max_num_seqs=2, enable_graphs=False,
                        seed=3)
        rid = eng.add_request(list(range(1, 25)),
                              SamplingParams(max_tokens=12, ignore_eos=True))
        for _ in range(200):
            eng.step()
            f = eng.get_finished(rid)
            if f:
                return f.output_ids
        raise AssertionError("did not finish")

    full = run(base)
    win = run(swa)
    assert len(win) == 12
    assert win != full  # the 24-token prompt exceeds the 8-token window


def test_mistral_config_has_window():
    assert CONFIGS["mistral-7b"].sliding_window == 4096
    assert CONFIGS["mistral-7b"].shard(2).sliding_window == 4096


def _mk_seq(sid, n, sp=None):
    from agentfield_amd.engine.sequence import Sequence
    return Sequence(seq_id=sid, prompt_ids=list(range(1, n + 1)),
                    sampling=sp or SamplingParams(max_tokens=64,
                                                  ignore_eos=True))


def test_rolling_window_reclaims_pages():
    """Pages wholly behind the band (minus the 64-token staging slack)
    return to the pool during decode; release() must not double-free."""
    from agentfield_amd.engine.scheduler import Scheduler, SchedulerConfig
    cfg = SchedulerConfig(max_num_seqs=2, max_prefill_tokens=4096,
                          page_size=4, num_pages=128, window_tokens=16)
    s = Scheduler(cfg)
    seq = _mk_seq(1, 40)
    assert s.add(seq)
    b = s.schedule()
    assert b.is_prefill
    free_after_prefill = s.alloc.num_free
    for _ in range(60):
        seq.output_ids.append(7)
        s.schedule()
    # lim = 100 - 16 - 64 = 20 tokens -> 5 pages reclaimed
    assert seq.freed_pages == 5
    # grow() keeps capacity for num_tokens+1: 40 -> 101 tokens = 10 -> 26
    # pages (+16), 5 reclaimed
    assert s.alloc.num_free == free_after_prefill - 16 + 5
    total_free_before_finish = s.alloc.num_free
    s.finish(seq)
    assert s.alloc.num_free == 127  # all pages back exactly once
    assert s.alloc.num_free > total_free_before_finish


def test_rolling_window_native_lockstep():
    """C++ scheduler reclaims identically to the Python oracle."""
    pytest.importorskip("agentfield_amd._native")
    from agentfield_amd._native import NativeScheduler
    from agentfield_amd.engine.scheduler import Scheduler, SchedulerConfig
    cfg = SchedulerConfig(max_num_seqs=2, max_prefill_tokens=4096,
                          page_size=4, num_pages=64, window_tokens=16)
    py = Scheduler(cfg)
    nat = NativeScheduler(2, 4096, 4, 64, 4096, False, 16)
    seq = _mk_seq(1, 40)
    py.add(seq)
    nat.add(1, 40)
    py.schedule()
    nat.schedule()
    for step in range(50):
        seq.output_ids.append(9)
        nat.note_token(1)
        py.schedule()
        nat.schedule()
        assert nat.num_free() == py.alloc.num_free, step
    py.finish(seq)
    nat.finish(1)
    assert nat.num_free() == py.alloc.num_free == 63


def test_engine_rolling_outputs_identical(monkeypatch):
    """AF_KV_ROLL on/off: identical outputs (reclaimed pages are never
    scored), strictly more free pages with rolling."""
    base = CONFIGS["tiny"]
    swa = LlamaConfig(name="tiny-swa2", hidden_size=base.hidden_size,
                      intermediate_size=base.intermediate_size,
                      num_layers=base.num_layers, num_heads=base.num_heads,
                      num_kv_heads=base.num_kv_heads,
                      vocab_size=base.vocab_size,
                      max_position=base.max_position, sliding_window=16)

    def run(roll: str):
        monkeypatch.setenv("AF_KV_ROLL", roll)
        eng = LLMEngine(swa, device="cpu", dtype=torch.float32, page_size=4,
                        num_pages=128, max_num_seqs=2, enable_graphs=False,
                        seed=5)
        rid = eng.add_request(list(range(1, 61)),
                              SamplingParams(max_tokens=48, ignore_eos=True))
        for _ in range(300):
            eng.step()
            f = eng.get_finished(rid)
            if f:
                return f.output_ids, eng.sched.alloc.num_free
        raise AssertionError("did not finish")

    out_roll, free_roll = run("1")
    out_flat, free_flat = run("0")
    assert out_roll == out_flat
    assert len(out_roll) == 48


def test_rolling_window_respects_prefix_sharing():
    """A windowed sequence whose early pages are SHARED via the prefix
    cache: rolling reclamation unrefs them, but the cache (and any other
    sequence reading them) keeps the pages alive until eviction."""
    from agentfield_amd.engine.prefix_cache import PrefixCachingScheduler
    from agentfield_amd.engine.scheduler import SchedulerConfig
    from agentfield_amd.engine.sequence import Sequence

    cfg = SchedulerConfig(max_num_seqs=4, max_prefill_tokens=4096,
                          page_size=4, num_pages=64, window_tokens=16)
    s = PrefixCachingScheduler(cfg)
    sp = SamplingParams(max_tokens=64, ignore_eos=True)
    a = Sequence(seq_id=1, prompt_ids=list(range(1, 33)), sampling=sp)
    s.add(a)
    s.schedule()
    a.num_prefilled = 32
    s.note_token(a)          # publish prompt pages into the cache
    a.output_ids.append(5)
    cached_pages = list(a.pages[:8])
    # duplicate prompt admits THROUGH the shared pages
    b = Sequence(seq_id=2, prompt_ids=list(range(1, 33)), sampling=sp)
    s.add(b)
    r = s.schedule()
    assert r.is_prefill and b.cached_prefix == 28
    b.num_prefilled = 32
    s.note_token(b)
    b.output_ids.append(5)
    # decode a far past the window: its early pages roll
    for _ in range(60):
        a.output_ids.append(7)
        b.output_ids.append(7)
        s.schedule()
    assert a.freed_pages > 0 and b.freed_pages > 0
    # the shared pages are still referenced by the CACHE: none returned
    # to the free list while the cache holds them
    for p in cached_pages:
        assert p in s.alloc.refs, p
        assert p not in s.alloc.free_list
    # both sequences finish; cache still owns the shared pages
    s.finish(a)
    s.finish(b)
    for p in cached_pages[:7]:  # 7 full prefix pages were published
        assert p in s.alloc.refs
    # total conservation: free + cache-held == all pages
    assert s.alloc.num_free + len(s.alloc.refs) == 63
