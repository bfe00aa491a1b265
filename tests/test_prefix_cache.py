"""Automatic prefix caching (engine/prefix_cache.py): shared prompt pages,
exactness, refcount hygiene, eviction, and preemption interplay."""
import random

import pytest

import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models import CONFIGS


def make(prefix_cache=True, num_pages=64, native=False, **kw):
    """White-box tests of the PYTHON PrefixCachingScheduler (the oracle
    the C++ NativeScheduler is lockstep-pinned to) — force it explicitly
    now that the native scheduler is the default.  native=True keeps the
    default C++ path."""
    import os
    old = os.environ.get("AF_NATIVE_PREFIX")
    os.environ["AF_NATIVE_PREFIX"] = "1" if native else "0"
    try:
        return LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                         page_size=4, num_pages=num_pages, max_num_seqs=4,
                         enable_graphs=False, prefix_cache=prefix_cache, **kw)
    finally:
        if old is None:
            os.environ.pop("AF_NATIVE_PREFIX", None)
        else:
            os.environ["AF_NATIVE_PREFIX"] = old


PROMPT = [7, 3, 9, 1, 8, 2, 6, 4, 5, 9, 2, 7, 1, 3, 8, 6, 4, 2, 9, 5, 7, 1]
SP = SamplingParams(max_tokens=6, ignore_eos=True)


def test_prefix_cache_hits_and_exact_tokens():
    base = make(prefix_cache=False).generate([PROMPT], SP)[0]
    eng = make()
    out1 = eng.generate([PROMPT], SP)[0]
    cold_prefill = eng.metrics["prefill_tokens"]
    assert out1 == base and eng.sched.cached_tokens == 0
    out2 = eng.generate([PROMPT], SP)[0]
    assert out2 == base
    # 22-token prompt, page 4: (22-1)//4 = 5 full pages = 20 tokens cached
    assert eng.sched.cached_tokens == 20
    assert eng.metrics["prefill_tokens"] == cold_prefill + 2  # only tail
    # shared prefix with a different tail also hits
    p2 = PROMPT[:20] + [9, 9, 9]
    base2 = make(prefix_cache=False).generate([p2], SP)[0]
    assert eng.generate([p2], SP)[0] == base2
    assert eng.sched.cached_tokens == 40  # +20 for p2's 5 shared pages


def test_prefix_cache_page_accounting():
    eng = make()
    for _ in range(3):
        eng.generate([PROMPT, PROMPT[:17]], SP)
    alloc = eng.sched.alloc
    # all sequence refs released; cache holds exactly one ref per entry
    assert len(alloc.refs) == len(eng.sched._cache)
    assert alloc.num_free == alloc.num_pages - 1 - len(eng.sched._cache)
    assert all(alloc.refs[p] == 1 for p in eng.sched._cache.values())


def test_prefix_cache_eviction_under_pressure():
    rng = random.Random(3)
    eng = make(num_pages=16)  # tiny pool forces eviction
    for i in range(12):
        p = [rng.randrange(500) for _ in range(rng.randint(8, 14))]
        out = eng.generate([p], SP)[0]
        assert len(out) == SP.max_tokens
    # pool never deadlocks and cache stays within the pool
    assert len(eng.sched._cache) <= 15


def test_prefix_cache_with_preemption_exact():
    """Preempted sequences re-admit THROUGH the cache (their own published
    pages) and still produce exact greedy tokens."""
    rng = random.Random(9)
    eng = make(num_pages=18, max_prefill_tokens=16, max_waiting=64)
    sp = SamplingParams(max_tokens=12, ignore_eos=True)
    prompts = [PROMPT, PROMPT[:18], PROMPT[:14], PROMPT]
    rids = {}
    for p in prompts:
        rids[eng.add_request(p, sp)] = p
    fins = {}
    for _ in range(2000):
        if len(fins) == len(rids):
            break
        eng.step()
        for r in list(rids):
            if r not in fins:
                f = eng.get_finished(r)
                if f:
                    fins[r] = f
    assert len(fins) == len(rids)
    ref = make(prefix_cache=False, num_pages=128)
    for r, p in rids.items():
        want = ref.generate([p], sp)[0]
        assert fins[r].output_ids == want, "prefix cache broke exactness"
    assert eng.sched.n_preempted > 0  # churn really happened
    assert eng.sched.cached_tokens > 0


def test_native_prefix_engine_end_to_end(monkeypatch):
    """The C++ prefix mode drives a full engine run: same exact tokens as
    the Python-scheduler engine, cache hits recorded."""
    pytest.importorskip("agentfield_amd._native")
    ref = make(prefix_cache=False)
    base = [ref.generate([PROMPT], SP)[0], ref.generate([PROMPT], SP)[0]]
    monkeypatch.setenv("AF_NATIVE_PREFIX", "1")
    eng = make(native=True)
    from agentfield_amd.engine.scheduler import NativeSchedulerAdapter
    assert isinstance(eng.sched, NativeSchedulerAdapter)
    got = [eng.generate([PROMPT], SP)[0], eng.generate([PROMPT], SP)[0]]
    assert got == base
    assert eng.sched.cached_tokens == 20  # second request hit 5 pages


def test_prefix_cache_with_speculation_exact():
    """Prefix cache + speculative decoding combined: draft verification
    writes KV only at positions >= prompt_len, which always land in the
    sequence's own fresh pages (shared pages cover strictly fewer full
    prompt pages), so the combination stays greedy-exact."""
    rep = [5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5, 9, 5]
    sp = SamplingParams(max_tokens=10, ignore_eos=True)
    base = make(prefix_cache=False).generate([rep], sp)[0]
    eng = make(spec_lookup=4)
    out1 = eng.generate([rep], sp)[0]
    out2 = eng.generate([rep], sp)[0]   # warm: cached prefix + speculation
    assert out1 == base and out2 == base
    assert eng.sched.cached_tokens > 0, "prefix cache never hit"
    assert eng.metrics["spec_drafted"] > 0, "speculation never fired"


def test_same_batch_duplicate_prompts_dedup():
    """Two identical prompts submitted together, cold cache: the second
    defers one round and then prefills as a cache hit instead of
    computing the same pages twice."""
    from agentfield_amd.engine.prefix_cache import PrefixCachingScheduler
    from agentfield_amd.engine.scheduler import SchedulerConfig
    from agentfield_amd.engine.sequence import Sequence

    def mkseq(sid, ids):
        return Sequence(seq_id=sid, prompt_ids=ids, sampling=SP)

    cfg = SchedulerConfig(max_num_seqs=8, max_prefill_tokens=4096,
                          page_size=4, num_pages=64)
    sched = PrefixCachingScheduler(cfg)
    a = mkseq(1, list(range(1, 21)))
    b = mkseq(2, list(range(1, 21)))
    sched.add(a)
    sched.add(b)
    r = sched.schedule()
    assert r.is_prefill and [s.seq_id for s in r.seqs] == [1]  # b deferred
    # prompt of a completes -> pages published
    a.num_prefilled = len(a.prompt_ids)
    sched.note_token(a)  # engine notes BEFORE appending the sampled token
    a.output_ids.append(5)
    r2 = sched.schedule()
    assert r2.is_prefill and [s.seq_id for s in r2.seqs] == [2]
    assert b.cached_prefix == 16  # 4 full pages shared, tail recomputed
    assert sched.cache_hits >= 4
    # distinct prompts are NOT deferred
    sched2 = PrefixCachingScheduler(cfg)
    c = mkseq(3, list(range(1, 21)))
    d = mkseq(4, list(range(100, 120)))
    sched2.add(c)
    sched2.add(d)
    r3 = sched2.schedule()
    assert [s.seq_id for s in r3.seqs] == [3, 4]


def test_same_batch_dedup_native_lockstep():
    pytest.importorskip("agentfield_amd._native")
    from agentfield_amd._native import NativeScheduler
    from agentfield_amd.engine.prefix_cache import prefix_hashes
    nat = NativeScheduler(8, 4096, 4, 64, 4096, True)
    h = prefix_hashes(list(range(1, 21)), 4)
    nat.add(1, 20, h)
    nat.add(2, 20, h)
    r = nat.schedule()
    assert r.is_prefill and list(r.seq_ids) == [1]
    nat.note_token(1)  # prompt prefilled: publish
    r2 = nat.schedule()
    assert list(r2.seq_ids) == [2]
    assert nat.cached_prefix(2) == 16
