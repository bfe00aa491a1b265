"""The C++ scheduler must behave exactly like the Python oracle."""
import random

import pytest

from agentfield_amd.engine import (SamplingParams, Scheduler, SchedulerConfig,
                                   Sequence)

try:
    from agentfield_amd._native import NativeScheduler
except ImportError:
    NativeScheduler = None

pytestmark = pytest.mark.skipif(NativeScheduler is None,
                                reason="_native not built")


def make_pair(**kw):
    cfg = SchedulerConfig(**kw)
    py = Scheduler(cfg)
    nat = NativeScheduler(cfg.max_num_seqs, cfg.max_prefill_tokens,
                          cfg.page_size, cfg.num_pages, cfg.max_waiting)
    return cfg, py, nat


def test_lockstep_random_workload():
    cfg, py, nat = make_pair(max_num_seqs=4, max_prefill_tokens=64,
                             page_size=4, num_pages=48)
    rng = random.Random(0)
    seqs: dict[int, Sequence] = {}
    next_id = 0
    remaining: dict[int, int] = {}

    for step in range(200):
        # occasionally add new requests
        if rng.random() < 0.4:
            n = rng.randint(1, 20)
            gen = rng.randint(1, 10)
            s = Sequence(next_id, list(range(n)), SamplingParams(max_tokens=gen))
            ok_py = py.add(s)
            ok_nat = nat.add(next_id, n)
            assert ok_py == ok_nat
            if ok_py:
                seqs[next_id] = s
                remaining[next_id] = gen
            next_id += 1

        b_py = py.schedule()
        r_nat = nat.schedule()
        if b_py is None:
            assert not r_nat.has_work
            continue
        assert r_nat.has_work
        assert b_py.is_prefill == r_nat.is_prefill
        assert [s.seq_id for s in b_py.seqs] == list(r_nat.seq_ids), \
            f"step {step}: batch mismatch"
        # native preemptions drop generated tokens like the oracle does
        for sid in r_nat.preempted:
            pass  # oracle already cleared output_ids internally
        for s in b_py.seqs:
            assert nat.pages(s.seq_id) == s.pages, f"step {step}: pages differ"
        # simulate one generated token per scheduled decode seq
        if not b_py.is_prefill:
            for s in list(b_py.seqs):
                s.output_ids.append(0)
                nat.note_token(s.seq_id)
                remaining[s.seq_id] -= 1
                if remaining[s.seq_id] <= 0:
                    py.finish(s)
                    nat.finish(s.seq_id)
        else:
            # prefill also emits the first token
            for s in list(b_py.seqs):
                s.output_ids.append(0)
                nat.note_token(s.seq_id)
                remaining[s.seq_id] -= 1
                if remaining[s.seq_id] <= 0:
                    py.finish(s)
                    nat.finish(s.seq_id)
        assert py.alloc.num_free == nat.num_free()
        assert py.num_queued() == nat.num_queued()
        assert py.num_running() == nat.num_running()

    assert py.n_preempted == nat.n_preempted()


def test_backpressure_parity():
    cfg, py, nat = make_pair(max_waiting=2)
    from agentfield_amd.engine import Sequence, SamplingParams
    for i in range(3):
        s = Sequence(i, [1], SamplingParams())
        assert py.add(s) == nat.add(i, 1)


def test_lockstep_prefix_cache():
    """The C++ prefix-cache mode must match engine/prefix_cache.py exactly:
    same page identities, same cached_prefix, same evictions, same
    preemptions — under a workload with heavy prompt repetition."""
    from agentfield_amd.engine.prefix_cache import (PrefixCachingScheduler,
                                                    prefix_hashes)
    cfg = SchedulerConfig(max_num_seqs=4, max_prefill_tokens=64,
                          page_size=4, num_pages=15)
    py = PrefixCachingScheduler(cfg)
    nat = NativeScheduler(cfg.max_num_seqs, cfg.max_prefill_tokens,
                          cfg.page_size, cfg.num_pages, cfg.max_waiting,
                          prefix_cache=True)
    rng = random.Random(5)
    base_prompts = [list(range(100, 100 + n)) for n in (9, 14, 22)]
    seqs, remaining = {}, {}
    next_id = 0
    for step in range(300):
        if rng.random() < 0.4:
            p = rng.choice(base_prompts)
            if rng.random() < 0.3:   # shared prefix, new tail
                p = p[:-2] + [rng.randrange(500), rng.randrange(500)]
            gen = rng.randint(4, 16)
            s = Sequence(next_id, list(p), SamplingParams(max_tokens=gen))
            ok_py = py.add(s)
            ok_nat = nat.add(next_id, len(p),
                             prefix_hashes(p, cfg.page_size))
            assert ok_py == ok_nat
            if ok_py:
                seqs[next_id] = s
                remaining[next_id] = gen
            next_id += 1
        b_py = py.schedule()
        r_nat = nat.schedule()
        if b_py is None:
            assert not r_nat.has_work
            continue
        assert r_nat.has_work and b_py.is_prefill == r_nat.is_prefill
        assert [s.seq_id for s in b_py.seqs] == list(r_nat.seq_ids), step
        for s in b_py.seqs:
            assert nat.pages(s.seq_id) == s.pages, \
                f"step {step}: page identity diverged"
            if b_py.is_prefill:
                assert nat.cached_prefix(s.seq_id) == \
                    getattr(s, "cached_prefix", 0), step
                s.num_prefilled = len(s.prompt_ids)  # engine would prefill
        for s in list(b_py.seqs):
            py.note_token(s)
            nat.note_token(s.seq_id)
            s.output_ids.append(0)
            remaining[s.seq_id] -= 1
            if remaining[s.seq_id] <= 0:
                py.finish(s)
                nat.finish(s.seq_id)
        assert py.alloc.num_free == nat.num_free(), step
        assert len(py._cache) == nat.cache_pages(), step
        assert py.cached_tokens == nat.cached_tokens(), step
        assert py.cache_hits == nat.cache_hits(), step
    assert py.n_preempted == nat.n_preempted()
    assert py.cached_tokens > 0, "workload never hit the cache"
    assert py.n_preempted > 0, "workload never preempted"
