"""Engine fuzz: random interleavings of add/step/cancel on the CPU tiny
model.  Invariants: every request reaches a terminal state, KV pages are
conserved, token counts respect sampling params."""
import random

import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models import CONFIGS


def test_engine_fuzz_add_step_cancel():
    rng = random.Random(42)
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=20, max_num_seqs=4,
                    max_prefill_tokens=16, enable_graphs=False,
                    max_waiting=64)
    total_pages = eng.sched.alloc.num_pages - 1
    live: dict[int, dict] = {}
    finished: dict[int, object] = {}
    cancelled: set[int] = set()
    submitted = 0

    for step in range(400):
        r = rng.random()
        if r < 0.35 and submitted < 60:
            n = rng.randint(4, 20)
            mt = rng.randint(8, 24)
            rid = eng.add_request([rng.randrange(500) for _ in range(n)],
                                  SamplingParams(max_tokens=mt,
                                                 ignore_eos=True))
            if rid is not None:
                live[rid] = {"max_tokens": mt}
                submitted += 1
        elif r < 0.45 and live:
            rid = rng.choice(list(live))
            if eng.cancel(rid):
                cancelled.add(rid)
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]

    # drain
    for _ in range(2000):
        if not eng.has_work():
            break
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]
    assert not live, f"requests stuck: {list(live)}"
    assert eng.sched.alloc.num_free == total_pages, "KV pages leaked"
    assert eng.sched.n_preempted > 0, "fuzz config should trigger preemption"
    # correctness: every non-cancelled request's greedy tokens must equal an
    # uninterrupted run (preemption/recompute must be transparent)
    ref_eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                        page_size=4, num_pages=128, max_num_seqs=4,
                        max_prefill_tokens=64, enable_graphs=False)
    checked = 0
    for rid, fin in finished.items():
        assert fin.finish_reason in ("length", "stop", "cancelled")
        if rid in cancelled and fin.finish_reason == "cancelled":
            continue
        assert len(fin.output_ids) == fin.sampling.max_tokens
        if checked < 12:  # bound the reference cost
            want = ref_eng.generate([fin.prompt_ids], fin.sampling)[0]
            assert fin.output_ids == want, \
                f"seq {rid} diverged after scheduling churn"
            checked += 1
    assert checked > 0


def test_engine_fuzz_with_speculation():
    """Same churn with prompt-lookup speculation enabled: multi-token
    acceptance must stay greedy-exact through preemption and recompute.
    Repetitive prompts make drafts fire; page pressure forces preemption."""
    rng = random.Random(7)
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=20, max_num_seqs=4,
                    max_prefill_tokens=16, enable_graphs=False,
                    max_waiting=64, spec_lookup=4)
    total_pages = eng.sched.alloc.num_pages - 1
    live: dict[int, dict] = {}
    finished = {}
    submitted = 0
    for step in range(400):
        if rng.random() < 0.3 and submitted < 40:
            pat = [rng.randrange(500) for _ in range(rng.randint(2, 4))]
            reps = rng.randint(2, 6)
            prompt = (pat * reps)[:rng.randint(4, 18)]
            mt = rng.randint(8, 24)
            rid = eng.add_request(prompt, SamplingParams(max_tokens=mt,
                                                         ignore_eos=True))
            if rid is not None:
                live[rid] = True
                submitted += 1
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]
    for _ in range(2000):
        if not eng.has_work():
            break
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]
    assert not live
    assert eng.sched.alloc.num_free == total_pages, "KV pages leaked"
    assert eng.metrics["spec_drafted"] > 0, "speculation never fired"
    assert eng.metrics["spec_accepted"] > 0, "no draft ever accepted"
    assert eng.sched.n_preempted > 0, "churn config should preempt"
    ref_eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                        page_size=4, num_pages=128, max_num_seqs=4,
                        max_prefill_tokens=64, enable_graphs=False)
    checked = 0
    for rid, fin in finished.items():
        assert len(fin.output_ids) == fin.sampling.max_tokens
        if checked < 10:
            want = ref_eng.generate([fin.prompt_ids], fin.sampling)[0]
            assert fin.output_ids == want, \
                f"seq {rid} diverged under speculation + churn"
            checked += 1
    assert checked > 0


def test_stream_events_exactly_once_under_preemption():
    """step() events (the streaming feed) never replay a token position
    after preemption: positions are strictly increasing per sequence."""
    rng = random.Random(11)
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=14, max_num_seqs=4,
                    max_prefill_tokens=16, enable_graphs=False,
                    max_waiting=64)
    counts = {}
    rids = []
    for i in range(6):
        rid = eng.add_request([rng.randrange(500) for _ in
                               range(rng.randint(6, 14))],
                              SamplingParams(max_tokens=10, ignore_eos=True))
        counts[rid] = 0
        rids.append(rid)
    fins = {}
    for _ in range(1500):
        for (rid, tok, done) in eng.step():
            counts[rid] += 1
        for rid in rids:
            if rid not in fins:
                f = eng.get_finished(rid)
                if f:
                    fins[rid] = f
        if len(fins) == len(rids):
            break
    assert len(fins) == len(rids)
    assert eng.sched.n_preempted > 0, "pool must force preemption"
    for rid, f in fins.items():
        assert counts[rid] == len(f.output_ids) == 10, \
            f"seq {rid}: {counts[rid]} events for {len(f.output_ids)} tokens"


def test_engine_fuzz_with_draft_model():
    """Preemption churn with the DRAFT-MODEL speculator: the draft KV
    (which reuses the sequences' page indices) must resync through
    release/re-admission, and every output stays greedy-exact."""
    rng = random.Random(11)
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=16, max_num_seqs=3,
                    max_prefill_tokens=16, enable_graphs=False,
                    max_waiting=64, spec_draft=CONFIGS["tiny"],
                    spec_draft_k=3)
    total_pages = eng.sched.alloc.num_pages - 1
    live, finished, submitted = {}, {}, 0
    for step in range(400):
        if rng.random() < 0.3 and submitted < 30:
            prompt = [rng.randrange(500) for _ in range(rng.randint(4, 16))]
            mt = rng.randint(8, 20)
            rid = eng.add_request(prompt, SamplingParams(max_tokens=mt,
                                                         ignore_eos=True))
            if rid is not None:
                live[rid] = True
                submitted += 1
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]
    for _ in range(3000):
        if not eng.has_work():
            break
        eng.step()
        for rid in list(live):
            fin = eng.get_finished(rid)
            if fin is not None:
                finished[rid] = fin
                del live[rid]
    assert not live
    assert eng.sched.alloc.num_free == total_pages, "KV pages leaked"
    assert eng.metrics["spec_steps"] > 0
    assert eng.sched.n_preempted > 0, "churn config should preempt"
    ref_eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                        page_size=4, num_pages=128, max_num_seqs=4,
                        max_prefill_tokens=64, enable_graphs=False)
    checked = 0
    for rid, fin in finished.items():
        assert len(fin.output_ids) == fin.sampling.max_tokens
        if checked < 8:
            want = ref_eng.generate([fin.prompt_ids], fin.sampling)[0]
            assert fin.output_ids == want, \
                f"seq {rid} diverged under draft-spec + churn"
            checked += 1
    assert checked > 0
