"""Scheduler + page-allocator semantics (CPU, no torch needed beyond import)."""
from agentfield_amd.engine import (PageAllocator, SamplingParams, Scheduler,
                                   SchedulerConfig, Sequence, SeqStatus)


def mkseq(i, n, max_tokens=8):
    return Sequence(i, list(range(n)), SamplingParams(max_tokens=max_tokens))


def test_allocator_reserves_null_page():
    a = PageAllocator(8)
    got = a.alloc(7)
    assert 0 not in got and len(set(got)) == 7
    try:
        a.alloc(1)
        assert False
    except MemoryError:
        pass
    a.free(got[:3])
    assert a.num_free == 3


def test_prefill_admission_respects_budget():
    s = Scheduler(SchedulerConfig(max_num_seqs=8, max_prefill_tokens=40,
                                  page_size=16, num_pages=64))
    for i in range(5):
        assert s.add(mkseq(i, 20))
    b = s.schedule()
    assert b.is_prefill and len(b.seqs) == 2  # 20+20 <= 40, third would exceed
    b2 = s.schedule()
    assert b2.is_prefill and len(b2.seqs) == 2
    assert s.num_running() == 4


def test_decode_after_prefill_and_finish():
    s = Scheduler(SchedulerConfig(max_num_seqs=4, max_prefill_tokens=64,
                                  page_size=4, num_pages=32))
    s.add(mkseq(0, 6))
    b = s.schedule()
    assert b.is_prefill
    seq = b.seqs[0]
    assert len(seq.pages) == 2  # 6 tokens / page 4
    seq.output_ids.append(1)    # 7 tokens
    b = s.schedule()
    assert not b.is_prefill
    # growth: 7+1 = 8 tokens fits in 2 pages; 9th triggers page 3
    seq.output_ids.extend([1, 1])
    s.schedule()
    assert len(seq.pages) == 3
    free_before = s.alloc.num_free
    s.finish(seq)
    assert s.alloc.num_free == free_before + 3
    assert s.schedule() is None


def test_preemption_on_oom():
    s = Scheduler(SchedulerConfig(max_num_seqs=4, max_prefill_tokens=64,
                                  page_size=1, num_pages=9))  # 8 usable
    s.add(mkseq(0, 4, max_tokens=16))
    s.add(mkseq(1, 4, max_tokens=16))
    b = s.schedule()
    assert len(b.seqs) == 2 and s.alloc.num_free == 0
    s1 = b.seqs[0]
    for seq in b.seqs:
        seq.output_ids.append(1)
    b = s.schedule()  # both need a page; only preemption can supply
    assert s.n_preempted >= 1
    assert b is not None and b.seqs == [s1]
    assert s.num_queued() == 1


def test_backpressure():
    s = Scheduler(SchedulerConfig(max_waiting=2))
    assert s.add(mkseq(0, 1)) and s.add(mkseq(1, 1))
    assert not s.add(mkseq(2, 1))
