"""Continuous-batching scheduler + KV page allocator (Python reference).

The native C++ implementation (agentfield_amd/native/scheduler.cpp, exposed
via agentfield_amd._native) is the production path; this module defines the
semantics and is the oracle for its tests.  The engine picks _native when the
extension is built.

Admission control is the GPU-side face of the control plane's bounded async
queue (reference: execute.go async worker pool): a bounded
waiting queue, a KV-page budget sized for 288 GB HBM3E, and preemption of the
youngest running sequence when decode runs out of pages.
"""
from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field

from .sequence import Sequence, SeqStatus


class PageAllocator:
    """Free-list page allocator.  Page 0 is reserved as the null page used by
    padded (dummy) decode lanes in hipGraph buckets."""

    def __init__(self, num_pages: int):
        self.num_pages = num_pages
        self.free_list = list(range(num_pages - 1, 0, -1))

    @property
    def num_free(self) -> int:
        return len(self.free_list)

    def alloc(self, n: int) -> list[int]:
        if n > len(self.free_list):
            raise MemoryError(f"KV allocator: need {n} pages, {len(self.free_list)} free")
        out = [self.free_list.pop() for _ in range(n)]
        return out

    def free(self, pages: list[int]) -> None:
        self.free_list.extend(pages)


@dataclass
class SchedulerConfig:
    max_num_seqs: int = 64
    max_prefill_tokens: int = 8192
    page_size: int = 16
    num_pages: int = 1024
    max_waiting: int = 4096
    # sliding-window models (cfg.sliding_window): pages wholly behind the
    # attention band are reclaimed during decode (rolling KV buffer).
    # 0 disables.  The 64-token slack covers the prefill kernel's KV-tile
    # staging below the band start (attn_prefill.hip kt_first).
    window_tokens: int = 0


ROLL_SLACK = 64


@dataclass
class ScheduleBatch:
    is_prefill: bool
    seqs: list[Sequence] = field(default_factory=list)


class Scheduler:
    def __init__(self, cfg: SchedulerConfig):
        self.cfg = cfg
        self.alloc = PageAllocator(cfg.num_pages)
        self.waiting: deque[Sequence] = deque()
        self.running: list[Sequence] = []
        self.n_preempted = 0

    # -- queue interface -------------------------------------------------
    def add(self, seq: Sequence) -> bool:
        """Returns False on backpressure (queue full -> control plane 503s)."""
        if len(self.waiting) >= self.cfg.max_waiting:
            return False
        self.waiting.append(seq)
        return True

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def num_queued(self) -> int:
        return len(self.waiting)

    def num_running(self) -> int:
        return len(self.running)

    # -- page bookkeeping ------------------------------------------------
    def _pages_needed(self, ntokens: int) -> int:
        return (ntokens + self.cfg.page_size - 1) // self.cfg.page_size

    def _roll(self, seq: Sequence) -> None:
        """Reclaim pages wholly behind the attention window.  The page ID
        stays in seq.pages (block-table rows keep their slot; the band
        mask guarantees those positions are never scored) but returns to
        the pool — long windowed sequences hold O(window) pages."""
        w = self.cfg.window_tokens
        if not w:
            return
        lim = seq.num_tokens - w - ROLL_SLACK
        while (seq.freed_pages + 1) * self.cfg.page_size <= lim:
            self.alloc.free([seq.pages[seq.freed_pages]])
            seq.freed_pages += 1

    def _grow(self, seq: Sequence) -> bool:
        """Ensure capacity for one more token; returns False if OOM."""
        need = self._pages_needed(seq.num_tokens + 1)
        if need > len(seq.pages):
            if self.alloc.num_free < 1:
                return False
            seq.pages.extend(self.alloc.alloc(1))
        return True

    def release(self, seq: Sequence) -> None:
        self.alloc.free(seq.pages[seq.freed_pages:])
        seq.pages = []
        seq.freed_pages = 0

    def _preempt_last(self) -> None:
        victim = self.running.pop()
        self.release(victim)
        victim.on_preempt()   # keep outputs; recompute their KV on readmit
        victim.status = SeqStatus.WAITING
        self.waiting.appendleft(victim)
        self.n_preempted += 1

    # -- the scheduling step ---------------------------------------------
    def schedule(self) -> ScheduleBatch | None:
        cfg = self.cfg
        # 1) admit prefills while budget allows
        batch: list[Sequence] = []
        tokens = 0
        while (self.waiting and len(self.running) + len(batch) < cfg.max_num_seqs):
            cand = self.waiting[0]
            # num_tokens includes outputs retained across preemption: their
            # KV recomputes as prefill on re-admission
            ntok = cand.num_tokens
            if batch and tokens + ntok > cfg.max_prefill_tokens:
                break
            need = self._pages_needed(ntok)
            if need > self.alloc.num_free:
                break
            self.waiting.popleft()
            cand.pages = self.alloc.alloc(need)
            cand.alloc_epoch += 1
            cand.status = SeqStatus.RUNNING
            batch.append(cand)
            tokens += ntok
        if batch:
            self.running.extend(batch)
            return ScheduleBatch(is_prefill=True, seqs=batch)

        # 2) otherwise decode everything running (grow pages, preempt on OOM)
        if not self.running:
            return None
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            self._roll(seq)
            while not self._grow(seq):
                if self.running[-1] is seq:
                    # can't preempt self and nothing else to free: defer
                    self.running.pop()
                    self.release(seq)
                    seq.on_preempt()
                    seq.status = SeqStatus.WAITING
                    self.waiting.appendleft(seq)
                    self.n_preempted += 1
                    i -= 1
                    break
                self._preempt_last()
            i += 1
        if not self.running:
            return None
        return ScheduleBatch(is_prefill=False, seqs=list(self.running))

    def note_token(self, seq: Sequence) -> None:
        """Interface parity with the native scheduler (token counting is
        implicit here via seq.num_tokens)."""

    def finish(self, seq: Sequence) -> None:
        seq.status = SeqStatus.FINISHED
        self.running.remove(seq)
        self.release(seq)


class _AllocShim:
    def __init__(self, nat, num_pages):
        self._nat = nat
        self.num_pages = num_pages

    @property
    def num_free(self):
        return self._nat.num_free()


class NativeSchedulerAdapter:
    """Engine-facing adapter over the C++ scheduler (agentfield_amd._native).
    Keeps the Python Sequence objects; page/queue decisions run natively.
    Semantics are pinned to the Python Scheduler by test_native_scheduler."""

    def __init__(self, cfg: SchedulerConfig, prefix_cache: bool = False):
        from agentfield_amd._native import NativeScheduler
        self.cfg = cfg
        self.prefix_cache = prefix_cache
        self.nat = NativeScheduler(cfg.max_num_seqs, cfg.max_prefill_tokens,
                                   cfg.page_size, cfg.num_pages,
                                   cfg.max_waiting, prefix_cache,
                                   cfg.window_tokens)
        self.seqs: dict[int, Sequence] = {}
        self.alloc = _AllocShim(self.nat, cfg.num_pages)

    @property
    def cached_tokens(self) -> int:
        return self.nat.cached_tokens() if self.prefix_cache else 0

    @property
    def cache_hits(self) -> int:
        return self.nat.cache_hits() if self.prefix_cache else 0

    @property
    def cache_pages(self) -> int:
        return self.nat.cache_pages() if self.prefix_cache else 0

    @property
    def n_preempted(self):
        return self.nat.n_preempted()

    def add(self, seq: Sequence) -> bool:
        hashes = []
        if self.prefix_cache:
            from .prefix_cache import prefix_hashes
            hashes = prefix_hashes(seq.prompt_ids, self.cfg.page_size)
        if not self.nat.add(seq.seq_id, len(seq.prompt_ids), hashes):
            return False
        self.seqs[seq.seq_id] = seq
        return True

    def has_work(self) -> bool:
        return self.nat.has_work()

    def num_queued(self) -> int:
        return self.nat.num_queued()

    def num_running(self) -> int:
        return self.nat.num_running()

    def note_token(self, seq: Sequence) -> None:
        self.nat.note_token(seq.seq_id)

    def schedule(self) -> ScheduleBatch | None:
        r = self.nat.schedule()
        for sid in r.preempted:
            s = self.seqs[sid]
            s.on_preempt()
            s.pages = []
            s.status = SeqStatus.WAITING
        if not r.has_work:
            return None
        batch = []
        for sid in r.seq_ids:
            s = self.seqs[sid]
            s.pages = self.nat.pages(sid)
            s.status = SeqStatus.RUNNING
            if r.is_prefill:
                s.alloc_epoch += 1  # fresh allocation: invalidate bt rows
                if self.prefix_cache:
                    s.cached_prefix = self.nat.cached_prefix(sid)
            batch.append(s)
        return ScheduleBatch(is_prefill=r.is_prefill, seqs=batch)

    def finish(self, seq: Sequence) -> None:
        seq.status = SeqStatus.FINISHED
        self.nat.finish(seq.seq_id)
        self.seqs.pop(seq.seq_id, None)


def make_scheduler(cfg: SchedulerConfig, prefer_native: bool = True):
    import os
    if prefer_native and os.environ.get("AF_NATIVE_SCHED", "1") != "0":
        try:
            return NativeSchedulerAdapter(cfg)
        except ImportError:
            pass
    return Scheduler(cfg)
