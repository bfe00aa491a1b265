"""Automatic prefix caching: share paged-KV blocks across requests with a
common prompt prefix.

Agent workloads repeat system prompts and few-shot preambles constantly
(the reference forwards them to providers verbatim on every call —
SURVEY.md §3.4); serving them locally makes the prefix KV reusable.  Pages
hold RoPE'd keys, so equal token prefixes at equal positions are
bit-identical — full prompt pages are content-addressed by a chained hash
and shared read-only via refcounts.  Only FULL pages are shared (the
partial tail page is always freshly computed), and at least one prompt
token is always recomputed so the engine still has a last-row logit to
sample from.

Opt-in via LLMEngine(prefix_cache=True); uses the Python scheduler (the
C++ NativeScheduler gets refcounted pages in round 2).
"""
from __future__ import annotations

from collections import OrderedDict

from .scheduler import ScheduleBatch, Scheduler, SchedulerConfig
from .sequence import Sequence, SeqStatus


def prefix_hashes(prompt: list[int], page_size: int) -> list[int]:
    """Chained content hash per FULL prompt page (shared by the Python
    scheduler and the C++ NativeScheduler's prefix mode)."""
    out, h = [], 0
    for i in range(len(prompt) // page_size):
        h = hash((h, tuple(prompt[i * page_size:(i + 1) * page_size])))
        out.append(h)
    return out


class RefcountAllocator:
    """Free-list allocator with refcounted sharing.  Page 0 stays reserved
    (null page for padded graph lanes)."""

    def __init__(self, num_pages: int):
        self.num_pages = num_pages
        self.free_list = list(range(num_pages - 1, 0, -1))
        self.refs: dict[int, int] = {}

    @property
    def num_free(self) -> int:
        return len(self.free_list)

    def alloc(self, n: int) -> list[int]:
        if n > len(self.free_list):
            raise MemoryError(f"KV allocator: need {n} pages, "
                              f"{len(self.free_list)} free")
        out = [self.free_list.pop() for _ in range(n)]
        for p in out:
            self.refs[p] = 1
        return out

    def ref(self, page: int) -> None:
        self.refs[page] += 1

    def free(self, pages: list[int]) -> None:
        for p in pages:
            self.refs[p] -= 1
            if self.refs[p] == 0:
                del self.refs[p]
                self.free_list.append(p)


class PrefixCachingScheduler(Scheduler):
    def __init__(self, cfg: SchedulerConfig):
        super().__init__(cfg)
        self.alloc = RefcountAllocator(cfg.num_pages)
        # chained-hash -> page, LRU-ordered; the cache itself holds one ref
        self._cache: OrderedDict[int, int] = OrderedDict()
        self._page_hash: dict[int, int] = {}
        self.cache_hits = 0       # pages served from cache
        self.cached_tokens = 0    # prompt tokens whose prefill was skipped

    @property
    def cache_pages(self) -> int:
        return len(self._cache)

    # ------------------------------------------------------------ hashing
    def _prefix_hashes(self, prompt: list[int]) -> list[int]:
        return prefix_hashes(prompt, self.cfg.page_size)

    # ------------------------------------------------------------- cache
    def _evict_one(self) -> bool:
        """Drop the least-recently-used cached page that nothing else
        references (cache ref is the only one)."""
        for h, page in self._cache.items():
            if self.alloc.refs.get(page) == 1:
                del self._cache[h]
                del self._page_hash[page]
                self.alloc.free([page])
                return True
        return False

    def _ensure_free(self, need: int) -> bool:
        while self.alloc.num_free < need:
            if not self._evict_one():
                return False
        return True

    def _register_pages(self, seq: Sequence) -> None:
        """Publish the sequence's full prompt pages into the cache."""
        hashes = self._prefix_hashes(seq.prompt_ids)
        # only pages the seq has fully WRITTEN (prompt fully prefilled)
        if seq.num_prefilled < len(seq.prompt_ids):
            return
        for h, page in zip(hashes, seq.pages):
            if h in self._cache:
                self._cache.move_to_end(h)
                continue
            self._cache[h] = page
            self._page_hash[page] = h
            self.alloc.ref(page)

    # --------------------------------------------------------- scheduling
    def schedule(self) -> ScheduleBatch | None:
        cfg = self.cfg
        batch: list[Sequence] = []
        tokens = 0
        claimed: set[int] = set()  # first-uncached-page hashes this batch
        while (self.waiting and
               len(self.running) + len(batch) < cfg.max_num_seqs):
            cand = self.waiting[0]
            prompt = cand.prompt_ids
            hashes = self._prefix_hashes(prompt)
            # longest cached full-page prefix, capped so >=1 prompt token
            # is always recomputed (the sampler needs its logit row)
            matched: list[int] = []
            max_full = (len(prompt) - 1) // cfg.page_size
            for h in hashes[:max_full]:
                page = self._cache.get(h)
                if page is None:
                    break
                matched.append(page)
            # same-batch dedup: if another candidate in THIS batch is about
            # to compute the page this one needs next, defer — one round
            # later the publisher's pages are in the cache and this prompt
            # prefills as a hit instead of a duplicate (FIFO preserved)
            nxt = (hashes[len(matched)] if len(matched) < max_full
                   else None)
            if nxt is not None and nxt in claimed:
                break
            cached_tok = len(matched) * cfg.page_size
            # num_tokens includes outputs retained across preemption
            ntok = cand.num_tokens - cached_tok
            if batch and tokens + ntok > cfg.max_prefill_tokens:
                break
            # PIN matched pages before any eviction: _ensure_free must not
            # reclaim a page this candidate is about to share
            for p in matched:
                self.alloc.ref(p)
                self._cache.move_to_end(self._page_hash[p])
            need = self._pages_needed(cand.num_tokens) - len(matched)
            if not self._ensure_free(need):
                self.alloc.free(matched)  # unpin; candidate stays queued
                break
            self.waiting.popleft()
            cand.pages = matched + self.alloc.alloc(need)
            cand.cached_prefix = cached_tok
            cand.alloc_epoch += 1
            cand.status = SeqStatus.RUNNING
            self.cache_hits += len(matched)
            self.cached_tokens += cached_tok
            if nxt is not None:
                claimed.add(nxt)
            batch.append(cand)
            tokens += ntok
        if batch:
            self.running.extend(batch)
            return ScheduleBatch(is_prefill=True, seqs=batch)

        if not self.running:
            return None
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            self._roll(seq)  # refcounted free: shared prefix pages survive
            while not (self._pages_needed(seq.num_tokens + 1) <=
                       len(seq.pages) or self._grow_cached(seq)):
                if self.running[-1] is seq:
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

    def _grow_cached(self, seq: Sequence) -> bool:
        if not self._ensure_free(1):
            return False
        seq.pages.extend(self.alloc.alloc(1))
        return True

    def note_token(self, seq: Sequence) -> None:
        # first generated token => prompt fully prefilled: publish pages
        if not seq.output_ids:
            self._register_pages(seq)

    def finish(self, seq: Sequence) -> None:
        seq.status = SeqStatus.FINISHED
        self.running.remove(seq)
        self.release(seq)

    def release(self, seq: Sequence) -> None:
        self.alloc.free(seq.pages[seq.freed_pages:])
        seq.pages = []
        seq.freed_pages = 0
        seq.cached_prefix = 0
