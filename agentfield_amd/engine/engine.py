"""The serving engine: continuous batching over the paged-KV Llama model.

One engine instance drives one GPU (DP replica) or one TP rank group.
Decode steps are hipGraph-captured per batch-size bucket: the whole
(embed -> L x layer -> norm -> lm_head -> sample) step replays with only
device-buffer updates between replays (guide: capture launch-bound inner
loops in hipGraphs).
"""
from __future__ import annotations

import time

import torch

from .. import ops
from ..models.llama import (AttnMetadata, KVCache, LlamaConfig,
                            LlamaForCausalLM)
from .scheduler import ScheduleBatch, SchedulerConfig, make_scheduler
from .sequence import SamplingParams, Sequence, SeqStatus

DECODE_BUCKETS = (1, 2, 4, 8, 16, 32, 64, 128, 256)
LP_TOPN = 8  # top-N alternatives reported per token when logprobs requested


def _bucket_for(n: int, max_bs: int) -> int:
    for b in DECODE_BUCKETS:
        if n <= b:
            return min(b, max_bs)
    return max_bs


def choose_nsplit(bs: int, hk: int) -> int:
    """Decode attention split-K factor: fill >=1024 workgroups (256 CUs x
    4 blocks for latency hiding; measured 2.4 TB/s at 512 WGs, bs=64)."""
    target = 1024
    ns = max(1, min(16, target // max(1, bs * hk)))
    return 1 << (ns.bit_length() - 1)


class LLMEngine:
    def __init__(self, cfg: LlamaConfig, device="cuda",
                 dtype=torch.bfloat16, page_size: int = 64,
                 kv_memory_frac: float = 0.80, num_pages: int | None = None,
                 max_num_seqs: int = 64, max_prefill_tokens: int = 8192,
                 max_waiting: int = 4096, enable_graphs: bool = True, eos_id: int = 2, seed: int = 0,
                 model: LlamaForCausalLM | None = None,
                 tp_group=None, spec_lookup: int = 0,
                 spec_draft=None, spec_draft_k: int = 4,
                 draft_model: "LlamaForCausalLM | None" = None,
                 prefix_cache: bool = False):
        self.cfg = cfg
        self.device = torch.device(device)
        self.is_gpu = self.device.type == "cuda"
        self.tp_group = tp_group
        if model is None:
            model = LlamaForCausalLM(cfg, device=device, dtype=dtype).init_random(seed)
        self.model = model.eval()
        if self.device.type == "cuda" and tp_group is None and \
                not getattr(model, "no_fused_decode", False):
            self.model.fold_norm_weights()

        if num_pages is None:
            if self.is_gpu:
                free, _total = torch.cuda.mem_get_info(self.device)
                budget = int(free * kv_memory_frac)
            else:
                budget = 64 << 20
            num_pages = max(16, budget // KVCache.bytes_per_page(cfg, page_size))
        max_pages_per_seq = (cfg.max_position + page_size - 1) // page_size
        self.max_pages_per_seq = max_pages_per_seq
        self.kv = KVCache(cfg, num_pages, page_size, self.device, dtype)
        # draft-model speculation (greedy-exact): a small model proposes
        # spec_draft_k tokens per step; the main model verifies them in
        # one chunked forward (_step_verify — shared with prompt-lookup).
        # The draft KV cache REUSES the sequences' page indices (its own
        # pool, same geometry), so preemption/allocation stay in sync for
        # free; seq.draft_len tracks how far the draft's KV is valid.
        self.draft = None
        self.spec_draft_k = int(spec_draft_k)
        # rejection-sampled speculation (temperature > 0) draws from its
        # own stream: deterministic per seed, distinct from the sampler's
        self._spec_gen = torch.Generator(
            device=self.device if self.is_gpu else "cpu")
        self._spec_gen.manual_seed((seed or 0x5EED) ^ 0x5A5A5A)
        if spec_draft is not None or draft_model is not None:
            from ..models import CONFIGS as _CFGS
            if draft_model is not None:
                dcfg = draft_model.cfg
            else:
                dcfg = (_CFGS[spec_draft] if isinstance(spec_draft, str)
                        else spec_draft)
                draft_model = LlamaForCausalLM(
                    dcfg, device=device, dtype=dtype).init_random(seed + 1)
            assert dcfg.vocab_size == cfg.vocab_size, \
                "draft and target must share a vocabulary"
            self.draft = draft_model.eval()
            self.draft_cfg = dcfg
            self.draft_kv = KVCache(dcfg, num_pages, page_size, self.device,
                                    dtype)
        import os as _os
        # rolling KV reclamation for sliding-window models (AF_KV_ROLL=0
        # keeps every page allocated — A/B/debug escape hatch)
        roll = (getattr(cfg, "sliding_window", 0)
                if _os.environ.get("AF_KV_ROLL", "1") != "0" else 0)
        sched_cfg = SchedulerConfig(
            max_num_seqs=max_num_seqs, max_prefill_tokens=max_prefill_tokens,
            page_size=page_size, num_pages=num_pages, max_waiting=max_waiting,
            window_tokens=roll)
        if prefix_cache:
            # The C++ NativeScheduler's prefix mode is the default (round-2:
            # lockstep-pinned to the Python oracle in
            # tests/test_native_scheduler.py and GPU-soaked by the DAG
            # bench); AF_NATIVE_PREFIX=0 falls back to the Python impl.
            import os
            use_native = os.environ.get("AF_NATIVE_PREFIX", "1") != "0"
            self.sched = None
            if use_native:
                try:
                    from .scheduler import NativeSchedulerAdapter
                    self.sched = NativeSchedulerAdapter(sched_cfg,
                                                        prefix_cache=True)
                except ImportError:
                    pass  # extension not built: Python fallback
            if self.sched is None:
                from .prefix_cache import PrefixCachingScheduler
                self.sched = PrefixCachingScheduler(sched_cfg)
        else:
            self.sched = make_scheduler(sched_cfg)
        self.page_size = page_size
        self.eos_id = eos_id
        self.max_num_seqs = max_num_seqs
        self._next_id = 0
        self._prefilling: list[Sequence] = []
        self._finished: dict[int, Sequence] = {}
        self.sampler = ops.SamplerState(max_num_seqs, self.device, seed=seed or 0x5EED)
        # MoE decode graph-captures too: the static-capacity dispatch
        # (models/llama.py MoEMLP) keeps every decode shape fixed
        self.enable_graphs = enable_graphs and self.is_gpu
        self._graphs: dict[int, dict] = {}
        self.metrics = {"prefill_tokens": 0, "decode_tokens": 0, "steps": 0,
                        "prefill_steps": 0, "decode_steps": 0,
                        "spec_steps": 0, "spec_drafted": 0, "spec_accepted": 0}
        # prompt-lookup speculative decoding (greedy-exact, opt-in):
        # draft-model speculation (greedy-exact): a small model proposes
        # spec_draft_k tokens per step; the main model verifies them in
        # one chunked forward (_step_verify — shared with prompt-lookup).
        # The draft KV cache REUSES the sequences' page indices (its own
        # pool, same geometry), so preemption/allocation stay in sync
        # for free; seq.draft_len tracks how far the draft's KV is valid.
        self.draft = None
        self.spec_draft_k = int(spec_draft_k)
        # rejection-sampled speculation (temperature > 0) draws from its
        # own stream: deterministic per seed, distinct from the sampler's
        self._spec_gen = torch.Generator(
            device=self.device if self.is_gpu else "cpu")
        self._spec_gen.manual_seed((seed or 0x5EED) ^ 0x5A5A5A)
        if spec_draft is not None or draft_model is not None:
            from ..models import CONFIGS as _CFGS
            if draft_model is not None:
                dcfg = draft_model.cfg
            else:
                dcfg = (_CFGS[spec_draft] if isinstance(spec_draft, str)
                        else spec_draft)
                draft_model = LlamaForCausalLM(
                    dcfg, device=device, dtype=dtype).init_random(seed + 1)
            assert dcfg.vocab_size == cfg.vocab_size,                 "draft and target must share a vocabulary"
            self.draft = draft_model.eval()
            self.draft_cfg = dcfg
        # draft up to spec_lookup tokens from n-gram matches in the
        # sequence's own context and verify them in ONE chunked-prefill
        # forward; every emitted token is the model's true greedy token.
        self.spec_lookup = int(spec_lookup)
        # static decode buffers (shared across graph buckets; sized to max)
        B = max_num_seqs
        dev = self.device
        self._dec = {
            "ids": torch.zeros(B, dtype=torch.int32, device=dev),
            "pos": torch.zeros(B, dtype=torch.int32, device=dev),
            "slots": torch.zeros(B, dtype=torch.int64, device=dev),
            "bt": torch.zeros(B, max_pages_per_seq, dtype=torch.int32, device=dev),
            "lens": torch.ones(B, dtype=torch.int32, device=dev),
            "temps": torch.zeros(B, dtype=torch.float32, device=dev),
            "topk": torch.zeros(B, dtype=torch.int32, device=dev),
            "topp": torch.ones(B, dtype=torch.float32, device=dev),
            "tokens": torch.zeros(B, dtype=torch.int32, device=dev),
            # logprob reporting (filled only in lp-keyed graphs)
            "lp_tok": torch.zeros(B, dtype=torch.float32, device=dev),
            "lp_vals": torch.zeros(B, LP_TOPN, dtype=torch.float32, device=dev),
            "lp_ids": torch.zeros(B, LP_TOPN, dtype=torch.int32, device=dev),
        }
        self._host = {k: torch.zeros_like(v, device="cpu").pin_memory()
                      if self.is_gpu else torch.zeros_like(v)
                      for k, v in self._dec.items()}
        # block-table dirty tracking: row -> (seq_id, pages_written)
        self._bt_rows: dict[int, tuple] = {}

    # ------------------------------------------------------------- requests
    def add_request(self, prompt_ids: list[int], sampling: SamplingParams | None = None,
                    on_token=None) -> int | None:
        sampling = sampling or SamplingParams()
        seq = Sequence(self._next_id, list(prompt_ids), sampling,
                       on_token=on_token, arrival_ns=time.monotonic_ns())
        if len(seq.prompt_ids) + sampling.max_tokens > self.cfg.max_position:
            raise ValueError("request exceeds model max_position")
        if not self.sched.add(seq):
            return None  # backpressure
        self._next_id += 1
        return seq.seq_id

    def get_finished(self, seq_id: int) -> Sequence | None:
        return self._finished.pop(seq_id, None)

    def cancel(self, seq_id: int) -> bool:
        """Cancel a queued or running request; frees its KV pages."""
        # waiting queue (python scheduler) or native adapter's seq map
        seqs = getattr(self.sched, "seqs", None)
        if seqs is not None:  # native adapter
            seq = seqs.get(seq_id)
        else:
            seq = next((s for s in list(self.sched.waiting) +
                        self.sched.running if s.seq_id == seq_id), None)
        if seq is None or seq.status == SeqStatus.FINISHED:
            return False
        if seq in self._prefilling:
            self._prefilling.remove(seq)
        if seqs is not None:
            self.sched.nat.finish(seq_id)
            seqs.pop(seq_id, None)
        else:
            if seq in self.sched.running:
                self.sched.running.remove(seq)
            elif seq in self.sched.waiting:
                self.sched.waiting.remove(seq)
            self.sched.release(seq)
        seq.status = SeqStatus.FINISHED
        seq.finish_reason = "cancelled"
        self._finished[seq_id] = seq
        return True

    def has_work(self) -> bool:
        return bool(self._prefilling) or self.sched.has_work()

    # ------------------------------------------------------------- stepping
    @torch.inference_mode()
    def step(self) -> list[tuple[int, int, bool]]:
        """Run one engine iteration.  Returns [(seq_id, token, done)].

        Long prompts prefill in chunks of max_prefill_tokens (paged prefill
        attention reads history through the block table), so admission cost
        is bounded regardless of prompt length."""
        if not self._prefilling:
            batch = self.sched.schedule()
            if batch is None:
                return []
            if batch.is_prefill:
                for s in batch.seqs:
                    # prefix cache: cached full pages skip their prefill
                    s.num_prefilled = getattr(s, "cached_prefix", 0)
                self._prefilling.extend(batch.seqs)
            else:
                self.metrics["steps"] += 1
                if (self.spec_lookup > 0 or self.draft is not None) and \
                        self.tp_group is None and \
                        all(s.sampling.top_k <= 0 and
                            s.sampling.top_p >= 1.0
                            and not s.sampling.json_mode
                            for s in batch.seqs):
                    if self.draft is not None:
                        drafts, qs = self._draft_model_propose(batch.seqs)
                    else:
                        drafts = {s.seq_id: self._draft_for(s)
                                  for s in batch.seqs}
                        qs = {}
                    if any(drafts.values()):
                        self.metrics["spec_steps"] += 1
                        return self._step_verify(batch, drafts, qs)
                self.metrics["decode_steps"] += 1
                tokens = self._step_decode(batch)
                return self._bookkeep(batch.seqs, tokens)
        self.metrics["steps"] += 1
        self.metrics["prefill_steps"] += 1
        done_seqs, tokens = self._step_prefill_chunk()
        return self._bookkeep(done_seqs, tokens)

    def _bookkeep(self, seqs, tokens) -> list[tuple[int, int, bool]]:
        events = []
        now = time.monotonic_ns()
        for seq, tok in zip(seqs, tokens):
            if not seq.output_ids:
                seq.first_token_ns = now
            self.sched.note_token(seq)
            done = seq.append(tok, self.eos_id)
            if done:
                seq.finish_ns = now
                self.sched.finish(seq)
                self._finished[seq.seq_id] = seq
            if self._emit(seq, done):
                if seq.on_token is not None:
                    seq.on_token(tok, done)
                events.append((seq.seq_id, tok, done))
        return events

    @staticmethod
    def _emit(seq: Sequence, done: bool) -> bool:
        """Stream exactly-once: preemption retains generated tokens (their
        KV recomputes as prefill), so new tokens always advance the
        high-water mark; the mark stays as a safety net against any future
        path that re-walks delivered tokens."""
        pos = len(seq.output_ids)
        if pos > getattr(seq, "_streamed", 0):
            seq._streamed = pos
            return True
        return done  # terminal event always delivered

    def _slot(self, seq: Sequence, tok_idx: int) -> int:
        return seq.pages[tok_idx // self.page_size] * self.page_size + \
            tok_idx % self.page_size

    def _step_prefill_chunk(self) -> tuple[list[Sequence], list[int]]:
        """Prefill up to max_prefill_tokens across the pending prompts;
        sample only for prompts whose last chunk completed.

        A sequence preempted mid-generation re-enters here with its
        output_ids retained: the recompute covers prompt AND outputs, so
        decode resumes exactly where it stopped (no resampling — streamed
        tokens stay the final tokens even at temperature>0)."""
        dev = self.device
        budget = self.sched.cfg.max_prefill_tokens
        seqs, chunks, known = [], [], []
        for seq in self._prefilling:
            if budget <= 0:
                break
            all_ids = seq.prompt_ids + seq.output_ids
            take = min(len(all_ids) - seq.num_prefilled, budget)
            if take <= 0:
                continue
            seqs.append(seq)
            chunks.append(take)
            known.append(all_ids)
            budget -= take
        ids, pos, slots, q_start = [], [], [], []
        bt_rows = []
        for seq, take, all_ids in zip(seqs, chunks, known):
            np0 = seq.num_prefilled
            ids.extend(all_ids[np0:np0 + take])
            pos.extend(range(np0, np0 + take))
            slots.extend(self._slot(seq, i) for i in range(np0, np0 + take))
            q_start.append(np0)
            row = torch.zeros(self.max_pages_per_seq, dtype=torch.int32)
            row[:len(seq.pages)] = torch.tensor(seq.pages, dtype=torch.int32)
            bt_rows.append(row)
        self.metrics["prefill_tokens"] += len(ids)
        cu_list = [0]
        for ln in chunks:
            cu_list.append(cu_list[-1] + ln)
        cu = torch.tensor(cu_list, dtype=torch.int32, device=dev)
        ids_t = torch.tensor(ids, dtype=torch.int32, device=dev)
        pos_t = torch.tensor(pos, dtype=torch.int32, device=dev)
        slots_t = torch.tensor(slots, dtype=torch.int64, device=dev)
        md = AttnMetadata(is_prefill=True, slots=slots_t, cu_seqlens=cu,
                          seq_lens=chunks,
                          q_start=torch.tensor(q_start, dtype=torch.int32,
                                               device=dev),
                          block_table=torch.stack(bt_rows).to(dev))
        done, done_rows = [], []
        for i, (seq, take, all_ids) in enumerate(zip(seqs, chunks, known)):
            seq.num_prefilled += take
            if seq.num_prefilled >= len(all_ids):
                done.append(seq)
                done_rows.append(cu_list[i + 1] - 1)
        for seq in done:
            self._prefilling.remove(seq)
        if not done:
            self.model(ids_t, pos_t, self.kv, md,
                       logit_rows=torch.zeros(1, dtype=torch.int32, device=dev))
            return [], []
        last_rows = torch.tensor(done_rows, dtype=torch.int32, device=dev)
        logits = self.model(ids_t, pos_t, self.kv, md, logit_rows=last_rows)
        if any(s.sampling.json_mode for s in done):
            logits = logits + self._json_mask(done).to(logits.dtype)
        temps = torch.tensor([s.sampling.temperature for s in done],
                             dtype=torch.float32, device=dev)
        kw = {}
        if any(s.sampling.top_k > 0 or s.sampling.top_p < 1.0 for s in done) \
                and dev.type == "cuda":
            kw = {"topk": torch.tensor([s.sampling.top_k for s in done],
                                       dtype=torch.int32, device=dev),
                  "topp": torch.tensor([s.sampling.top_p for s in done],
                                       dtype=torch.float32, device=dev)}
        toks = ops.sample(logits, temps, self.sampler, **kw)
        if any(s.sampling.logprobs > 0 for s in done):
            self._attach_row_logprobs(done, logits, toks)
        return done, toks.cpu().tolist()

    def _attach_row_logprobs(self, seqs, logits, toks) -> None:
        """Per-row logprob reporting for eager paths (prefill-final and
        speculative verify): rows of `logits` align with `seqs`/`toks`."""
        lf = logits.float()
        lse = torch.logsumexp(lf, dim=-1)
        chosen = (lf.gather(1, toks.long().unsqueeze(1)).squeeze(1) -
                  lse).cpu().tolist()
        tv, ti = lf.topk(LP_TOPN, dim=-1)
        tv = (tv - lse.unsqueeze(1)).cpu().tolist()
        ti = ti.cpu().tolist()
        for i, seq in enumerate(seqs):
            nreq = seq.sampling.logprobs
            if nreq <= 0:
                continue
            if seq.logprobs is None:
                seq.logprobs = []
            seq.logprobs.append({"logprob": chosen[i],
                                 "top": list(zip(ti[i][:nreq],
                                                 tv[i][:nreq]))})

    # -- grammar-constrained JSON decoding ----------------------------------
    # Byte tokenizer: the byte PDA masks directly (jsonfsm.py).  HF/BPE
    # vocabularies: attach a TokenJsonGrammar (set_token_grammar) and the
    # same seam masks whole tokens (token_grammar.py).
    def set_token_grammar(self, grammar) -> None:
        self.token_grammar = grammar

    def _json_allowed(self, seq: Sequence) -> list[int]:
        """Token ids legal for seq's next token under the JSON grammar,
        completable within its remaining budget.  The FSM is cached on the
        sequence and rebuilt after preemption (output_ids reset)."""
        from .jsonfsm import EOS_ID, JsonFSM
        grammar = getattr(self, "token_grammar", None)
        fsm = getattr(seq, "_fsm", None)
        pos = getattr(seq, "_fsm_pos", 0)
        if fsm is None or pos > len(seq.output_ids):
            spec = seq.sampling.json_schema
            if spec is not None:
                from .schemafsm import SchemaFSM, SchemaSpec, make_fsm
                if isinstance(spec, SchemaSpec):
                    fsm = SchemaFSM(spec)
                else:
                    fsm = make_fsm(spec)  # dict (incl. root anyOf)
            else:
                fsm = JsonFSM()
            pos = 0
        try:
            for tid in seq.output_ids[pos:]:
                if grammar is not None:
                    grammar.advance_token(fsm, tid)
                    continue
                b = tid - 4  # ByteTokenizer offset
                if 0 <= b < 256:
                    fsm.advance(b)
        except (ValueError, AssertionError):
            # a grammar-invalid token in history means the constraint was
            # already violated (should be impossible; the sampler-overflow
            # bug fixed in csrc/sampling.hip was the one known cause).
            # Fail OPEN for this sequence rather than killing the engine
            # loop: drop the constraint and let it finish unconstrained.
            import logging
            logging.getLogger("agentfield.engine").warning(
                "seq %d: grammar-invalid token in history; disabling "
                "json_mode for it", seq.seq_id)
            seq.sampling.json_mode = False
            return list(range(self.cfg.vocab_size))
        seq._fsm, seq._fsm_pos = fsm, len(seq.output_ids)
        remaining = seq.sampling.max_tokens - len(seq.output_ids)
        if grammar is not None:
            return grammar.allowed_token_ids(fsm, remaining)
        ids = fsm.allowed_token_ids(remaining)
        return [self.eos_id if i == EOS_ID else i for i in ids]

    def _json_mask(self, seqs: list[Sequence]) -> torch.Tensor:
        """Additive [len(seqs), V] mask: 0 for allowed, -inf elsewhere for
        json_mode rows; all-0 for unconstrained rows."""
        V = self.cfg.vocab_size
        mask = torch.zeros(len(seqs), V)
        for i, seq in enumerate(seqs):
            if seq.sampling.json_mode:
                # -1e30 not -inf: stays finite through the bf16 top-k/p
                # histogram sampler while still contributing zero mass
                mask[i] = -1e30
                mask[i, self._json_allowed(seq)] = 0.0
        return mask.to(self.device)

    # -- speculative decode: draft model (greedy-exact) ---------------------
    def _spec_cap(self, seq: Sequence, k: int) -> int:
        """Cap a draft to the generation budget (minus the bonus token)
        and to KV capacity the scheduler already allocated."""
        n = seq.num_tokens
        remaining = len(seq.prompt_ids) + seq.sampling.max_tokens - n
        capacity = len(seq.pages) * self.page_size - n
        return max(0, min(k, remaining - 1, capacity))

    @torch.no_grad()
    def _draft_model_propose(self, seqs: list[Sequence]) -> tuple:
        """Autoregressive k-token proposals from the draft model.  First a
        varlen catch-up chunk brings each sequence's draft KV up to its
        current length (tokens the target emitted without the draft —
        bonus tokens, re-admissions after preemption), then k batched
        decode rounds propose — argmax for greedy sequences, a sample
        from the draft's temperature-scaled distribution otherwise (the
        q the rejection-sampling verify divides by).  Returns
        (drafts, q_rows): q_rows[seq_id] is a [k, V] tensor of draft
        probabilities for sampled sequences, None entries for greedy."""
        dev = self.device
        ks = {s.seq_id: self._spec_cap(s, self.spec_draft_k) for s in seqs}
        live = [s for s in seqs if ks[s.seq_id] > 0]
        if not live:
            return {s.seq_id: [] for s in seqs}, {}
        # ---- catch-up: feed ctx[draft_len : n-1] (KV only) ----
        cu_ids, cu_pos, cu_slots, cu_qs, cu_lens, cu_bt = [], [], [], [], [], []
        for s in live:
            dl = getattr(s, "draft_len", 0)
            n = s.num_tokens
            if dl >= n - 1:
                continue
            ctx = s.prompt_ids + s.output_ids
            chunk = ctx[dl:n - 1]
            cu_ids.extend(chunk)
            cu_pos.extend(range(dl, n - 1))
            cu_slots.extend(self._slot(s, i) for i in range(dl, n - 1))
            cu_qs.append(dl)
            cu_lens.append(len(chunk))
            row = torch.zeros(self.max_pages_per_seq, dtype=torch.int32)
            row[:len(s.pages)] = torch.tensor(s.pages, dtype=torch.int32)
            cu_bt.append(row)
            s.draft_len = n - 1
        if cu_ids:
            cu = [0]
            for ln in cu_lens:
                cu.append(cu[-1] + ln)
            md = AttnMetadata(
                is_prefill=True,
                slots=torch.tensor(cu_slots, dtype=torch.int64, device=dev),
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
                seq_lens=cu_lens,
                q_start=torch.tensor(cu_qs, dtype=torch.int32, device=dev),
                block_table=torch.stack(cu_bt).to(dev))
            self.draft(torch.tensor(cu_ids, dtype=torch.int32, device=dev),
                       torch.tensor(cu_pos, dtype=torch.int32, device=dev),
                       self.draft_kv, md)
        # ---- k decode rounds over the live batch ----
        B = len(live)
        bt = torch.zeros(B, self.max_pages_per_seq, dtype=torch.int32)
        for i, s in enumerate(live):
            bt[i, :len(s.pages)] = torch.tensor(s.pages, dtype=torch.int32)
        bt = bt.to(dev)
        cur = torch.tensor([s.last_token for s in live], dtype=torch.int32,
                           device=dev)
        base = torch.tensor([s.num_tokens for s in live], dtype=torch.int32,
                            device=dev)
        out: dict = {s.seq_id: [] for s in seqs}
        temps = [s.sampling.temperature for s in live]
        any_sampled = any(t > 0.0 for t in temps)
        q_rows: dict = {s.seq_id: [] for s in live
                        if s.sampling.temperature > 0.0}
        kmax = max(ks[s.seq_id] for s in live)
        for j in range(kmax):
            pos = base - 1 + j
            # lanes past their own cap keep decoding (batch stays static)
            # but write KV to the reserved null page; their outputs are
            # discarded below
            slots = torch.tensor(
                [self._slot(s, s.num_tokens - 1 + j)
                 if j < ks[s.seq_id] else 0 for s in live],
                dtype=torch.int64, device=dev)
            md = AttnMetadata(is_prefill=False, slots=slots,
                              block_table=bt, seq_lens_t=base + j,
                              nsplit=choose_nsplit(B,
                                                   self.draft_cfg.num_kv_heads))
            logits = self.draft(cur, pos, self.draft_kv, md)
            nxt = logits.argmax(dim=-1).to(torch.int32)
            if any_sampled:
                nl = [int(x) for x in nxt.cpu()]
                for i, s in enumerate(live):
                    t = temps[i]
                    if t <= 0.0:
                        continue
                    q = torch.softmax(logits[i].float() / t, dim=-1)
                    tok = int(torch.multinomial(q, 1,
                                                generator=self._spec_gen))
                    nl[i] = tok
                    if j < ks[s.seq_id]:
                        q_rows[s.seq_id].append(q)
                nxt = torch.tensor(nl, dtype=torch.int32, device=dev)
            cur = nxt
            toks = cur.cpu().tolist()
            for i, s in enumerate(live):
                if j < ks[s.seq_id]:
                    out[s.seq_id].append(int(toks[i]))
        qs = {sid: torch.stack(rows) for sid, rows in q_rows.items() if rows}
        return out, qs

    # -- speculative decode (prompt lookup, greedy-exact) -------------------
    def _draft_for(self, seq: Sequence) -> list[int]:
        """Propose a continuation from the most recent earlier occurrence of
        the context's tail bigram.  Capped to (a) remaining generation
        budget minus the bonus token and (b) KV capacity already allocated
        by the scheduler — so verification never needs new pages."""
        k = self.spec_lookup
        n = seq.num_tokens
        remaining = len(seq.prompt_ids) + seq.sampling.max_tokens - n
        capacity = len(seq.pages) * self.page_size - n
        k = min(k, remaining - 1, capacity)
        if k <= 0:
            return []
        ctx = seq.prompt_ids + seq.output_ids
        # longest-suffix match first: a 4-gram match predicts the
        # continuation better than a bigram (acceptance is what pays;
        # correctness never depends on draft quality)
        for g in (4, 3, 2):
            if len(ctx) <= g:
                continue
            tail = ctx[-g:]
            for i in range(len(ctx) - g - 1, -1, -1):
                if ctx[i:i + g] == tail:
                    d = ctx[i + g:i + g + k]
                    if d:
                        return d
        return []

    def _step_verify(self, batch: ScheduleBatch, drafts: dict,
                     qs: dict | None = None):
        """One chunked-prefill forward verifies each sequence's draft: chunk
        = [last_token] + draft at positions n-1..n-1+k.  Greedy sequences:
        row j's argmax is the model's true greedy token after consuming
        draft[:j]; the longest matching prefix is accepted plus one bonus
        token.  Sampled sequences (temperature > 0) go through rejection
        sampling against the draft's q (spec_sampling.accept_resample),
        which keeps the emitted distribution EXACTLY the target's.
        Rejected draft positions leave garbage KV beyond the sequence
        length, which is overwritten before it can ever be read
        (attention is length-bounded)."""
        dev = self.device
        seqs = batch.seqs
        ids, pos, slots, q_start, bt_rows, chunks = [], [], [], [], [], []
        for seq in seqs:
            d = drafts.get(seq.seq_id) or []
            n = seq.num_tokens
            chunk = [seq.last_token] + d
            ids.extend(chunk)
            pos.extend(range(n - 1, n - 1 + len(chunk)))
            slots.extend(self._slot(seq, i)
                         for i in range(n - 1, n - 1 + len(chunk)))
            q_start.append(n - 1)
            chunks.append(len(chunk))
            row = torch.zeros(self.max_pages_per_seq, dtype=torch.int32)
            row[:len(seq.pages)] = torch.tensor(seq.pages, dtype=torch.int32)
            bt_rows.append(row)
            self.metrics["spec_drafted"] += len(d)
        cu_list = [0]
        for ln in chunks:
            cu_list.append(cu_list[-1] + ln)
        md = AttnMetadata(
            is_prefill=True,
            slots=torch.tensor(slots, dtype=torch.int64, device=dev),
            cu_seqlens=torch.tensor(cu_list, dtype=torch.int32, device=dev),
            seq_lens=chunks,
            q_start=torch.tensor(q_start, dtype=torch.int32, device=dev),
            block_table=torch.stack(bt_rows).to(dev))
        logits = self.model(torch.tensor(ids, dtype=torch.int32, device=dev),
                            torch.tensor(pos, dtype=torch.int32, device=dev),
                            self.kv, md)
        greedy_t = logits.argmax(dim=-1).to(torch.int64)
        greedy = greedy_t.cpu().tolist()
        tok_lists, lp_rows, lp_seqs = [], [], []
        sp_rows, sp_chosen, sp_seqs = [], [], []  # sampled-branch logprobs
        for i, seq in enumerate(seqs):
            t = greedy[cu_list[i]:cu_list[i + 1]]
            d = drafts.get(seq.seq_id) or []
            temp = seq.sampling.temperature
            if temp > 0.0:
                from .spec_sampling import accept_resample
                rows = logits[cu_list[i]:cu_list[i + 1]]
                p_rows = torch.softmax(rows.float() / temp, dim=-1)
                # d may be empty (capacity/budget-capped seq riding a
                # spec step): accept_resample degenerates to one sample
                # from p_0 — NOT the greedy path, which would silently
                # freeze this sequence's temperature to 0
                toks = accept_resample(p_rows, d,
                                       (qs or {}).get(seq.seq_id),
                                       self._spec_gen)
                tok_lists.append(toks)
                if self.draft is not None:
                    seq.draft_len = seq.num_tokens + len(toks) - 1
                self.metrics["spec_accepted"] += len(toks) - 1
                self.metrics["decode_tokens"] += len(toks)
                if seq.sampling.logprobs > 0:
                    for j, tk in enumerate(toks):
                        sp_rows.append(cu_list[i] + j)
                        sp_chosen.append(tk)
                        sp_seqs.append(seq)
                continue
            a = 0
            while a < len(d) and t[a] == d[a]:
                a += 1
            tok_lists.append(t[:a + 1])  # accepted prefix + bonus token
            if self.draft is not None:
                # draft KV valid through position n-1+a (last_token +
                # the accepted prefix); the bonus token catches up next
                # round
                seq.draft_len = seq.num_tokens + a
            self.metrics["spec_accepted"] += a
            self.metrics["decode_tokens"] += a + 1
            if seq.sampling.logprobs > 0:
                for j in range(a + 1):  # one helper row per emitted token
                    lp_rows.append(cu_list[i] + j)
                    lp_seqs.append(seq)
        if lp_rows:
            rows = torch.tensor(lp_rows, dtype=torch.int64,
                                device=logits.device)
            self._attach_row_logprobs(lp_seqs, logits[rows], greedy_t[rows])
        if sp_rows:
            rows = torch.tensor(sp_rows, dtype=torch.int64,
                                device=logits.device)
            chosen = torch.tensor(sp_chosen, dtype=torch.int64,
                                  device=logits.device)
            self._attach_row_logprobs(sp_seqs, logits[rows], chosen)
        return self._bookkeep_multi(seqs, tok_lists)

    def _bookkeep_multi(self, seqs, tok_lists):
        events = []
        now = time.monotonic_ns()
        for seq, toks in zip(seqs, tok_lists):
            for tok in toks:
                if not seq.output_ids:
                    seq.first_token_ns = now
                self.sched.note_token(seq)
                done = seq.append(tok, self.eos_id)
                if done:
                    seq.finish_ns = now
                    self.sched.finish(seq)
                    self._finished[seq.seq_id] = seq
                if self._emit(seq, done):
                    if seq.on_token is not None:
                        seq.on_token(tok, done)
                    events.append((seq.seq_id, tok, done))
                if done:
                    break
            if seq.logprobs is not None:
                # eos mid-acceptance: drop logprobs for discarded draft rows
                del seq.logprobs[len(seq.output_ids):]
        return events

    # -- decode path (graph-captured on GPU) --------------------------------
    def _fill_decode_buffers(self, seqs: list[Sequence], bs: int) -> None:
        h = self._host
        for i, seq in enumerate(seqs):
            n = seq.num_tokens
            h["ids"][i] = seq.last_token
            h["pos"][i] = n - 1
            h["slots"][i] = self._slot(seq, n - 1)
            h["lens"][i] = n
            npg = len(seq.pages)
            h["bt"][i, :npg] = torch.tensor(seq.pages, dtype=torch.int32)
            h["temps"][i] = seq.sampling.temperature
            h["topk"][i] = seq.sampling.top_k
            h["topp"][i] = seq.sampling.top_p
        for i in range(len(seqs), bs):  # dummy lanes -> null page 0
            h["ids"][i] = 0
            h["pos"][i] = 0
            h["slots"][i] = 0
            h["lens"][i] = 1
            h["bt"][i, 0] = 0
            h["temps"][i] = 0.0
            h["topk"][i] = 0
            h["topp"][i] = 1.0
        d = self._dec
        nb = self.is_gpu
        for k in ("ids", "pos", "slots", "lens", "temps", "topk", "topp"):
            d[k][:bs].copy_(h[k][:bs], non_blocking=nb)
        d["bt"][:bs].copy_(h["bt"][:bs], non_blocking=nb)

    def _decode_forward(self, bs: int, nsplit: int, scratch, tkp: bool,
                        lp: bool = False, mask: torch.Tensor | None = None):
        d = self._dec
        md = AttnMetadata(is_prefill=False, slots=d["slots"][:bs],
                          block_table=d["bt"][:bs], seq_lens_t=d["lens"][:bs],
                          nsplit=nsplit, decode_scratch=scratch)
        logits = self.model(d["ids"][:bs], d["pos"][:bs], self.kv, md)
        if mask is not None:  # grammar-constrained rows (eager only)
            logits = logits + mask.to(logits.dtype)
        kw = {"topk": d["topk"][:bs], "topp": d["topp"][:bs]} if tkp else {}
        ops.sample(logits, d["temps"][:bs], self.sampler,
                   out=d["tokens"][:bs], **kw)
        if lp:  # graph-capturable logprob reporting
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=-1)
            d["lp_tok"][:bs] = lf.gather(
                1, d["tokens"][:bs].long().unsqueeze(1)).squeeze(1) - lse
            tv, ti = lf.topk(LP_TOPN, dim=-1)
            d["lp_vals"][:bs] = tv - lse.unsqueeze(1)
            d["lp_ids"][:bs] = ti.to(torch.int32)

    def _make_scratch(self, bs: int, nsplit: int):
        if nsplit <= 1:
            return None
        cfg = self.cfg
        G = cfg.num_heads // cfg.num_kv_heads
        po = torch.empty(bs, cfg.num_kv_heads, nsplit, G, cfg.head_dim,
                         dtype=torch.float32, device=self.device)
        pml = torch.empty(bs, cfg.num_kv_heads, nsplit, G, 2,
                          dtype=torch.float32, device=self.device)
        return po, pml

    def _get_graph(self, bs: int, tkp: bool, lp: bool = False):
        key = (bs, tkp, lp)
        entry = self._graphs.get(key)
        if entry is not None:
            return entry
        nsplit = choose_nsplit(bs, self.cfg.num_kv_heads)
        scratch = self._make_scratch(bs, nsplit)
        # warm up allocator/kernels on a side stream, then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._decode_forward(bs, nsplit, scratch, tkp, lp)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._decode_forward(bs, nsplit, scratch, tkp, lp)
        entry = {"graph": g, "nsplit": nsplit, "scratch": scratch}
        self._graphs[key] = entry
        return entry

    def _attach_decode_logprobs(self, seqs: list[Sequence], n: int) -> None:
        d, h = self._dec, self._host
        if self.is_gpu:
            for k in ("lp_tok", "lp_vals", "lp_ids"):
                h[k][:n].copy_(d[k][:n], non_blocking=True)
            torch.cuda.current_stream().synchronize()
            src = h
        else:
            src = d
        vals, ids_, tokl = (src["lp_vals"][:n].tolist(),
                            src["lp_ids"][:n].tolist(),
                            src["lp_tok"][:n].tolist())
        for i, seq in enumerate(seqs):
            nreq = seq.sampling.logprobs
            if nreq <= 0:
                continue
            if seq.logprobs is None:
                seq.logprobs = []
            top = list(zip(ids_[i][:nreq], vals[i][:nreq]))
            seq.logprobs.append({"logprob": tokl[i], "top": top})

    def _step_decode(self, batch: ScheduleBatch) -> list[int]:
        seqs = batch.seqs
        n = len(seqs)
        self.metrics["decode_tokens"] += n
        bs = _bucket_for(n, self.max_num_seqs)
        tkp = any(s.sampling.top_k > 0 or s.sampling.top_p < 1.0 for s in seqs)
        lp = any(s.sampling.logprobs > 0 for s in seqs)
        jm = any(s.sampling.json_mode for s in seqs)
        self._fill_decode_buffers(seqs, bs)
        if self.enable_graphs and not jm:
            self._get_graph(bs, tkp, lp)["graph"].replay()
        else:
            # json grammar masks are data-dependent -> eager decode
            mask = None
            if jm:
                mask = torch.zeros(bs, self.cfg.vocab_size,
                                   device=self.device)
                mask[:n] = self._json_mask(seqs)
            nsplit = choose_nsplit(bs, self.cfg.num_kv_heads) if self.is_gpu else 1
            self._decode_forward(bs, nsplit, self._make_scratch(bs, nsplit),
                                 tkp, lp, mask)
        if lp:
            self._attach_decode_logprobs(seqs, n)
        toks = self._dec["tokens"][:n]
        if self.is_gpu:
            self._host["tokens"][:n].copy_(toks, non_blocking=True)
            torch.cuda.current_stream().synchronize()
            return self._host["tokens"][:n].tolist()
        return toks.tolist()

    # ------------------------------------------------------------- generate
    def generate(self, prompts: list[list[int]],
                 sampling: SamplingParams | None = None) -> list[list[int]]:
        """Synchronous batch generate (used by tests and app.ai())."""
        idmap = {}
        for p in prompts:
            rid = self.add_request(p, sampling)
            if rid is None:
                raise RuntimeError("engine queue full")
            idmap[rid] = None
        while self.has_work() and any(v is None for v in idmap.values()):
            self.step()
            for rid in list(idmap):
                if idmap[rid] is None:
                    fin = self.get_finished(rid)
                    if fin is not None:
                        idmap[rid] = fin.output_ids
        return [idmap[rid] for rid in sorted(idmap)]
