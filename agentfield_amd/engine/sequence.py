"""Request/sequence state for the serving engine."""
from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Callable


@dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0
    top_k: int = 0          # 0 = disabled
    top_p: float = 1.0      # >= 1 = disabled
    stop_token_ids: tuple = ()
    ignore_eos: bool = False
    logprobs: int = 0       # >0: report chosen-token logprob + top-N
                            # alternatives per emitted token (N <= 8)
    json_mode: bool = False  # grammar-constrained valid-JSON decoding
                             # (engine/jsonfsm.py / token_grammar.py)
    json_schema: object = None  # SchemaSpec (or raw dict): constrain to a
                                # JSON-Schema subset, not just syntax
                                # (engine/schemafsm.py)


class SeqStatus(enum.Enum):
    WAITING = 0
    RUNNING = 1
    FINISHED = 2
    PREEMPTED = 3


@dataclass
class Sequence:
    seq_id: int
    prompt_ids: list[int]
    sampling: SamplingParams
    status: SeqStatus = SeqStatus.WAITING
    num_prefilled: int = 0
    alloc_epoch: int = 0   # bumped on each (re)admission page allocation
    output_ids: list[int] = field(default_factory=list)
    pages: list[int] = field(default_factory=list)
    freed_pages: int = 0   # leading pages reclaimed by the rolling window
    draft_len: int = 0     # tokens whose KV the spec DRAFT model holds
    finish_reason: str | None = None
    # per output token, when sampling.logprobs > 0:
    # {"logprob": float, "top": [(token_id, logprob), ...]}
    logprobs: list | None = None
    on_token: Callable | None = None     # streaming callback (token_id, done)
    arrival_ns: int = 0
    first_token_ns: int = 0
    finish_ns: int = 0

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)

    @property
    def last_token(self) -> int:
        return self.output_ids[-1] if self.output_ids else self.prompt_ids[-1]

    def on_preempt(self) -> None:
        """Preemption keeps already-generated tokens (vLLM-style recompute):
        on re-admission the prompt AND retained outputs re-prefill, then
        decoding continues from where it stopped.  Streamed output therefore
        never diverges from final output_ids, even for sampled sequences —
        dropping tokens and resampling would splice old-sample prefix with
        new-sample suffix for temperature>0 streams."""
        self.num_prefilled = 0
        self.cached_prefix = 0
        self.draft_len = 0  # draft KV pages were released with the rest

    def append(self, tok: int, eos_id: int) -> bool:
        """Append a generated token; returns True when the sequence finished."""
        self.output_ids.append(tok)
        sp = self.sampling
        if not sp.ignore_eos and (tok == eos_id or tok in sp.stop_token_ids):
            self.finish_reason = "stop"
            return True
        if len(self.output_ids) >= sp.max_tokens:
            self.finish_reason = "length"
            return True
        return False
