"""Token-level JSON grammar masking for multi-byte (HF/BPE) vocabularies.

The byte tokenizer gets exact JSON masking directly from the byte PDA
(jsonfsm.py).  Real vocabularies map one token to SEVERAL bytes, so a
token is admissible iff the PDA accepts its ENTIRE byte string and the
value remains closable within the remaining token budget.  This module
compiles that per decode step by walking a byte trie of the vocabulary
with cloned PDA states, pruning at the first illegal byte.

Cost model: one trie walk per UNIQUE PDA signature (state, stack,
in-literal progress, budget class); results are memoized, and a decode
step inside a long string or number re-uses the cached mask, so the walk
cost amortizes to O(1) per token after warmup.  The budget enters the
signature only until it stops binding (min_close is bounded by
stack depth + a small constant), keeping the cache finite.

Completion guarantee: a token is admitted only if min_close(after it)
<= remaining-1, counting ONE byte of progress per future token — safe
for any vocabulary that contains single-byte fallback tokens for the
structural bytes (true of Llama-3/GPT byte-level BPE vocabularies; the
builder verifies this and refuses otherwise).
"""
from __future__ import annotations

from .jsonfsm import JsonFSM

# the bytes min-close completions are built from: closers, string quote,
# key scaffold ':','0', and the tails of true/false/null
_STRUCTURAL = b'"}]:0rueals'


class _Node:
    __slots__ = ("edges", "tokens")

    def __init__(self):
        self.edges: dict[int, _Node] = {}
        self.tokens: list[int] = []


class TokenJsonGrammar:
    def __init__(self, vocab: list[bytes | None], eos_id: int):
        """vocab[token_id] = the token's byte string (None = never legal,
        e.g. special tokens)."""
        self.eos_id = eos_id
        self.vocab = vocab
        self.root = _Node()
        singles = set()
        for tid, bs in enumerate(vocab):
            if not bs:
                continue
            node = self.root
            for b in bs:
                nxt = node.edges.get(b)
                if nxt is None:
                    nxt = node.edges[b] = _Node()
                node = nxt
            node.tokens.append(tid)
            if len(bs) == 1:
                singles.add(bs[0])
        missing = [bytes([b]) for b in _STRUCTURAL if b not in singles]
        if missing:
            raise ValueError(
                "vocabulary lacks single-byte fallback tokens for "
                f"{missing}; the budget guarantee would not hold")
        self._mask_cache: dict[tuple, list[int]] = {}

    # ------------------------------------------------------------- engine API
    def advance_token(self, fsm: JsonFSM, tid: int) -> None:
        bs = self.vocab[tid] if 0 <= tid < len(self.vocab) else None
        if bs is None:
            raise ValueError(f"token {tid} has no byte mapping")
        for b in bs:
            fsm.advance(b)

    SCHEMA_BUDGET_CAP = 96

    @classmethod
    def _sig(cls, fsm, remaining: int) -> tuple:
        if hasattr(fsm, "sig"):
            # SchemaFSM: min_close is not stack-bounded, so masks for
            # remaining >= CAP are computed AT the cap (over-strict for
            # larger budgets, never under-strict) and shared
            return (fsm.sig(), min(remaining, cls.SCHEMA_BUDGET_CAP))
        # JsonFSM: the budget stops binding once it exceeds any
        # reachable min_close
        cap = len(fsm.stack) + 12
        return (fsm.state, tuple(fsm.stack), fsm.key_str, fsm.lit,
                fsm.hex_left, fsm.utf_left, fsm.utf_lo, fsm.utf_hi,
                min(remaining, cap))

    def allowed_token_ids(self, fsm, remaining: int) -> list[int]:
        if hasattr(fsm, "sig"):
            remaining = min(remaining, self.SCHEMA_BUDGET_CAP)
        sig = self._sig(fsm, remaining)
        hit = self._mask_cache.get(sig)
        if hit is not None:
            return hit
        out: list[int] = [self.eos_id] if fsm.complete() else []
        budget = remaining - 1
        stack = [(self.root, fsm)]
        while stack:
            node, f = stack.pop()
            for b, child in node.edges.items():
                f2 = f.clone()
                try:
                    f2.advance(b)
                except (ValueError, AssertionError):
                    continue
                if child.tokens and f2.min_close() <= budget:
                    out.extend(child.tokens)
                if child.edges:
                    stack.append((child, f2))
        self._mask_cache[sig] = out
        return out


def vocab_bytes_from_hf(tok, vocab_size: int) -> list[bytes | None]:
    """Token-id -> byte-string table for an HF tokenizer.  Special tokens
    (and ids that decode to nothing) map to None so the grammar never
    admits them."""
    special = set(getattr(tok, "all_special_ids", []) or [])
    out: list[bytes | None] = [None] * vocab_size
    for tid in range(vocab_size):
        if tid in special:
            continue
        try:
            s = tok.decode([tid])
        except Exception:
            continue
        if s:
            out[tid] = s.encode("utf-8")
    return out
