"""Token-level JSON grammar (engine/token_grammar.py): the valid-JSON
guarantee must hold for multi-byte (HF/BPE-style) vocabularies, not just
the byte tokenizer."""
import json
import random

import pytest
import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.engine.jsonfsm import JsonFSM  # noqa: F401
from agentfield_amd.engine.token_grammar import TokenJsonGrammar
from agentfield_amd.models import CONFIGS

# synthetic BPE-ish vocab aligned with the tiny engine (vocab_size 512):
# ids 0-3 specials, 4..259 all single bytes, 260+ multi-byte merges
MERGES = [b'{"', b'": ', b'", "', b'"}', b'true', b'false', b'null',
          b'{"a": ', b'[1, ', b'123', b'0.5', b'e-2', b'": [', b']}',
          b'xy', b'hello', b'\\"', b', ', b'["', b'"]']


def make_vocab(size=512):
    vocab = [None] * size
    for b in range(256):
        vocab[4 + b] = bytes([b])
    for i, m in enumerate(MERGES):
        vocab[260 + i] = m
    return vocab


class FakeBPE:
    """Quacks like the sdk.ai._HF wrapper for grammar purposes."""

    def __init__(self):
        self.vocab = make_vocab()
        self.eos_id = 2

    def decode(self, ids):
        return b"".join(self.vocab[i] or b"" for i in ids).decode(
            "utf-8", errors="replace")


def test_grammar_masks_fresh_state():
    g = TokenJsonGrammar(make_vocab(), eos_id=2)
    allowed = set(g.allowed_token_ids(JsonFSM(), remaining=64))
    assert 4 + ord("{") in allowed
    assert 260 + MERGES.index(b'{"') in allowed       # multi-byte opener
    assert 260 + MERGES.index(b'true') in allowed
    assert 4 + ord("}") not in allowed                # close before open
    assert 2 not in allowed                           # not complete yet
    # after '{' the '"}'-ish tokens become legal, ':' not yet
    f = JsonFSM()
    f.advance(ord("{"))
    allowed = set(g.allowed_token_ids(f, remaining=64))
    assert 4 + ord("}") in allowed
    assert 4 + ord(":") not in allowed


def test_grammar_budget_forces_closure():
    g = TokenJsonGrammar(make_vocab(), eos_id=2)
    f = JsonFSM()
    for b in b'{"a": [1':
        f.advance(b)
    # plenty of budget: can keep nesting
    wide = set(g.allowed_token_ids(f, remaining=64))
    assert 4 + ord(",") in wide
    # 3 tokens left: must close "]}" after this one -> no new nesting
    tight = set(g.allowed_token_ids(f, remaining=3))
    assert 4 + ord("[") not in tight
    assert 4 + ord("]") in tight
    assert 260 + MERGES.index(b']}') in tight


def test_grammar_mask_cache_reused():
    g = TokenJsonGrammar(make_vocab(), eos_id=2)
    f = JsonFSM()
    for b in b'{"key": "abc':
        f.advance(b)
    a1 = g.allowed_token_ids(f, remaining=60)
    n = len(g._mask_cache)
    f2 = f.clone()
    f2.advance(ord("d"))  # same signature: still mid-string
    a2 = g.allowed_token_ids(f2, remaining=59)
    assert len(g._mask_cache) == n, "signature cache missed"
    assert a1 == a2


def test_grammar_rejects_vocab_without_fallbacks():
    vocab = make_vocab()
    vocab[4 + ord("}")] = None  # no single '}' token
    with pytest.raises(ValueError):
        TokenJsonGrammar(vocab, eos_id=2)


def test_engine_json_mode_with_bpe_vocab():
    """End-to-end: every sampled output under json_mode with the
    multi-byte vocab decodes to valid JSON (the reference can only
    prompt-and-validate; this engine guarantees syntax)."""
    tok = FakeBPE()
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=128, max_num_seqs=4,
                    enable_graphs=False, seed=6)
    eng.set_token_grammar(TokenJsonGrammar(tok.vocab, eos_id=2))
    rng = random.Random(0)
    rids = [eng.add_request([1, 4 + rng.randrange(256), 9],
                            SamplingParams(max_tokens=24, temperature=0.9,
                                           json_mode=True))
            for _ in range(6)]
    outs = {}
    for _ in range(600):
        eng.step()
        for r in rids:
            if r not in outs:
                f = eng.get_finished(r)
                if f:
                    outs[r] = f.output_ids
        if len(outs) == len(rids):
            break
    assert len(outs) == len(rids)
    for r, ids in outs.items():
        body = ids[:-1] if ids and ids[-1] == 2 else ids
        text = tok.decode(body)
        json.loads(text)  # must parse — the whole point


def test_grammar_fuzz_always_parses():
    """Random walks sampling ONLY from allowed sets must always produce
    parseable JSON within budget — including multi-byte tokens that
    smuggle bytes the byte-mask path never offers (the GPU-found
    control-chars-in-string bug class)."""
    import random
    tok = FakeBPE()
    g = TokenJsonGrammar(tok.vocab, eos_id=2)
    rng = random.Random(7)
    for trial in range(200):
        f = JsonFSM()
        out = []
        remaining = rng.choice([6, 12, 24])
        budget = remaining
        while budget > 0:
            allowed = g.allowed_token_ids(f, budget)
            assert allowed, (trial, out, f.state, f.stack)
            t = rng.choice(allowed)
            if t == 2:
                break
            g.advance_token(f, t)
            out.append(t)
            budget -= 1
        text = tok.decode(out)
        json.loads(text)  # must parse, every time


def test_runner_attaches_grammar_for_hf_tokenizer():
    """app.ai(json_only=True) with an HF-style tokenizer compiles and
    attaches the token grammar instead of raising (round-1 behavior)."""
    from agentfield_amd.sdk.ai import AIConfig, EngineRunner

    class HFLike:
        def __init__(self):
            self.tok = FakeBPE()
            self.eos_id = 2

        def encode(self, text, add_bos=True):
            return [1] + [4 + b for b in text.encode()[:8]]

        def decode(self, ids):
            return self.tok.decode([i for i in ids if i > 3])

    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=128, max_num_seqs=4,
                    enable_graphs=False, seed=6)
    runner = EngineRunner(eng, HFLike())
    out = runner.generate_text("make json",
                               AIConfig(model="tiny", max_tokens=20,
                                        temperature=0.8, json_only=True))
    json.loads(out)
    assert eng.token_grammar is not None
    runner.shutdown()


def test_token_grammar_with_root_anyof():
    """MultiFSM (root anyOf) through the BPE token-trie walk: every walk
    yields a document matching one alternative."""
    import random as _random

    from agentfield_amd.engine.schemafsm import make_fsm
    tok = FakeBPE()
    g = TokenJsonGrammar(tok.vocab, eos_id=2)
    schema = {"anyOf": [
        {"type": "object", "properties": {"a": {"type": "integer"}},
         "required": ["a"]},
        {"type": "object", "properties": {"s": {"enum": ["x", "yy"]}},
         "required": ["s"]},
    ]}
    rng = _random.Random(13)
    for trial in range(30):
        f = make_fsm(schema)
        out = []
        budget = 24
        while budget > 0:
            ids = g.allowed_token_ids(f, budget)
            assert ids, (trial, out)
            t = rng.choice(ids)
            if t == 2:
                break
            g.advance_token(f, t)
            out.append(t)
            budget -= 1
        data = json.loads(tok.decode(out))
        assert ("a" in data and isinstance(data["a"], int)) or \
               ("s" in data and data["s"] in ("x", "yy"))
