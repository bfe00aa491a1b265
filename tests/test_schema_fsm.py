"""Schema-constrained decoding (engine/schemafsm.py): outputs must parse
AND conform — keys, types, enums, required properties, budget closure."""
import json
import random
import sys
from pathlib import Path

import pytest
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent))

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.engine.schemafsm import SchemaFSM, SchemaSpec
from agentfield_amd.engine.token_grammar import TokenJsonGrammar
from agentfield_amd.models import CONFIGS
from test_token_grammar import FakeBPE

SCHEMA = {
    "type": "object",
    "properties": {
        "name": {"type": "string"},
        "age": {"type": "integer"},
        "mood": {"enum": ["happy", "sad", 42]},
        "tags": {"type": "array", "items": {"type": "string"}},
        "ok": {"type": "boolean"},
        "meta": {},  # no type: any JSON value
    },
    "required": ["name", "age"],
}


def drive(fsm, data: bytes):
    for b in data:
        fsm.advance(b)
    return fsm


def test_schema_accepts_conforming_document():
    doc = (b'{"name": "bo", "age": -12, "mood": "sad", '
           b'"tags": ["x", "y"], "ok": true, "meta": {"z": [1, null]}}')
    f = drive(SchemaFSM(SchemaSpec(SCHEMA)), doc)
    assert f.complete()
    assert f.min_close() == 0


def test_schema_rejects_violations():
    spec = SchemaSpec(SCHEMA)
    cases = [
        b'{"nope',              # unknown key
        b'{"name": 5',          # wrong type for name
        b'{"age": "x',          # string where integer required
        b'{"age": 1.',          # float where integer required
        b'{"mood": "mad',       # not in enum
        b'{"tags": [1',         # array item type
        b'{"ok": null',         # null where boolean
        b'[1]'[0:1],            # top level must be an object -> '[' fails
        b'{"name": "a", "name',  # duplicate key
    ]
    for case in cases:
        f = SchemaFSM(spec)
        with pytest.raises(ValueError):
            drive(f, case)


def test_schema_required_enforced_on_close():
    spec = SchemaSpec(SCHEMA)
    f = drive(SchemaFSM(spec), b'{"name": "a"')
    with pytest.raises(ValueError):
        f.advance(ord("}"))  # age still missing
    f2 = drive(SchemaFSM(spec), b'{"name": "a", "age": 3')
    f2.advance(ord("}"))  # lazy number end -> pops, then close
    assert f2.complete()


def test_schema_byte_masks_and_budget():
    spec = SchemaSpec({"type": "object",
                       "properties": {"a": {"type": "integer"},
                                      "b": {"type": "string"}},
                       "required": ["a", "b"]})
    f = SchemaFSM(spec)
    ids = f.allowed_token_ids(64)
    ws = {b + 4 for b in b" \t\n\r"}
    assert set(ids) - ws == {ord("{") + 4}  # only the opening brace (+ws)
    # mid-key: only bytes extending candidate keys
    f = drive(SchemaFSM(spec), b'{"')
    allowed = {i - 4 for i in f.allowed_token_ids(64)}
    assert allowed == {ord("a"), ord("b")}  # keys only, no ws inside
    # tight budget: must pick the completion that still closes
    f = drive(SchemaFSM(spec), b'{"a": 1, "b": "x')
    # '"' then '}' = 2 bytes to close
    ids = f.allowed_token_ids(2)
    assert ord('"') + 4 in ids
    # a content byte would still need '"}' after -> excluded at budget 2
    assert ord("y") + 4 not in ids
    # EOS only when complete
    done = drive(SchemaFSM(spec), b'{"a": 1, "b": "x"}')
    assert done.complete()
    assert 2 in done.allowed_token_ids(5)


def test_schema_fuzz_conforms(tmp_path):
    """Random walks over byte masks always yield documents that parse
    AND validate against the schema subset."""
    spec = SchemaSpec(SCHEMA)
    rng = random.Random(3)
    for trial in range(120):
        f = SchemaFSM(spec)
        out = bytearray()
        budget = rng.choice([24, 40, 80])
        while budget > 0:
            ids = f.allowed_token_ids(budget)
            assert ids, (trial, bytes(out), f.stack)
            t = rng.choice(ids)
            if t == 2:
                break
            f.advance(t - 4)
            out.append(t - 4)
            budget -= 1
        data = json.loads(bytes(out))
        assert isinstance(data, dict)
        assert {"name", "age"} <= set(data)
        assert isinstance(data["name"], str)
        assert isinstance(data["age"], int)
        assert set(data) <= set(SCHEMA["properties"])
        if "mood" in data:
            assert data["mood"] in ("happy", "sad", 42)
        if "tags" in data:
            assert all(isinstance(x, str) for x in data["tags"])
        if "ok" in data:
            assert isinstance(data["ok"], bool)


def test_schema_token_grammar_bpe():
    """Schema constraints through the token-trie walk (BPE vocab)."""
    tok = FakeBPE()
    g = TokenJsonGrammar(tok.vocab, eos_id=2)
    spec = SchemaSpec({"type": "object",
                       "properties": {"a": {"type": "integer"}},
                       "required": ["a"]})
    rng = random.Random(5)
    for trial in range(40):
        f = SchemaFSM(spec)
        out = []
        budget = 20
        while budget > 0:
            ids = g.allowed_token_ids(f, budget)
            assert ids, (trial, out, f.stack)
            t = rng.choice(ids)
            if t == 2:
                break
            g.advance_token(f, t)
            out.append(t)
            budget -= 1
        data = json.loads(tok.decode(out))
        assert set(data) == {"a"} and isinstance(data["a"], int)


def test_engine_schema_mode_end_to_end():
    """Engine-level: sampled outputs under json_schema conform."""
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=128, max_num_seqs=4,
                    enable_graphs=False, seed=8)
    spec = SchemaSpec({"type": "object",
                       "properties": {"x": {"type": "integer"},
                                      "s": {"enum": ["a", "bb"]}},
                       "required": ["x"]})
    rids = [eng.add_request([1, 30 + i, 9],
                            SamplingParams(max_tokens=20, temperature=0.9,
                                           json_mode=True,
                                           json_schema=spec))
            for i in range(4)]
    outs = {}
    for _ in range(400):
        eng.step()
        for r in rids:
            if r not in outs:
                fin = eng.get_finished(r)
                if fin:
                    outs[r] = fin.output_ids
        if len(outs) == len(rids):
            break
    assert len(outs) == len(rids)
    for ids in outs.values():
        body = ids[:-1] if ids and ids[-1] == 2 else ids
        data = json.loads(bytes(b - 4 for b in body).decode())
        assert isinstance(data["x"], int)
        assert set(data) <= {"x", "s"}
        if "s" in data:
            assert data["s"] in ("a", "bb")


def test_ai_schema_guarantee():
    """app.ai(schema=...) conformance through the runner."""
    from agentfield_amd.sdk.ai import AIConfig, ByteTokenizer, EngineRunner
    eng = LLMEngine(CONFIGS["tiny"], device="cpu", dtype=torch.float32,
                    page_size=4, num_pages=128, max_num_seqs=4,
                    enable_graphs=False, seed=8)
    runner = EngineRunner(eng, ByteTokenizer(CONFIGS["tiny"].vocab_size))
    cfg = AIConfig(model="tiny", max_tokens=24, temperature=0.8,
                   json_only=True,
                   json_schema={"type": "object",
                                "properties": {"answer": {"type": "string"}},
                                "required": ["answer"]})
    out = runner.generate_text("question", cfg)
    data = json.loads(out)
    assert set(data) == {"answer"} and isinstance(data["answer"], str)
    runner.shutdown()


def test_root_anyof_tool_call_shapes():
    """Root-level anyOf (NFA of SchemaFSM alternatives): masks admit the
    union, advance prunes dead branches, completion/min_close take the
    best alternative — the shape tool-calling compiles to."""
    from agentfield_amd.engine.schemafsm import MultiFSM, make_fsm
    schema = {"anyOf": [
        {"type": "object",
         "properties": {"name": {"enum": ["get_time"]},
                        "arguments": {"type": "object",
                                      "properties": {"tz": {"type": "string"}},
                                      "required": ["tz"]}},
         "required": ["name", "arguments"]},
        {"type": "object",
         "properties": {"name": {"enum": ["add"]},
                        "arguments": {"type": "object",
                                      "properties": {"a": {"type": "integer"},
                                                     "b": {"type": "integer"}},
                                      "required": ["a", "b"]}},
         "required": ["name", "arguments"]},
    ]}
    f = make_fsm(schema)
    assert isinstance(f, MultiFSM)
    doc = b'{"name": "add", "arguments": {"a": 1, "b": -2}}'
    for b in doc:
        f.advance(b)
    assert f.complete()
    # pruning: after the discriminator only the matching branch survives
    f2 = make_fsm(schema)
    for b in b'{"name": "g':
        f2.advance(b)
    assert len(f2.alts) == 1
    with pytest.raises(ValueError):
        f2.advance(ord("x"))  # 'gx' matches no enum
    # fuzz: random mask walks always produce a doc matching ONE branch
    rng = random.Random(9)
    for _ in range(40):
        g = make_fsm(schema)
        out = bytearray()
        budget = 70
        while budget > 0:
            ids = g.allowed_token_ids(budget)
            assert ids
            t = rng.choice(ids)
            if t == 2:
                break
            g.advance(t - 4)
            out.append(t - 4)
            budget -= 1
        data = json.loads(bytes(out).decode("utf-8", errors="replace"))
        assert data["name"] in ("get_time", "add")
        if data["name"] == "add":
            assert set(data["arguments"]) <= {"a", "b"}
