"""Unit + fuzz coverage for the byte-level JSON grammar automaton
(engine/jsonfsm.py) that backs grammar-constrained decoding."""
import json
import random

import pytest

from agentfield_amd.engine.jsonfsm import BYTE_OFFSET, EOS_ID, JsonFSM

VALID_DOCS = [
    '{"a": 1, "b": [true, null, -2.5e+3], "c": {"d": ""}}',
    "[]", "{}", '[[], {}, "x\\u00e9\\n", 0.1]', "-0.5", "true",
    '  {"k" : [ 1 , 2 ]}  ', '"hi"', "0", "123", "1e9", "0.25",
    '{"nested": {"deep": [{"x": [""]}]}}', "null", "false",
]


@pytest.mark.parametrize("doc", VALID_DOCS)
def test_accepts_valid_json(doc):
    f = JsonFSM()
    for byte in doc.encode():
        assert byte in f._allowed_raw(), (doc, chr(byte), f.state)
        f.advance(byte)
    assert f.complete(), (doc, f.state, f.stack)


def test_rejects_invalid_transitions():
    f = JsonFSM()
    f.advance(ord("{"))
    assert ord(",") not in f._allowed_raw()   # no leading comma
    assert ord("}") in f._allowed_raw()       # empty object ok
    f2 = JsonFSM()
    f2.advance(ord("["))
    f2.advance(ord("1"))
    f2.advance(ord(","))
    assert ord("]") not in f2._allowed_raw()  # trailing comma banned
    f3 = JsonFSM()
    f3.advance(ord("0"))
    assert ord("1") not in f3._allowed_raw()  # leading-zero numbers banned


def test_budget_walks_always_complete():
    """Random walks through allowed_token_ids(budget) must always yield a
    parseable document within the budget — the completion-forcing filter
    can never strand an open structure."""
    rng = random.Random(0)
    for _ in range(3000):
        f = JsonFSM()
        budget = rng.randint(1, 48)
        out = bytearray()
        while True:
            ids = f.allowed_token_ids(budget - len(out))
            assert ids, (bytes(out), f.state, f.stack, budget)
            tid = rng.choice(ids)
            if tid == EOS_ID:
                break
            out.append(tid - BYTE_OFFSET)
            f.advance(out[-1])
            if len(out) >= budget:
                break
        json.loads(out.decode("utf-8", errors="replace").strip())


def test_eos_only_when_complete():
    f = JsonFSM()
    assert EOS_ID not in f.allowed_token_ids(10)
    f.advance(ord("7"))
    assert EOS_ID in f.allowed_token_ids(10)   # bare number is complete
    f2 = JsonFSM()
    for b in b'{"a':
        f2.advance(b)
    assert EOS_ID not in f2.allowed_token_ids(10)
