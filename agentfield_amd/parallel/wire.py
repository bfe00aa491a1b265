"""Fixed-layout tensor codec for TP request sync.

The TP group must make identical scheduling decisions, so rank 0
broadcasts each batch of newly-submitted requests before stepping.
Pickle (`broadcast_object_list`) on the decode hot path costs Python
serialization per step at TP=8; this codec packs requests into one int32
tensor so the broadcast is a plain RCCL payload.

Layout: [n_requests] then per request a 10-word header
  [prompt_len, max_tokens, temp_u, top_k, top_p_u, ignore_eos,
   logprobs, json_mode, n_stop, n_schema]
followed by prompt ids, stop token ids, then n_schema words of
UTF-8 JSON-schema bytes (one byte per word; a per-request one-time
cost, not per-step).  Float fields are scaled by 1e6 (temperature/
top_p resolution far below sampling noise).  Every TP rank samples
with the same seed, so the schema must decode identically on all
ranks — json.dumps with sort_keys=False preserves the submitter's
property order, which the mask construction depends on.
"""
from __future__ import annotations

import json

import torch

from ..engine import SamplingParams

_SCALE = 1_000_000
_HDR = 10


def _schema_json(sp: SamplingParams) -> bytes:
    js = sp.json_schema
    if js is None:
        return b""
    src = getattr(js, "source", js)  # SchemaSpec keeps its input JSON
    return json.dumps(src, separators=(",", ":")).encode()


def encode_requests(requests: list[tuple[list[int], SamplingParams]]
                    ) -> torch.Tensor:
    words: list[int] = [len(requests)]
    for ids, sp in requests:
        stop = list(sp.stop_token_ids or ())
        sj = _schema_json(sp)
        words += [len(ids), sp.max_tokens, int(sp.temperature * _SCALE),
                  sp.top_k, int(sp.top_p * _SCALE), int(sp.ignore_eos),
                  sp.logprobs, int(sp.json_mode), len(stop), len(sj)]
        words += list(ids)
        words += stop
        words += list(sj)
    return torch.tensor(words, dtype=torch.int32)


def decode_requests(t: torch.Tensor
                    ) -> list[tuple[list[int], SamplingParams]]:
    w = t.tolist()
    n = w[0]
    out = []
    at = 1
    for _ in range(n):
        (plen, max_tokens, temp_u, top_k, top_p_u, ignore_eos,
         logprobs, json_mode, n_stop, n_schema) = w[at:at + _HDR]
        at += _HDR
        ids = w[at:at + plen]
        at += plen
        stop = tuple(w[at:at + n_stop])
        at += n_stop
        schema = None
        if n_schema:
            schema = json.loads(bytes(w[at:at + n_schema]).decode())
            at += n_schema
        sp = SamplingParams(max_tokens=max_tokens,
                            temperature=temp_u / _SCALE,
                            top_k=top_k, top_p=top_p_u / _SCALE,
                            stop_token_ids=stop,
                            ignore_eos=bool(ignore_eos),
                            logprobs=logprobs,
                            json_mode=bool(json_mode),
                            json_schema=schema)
        out.append((ids, sp))
    return out
