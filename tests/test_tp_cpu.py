"""Tensor-parallel correctness on CPU: world_size=2 over gloo.
TP=2 sharded model must reproduce the TP=1 model's logits and greedy tokens.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp

from agentfield_amd.models.llama import LlamaConfig

TP_CFG = LlamaConfig(name="tiny-tp", hidden_size=512, intermediate_size=1024,
                     num_layers=2, num_heads=4, num_kv_heads=2,
                     vocab_size=512, max_position=256)


def _run_rank(rank, world, port, fn_name, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        if rank == 0:
            out_q.put(("ok", result))
    except Exception as e:
        import traceback
        if rank == 0:
            out_q.put(("err", f"{e}\n{traceback.format_exc()}"))
        raise
    finally:
        dist.destroy_process_group()


def _spawn(fn_name):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29612 + abs(hash(fn_name)) % 500
    procs = [ctx.Process(target=_run_rank, args=(r, 2, port, fn_name, q))
             for r in range(2)]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=300)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    return payload


def _logits_case(rank, world):
    from agentfield_amd.models.llama import AttnMetadata, KVCache
    from agentfield_amd.parallel import build_tp_model
    import torch.distributed as dist

    torch.manual_seed(0)
    dev = "cpu"
    model = build_tp_model(TP_CFG, world, rank, dev, dtype=torch.float32,
                           group=None, base_seed=7)
    T = 12
    ids = torch.randint(0, TP_CFG.vocab_size, (T,), dtype=torch.int32)
    pos = torch.arange(T, dtype=torch.int32)
    kv = KVCache(model.cfg, 32, 4, dev, torch.float32)
    md = AttnMetadata(is_prefill=True, slots=torch.arange(T, dtype=torch.int64),
                      cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
                      seq_lens=[T],
                      q_start=torch.zeros(1, dtype=torch.int32),
                      block_table=torch.arange(32, dtype=torch.int32)[None, :])
    # vocab-parallel lm_head must actually be sharded...
    assert model.lm_head.shape[0] == TP_CFG.vocab_size // world
    logits = model(ids, pos, kv, md)
    # ...while gathered logits cover the full vocab
    assert logits.shape[-1] == TP_CFG.vocab_size

    # reference: full (TP=1) model, same deterministic weights
    full = build_tp_model(TP_CFG, 1, 0, dev, dtype=torch.float32, base_seed=7)
    kv2 = KVCache(full.cfg, 32, 4, dev, torch.float32)
    want = full(ids, pos, kv2, md)
    diff = (logits - want).abs().max().item()
    assert diff < 1e-3, f"rank {rank}: TP logits diverge, max diff {diff}"
    return diff


def _generate_case(rank, world):
    from agentfield_amd.engine import SamplingParams
    from agentfield_amd.parallel import TPEngineGroup

    grp = TPEngineGroup(TP_CFG, "cpu", dtype=torch.float32, base_seed=7,
                        num_pages=64, page_size=4, max_num_seqs=4,
                        enable_graphs=False)
    prompts = [[1, 5, 9, 20], [3, 7, 2]]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    rids = grp.broadcast_and_submit([(p, sp) for p in prompts]
                                    if rank == 0 else None)
    outs = {r: None for r in rids}
    for _ in range(200):
        if not grp.has_work():
            break
        grp.step()
        for r in rids:
            if outs[r] is None:
                fin = grp.get_finished(r)
                if fin is not None:
                    outs[r] = fin.output_ids
    assert all(v is not None for v in outs.values())

    # single-rank reference
    from agentfield_amd.engine import LLMEngine
    from agentfield_amd.parallel import build_tp_model
    full = build_tp_model(TP_CFG, 1, 0, "cpu", dtype=torch.float32, base_seed=7)
    eng = LLMEngine(full.cfg, device="cpu", dtype=torch.float32, model=full,
                    num_pages=64, page_size=4, max_num_seqs=4,
                    enable_graphs=False)
    want = eng.generate(prompts, sp)
    got = [outs[r] for r in rids]
    assert got == want, f"rank {rank}: TP tokens {got} != {want}"
    return got


MOE_CFG = LlamaConfig(name="tiny-moe-tp", hidden_size=512,
                      intermediate_size=1024, num_layers=2, num_heads=4,
                      num_kv_heads=2, vocab_size=512, max_position=256,
                      num_experts=4, num_experts_per_tok=2)


def _moe_logits_case(rank, world):
    from agentfield_amd.models.llama import AttnMetadata, KVCache
    from agentfield_amd.parallel import build_tp_model

    torch.manual_seed(0)
    model = build_tp_model(MOE_CFG, world, rank, "cpu", dtype=torch.float32,
                           group=None, base_seed=11)
    # expert FFNs sharded, router replicated
    assert model.layers[0].mlp.gate_up.shape == \
        (4, 2 * MOE_CFG.intermediate_size // world, MOE_CFG.hidden_size)
    assert model.layers[0].mlp.router.shape == (4, MOE_CFG.hidden_size)
    T = 10
    ids = torch.randint(0, MOE_CFG.vocab_size, (T,), dtype=torch.int32)
    pos = torch.arange(T, dtype=torch.int32)
    md = AttnMetadata(is_prefill=True,
                      slots=torch.arange(T, dtype=torch.int64),
                      cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
                      seq_lens=[T],
                      q_start=torch.zeros(1, dtype=torch.int32),
                      block_table=torch.arange(32, dtype=torch.int32)[None, :])
    logits = model(ids, pos, KVCache(model.cfg, 32, 4, "cpu", torch.float32),
                   md)
    full = build_tp_model(MOE_CFG, 1, 0, "cpu", dtype=torch.float32,
                          base_seed=11)
    want = full(ids, pos, KVCache(full.cfg, 32, 4, "cpu", torch.float32), md)
    diff = (logits - want).abs().max().item()
    assert diff < 1e-3, f"rank {rank}: MoE TP logits diverge, max {diff}"
    return diff


def test_request_wire_codec_roundtrip():
    """The TP request broadcast codec must round-trip every sampling
    field exactly (no pickle on the hot path — VERDICT r1)."""
    from agentfield_amd.engine import SamplingParams
    from agentfield_amd.parallel.wire import (decode_requests,
                                              encode_requests)
    reqs = [
        ([1, 2, 3], SamplingParams()),
        ([9] * 50, SamplingParams(max_tokens=7, temperature=0.85,
                                  top_k=40, top_p=0.95,
                                  stop_token_ids=(2, 17),
                                  ignore_eos=True, logprobs=3,
                                  json_mode=True)),
        ([], SamplingParams(max_tokens=1)),
    ]
    out = decode_requests(encode_requests(reqs))
    assert len(out) == len(reqs)
    for (ids, sp), (ids2, sp2) in zip(reqs, out):
        assert ids == ids2
        assert sp2.max_tokens == sp.max_tokens
        assert abs(sp2.temperature - sp.temperature) < 1e-5
        assert sp2.top_k == sp.top_k
        assert abs(sp2.top_p - sp.top_p) < 1e-5
        assert tuple(sp2.stop_token_ids) == tuple(sp.stop_token_ids)
        assert sp2.ignore_eos == sp.ignore_eos
        assert sp2.logprobs == sp.logprobs
        assert sp2.json_mode == sp.json_mode
    assert encode_requests([]).tolist() == [0]


def _incremental_submit_case(rank, world):
    """Requests submitted over multiple steps (the serving pattern):
    empty broadcasts must be cheap no-ops and late joiners must land on
    every rank identically."""
    from agentfield_amd.engine import SamplingParams
    from agentfield_amd.parallel import TPEngineGroup

    grp = TPEngineGroup(TP_CFG, "cpu", dtype=torch.float32, base_seed=7,
                        num_pages=64, page_size=4, max_num_seqs=4,
                        enable_graphs=False)
    sp = SamplingParams(max_tokens=5, ignore_eos=True)
    rids = grp.broadcast_and_submit([([1, 5, 9], sp)]
                                    if rank == 0 else None)
    outs = {}
    joined = False
    for step in range(300):
        if not grp.has_work() and joined:
            break
        grp.step()
        new = grp.broadcast_and_submit(
            [([3, 7, 2, 11], sp)] if (rank == 0 and step == 2) else None)
        if new:
            rids += new
            joined = True
        elif rank != 0 and step == 2:
            joined = True
        for r in rids:
            if r not in outs:
                fin = grp.get_finished(r)
                if fin is not None:
                    outs[r] = fin.output_ids
    assert len(outs) == 2 and all(len(v) == 5 for v in outs.values())
    return sorted(outs.items())


def test_tp2_incremental_submit():
    _spawn("_incremental_submit_case")


def test_tp2_logits_match_tp1():
    _spawn("_logits_case")


def test_tp2_moe_logits_match_tp1():
    _spawn("_moe_logits_case")


def test_tp2_generate_matches_tp1():
    _spawn("_generate_case")


def test_request_wire_codec_carries_schema():
    """json_schema travels on the TP wire so every rank masks identically
    (ranks sample with the same seed; divergent masks would fork the
    token stream)."""
    from agentfield_amd.engine import SamplingParams
    from agentfield_amd.engine.schemafsm import SchemaSpec
    from agentfield_amd.parallel.wire import decode_requests, encode_requests
    schema = {"type": "object",
              "properties": {"a": {"type": "integer"},
                             "b": {"enum": ["x", 2]}},
              "required": ["a"]}
    reqs = [([1, 2, 3], SamplingParams(max_tokens=8, json_mode=True,
                                       json_schema=schema)),
            ([4], SamplingParams(max_tokens=4)),
            ([5, 6], SamplingParams(max_tokens=4, json_mode=True,
                                    json_schema=SchemaSpec(schema)))]
    out = decode_requests(encode_requests(reqs))
    assert out[0][1].json_schema == schema
    assert out[1][1].json_schema is None
    assert out[2][1].json_schema == schema  # SchemaSpec round-trips via source
    # property order (mask construction depends on it) survives the wire
    assert list(out[0][1].json_schema["properties"]) == ["a", "b"]
