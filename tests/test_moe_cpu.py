"""Sparse MoE (Mixtral-family) tests on CPU: layer math vs a manual
reference, engine E2E, norm folding exactness, checkpoint roundtrip."""
import pytest
import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models import CONFIGS
from agentfield_amd.models.llama import (AttnMetadata, KVCache,
                                         LlamaForCausalLM, MoEMLP)


def _md_prefill(T, dev="cpu"):
    return AttnMetadata(
        is_prefill=True, slots=torch.arange(T, dtype=torch.int64),
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32), seq_lens=[T],
        q_start=torch.zeros(1, dtype=torch.int32),
        block_table=torch.arange(64, dtype=torch.int32)[None, :])


def test_moe_layer_matches_manual_reference():
    cfg = CONFIGS["tiny-moe"]
    torch.manual_seed(0)
    mlp = MoEMLP(cfg)
    with torch.no_grad():
        for p in mlp.parameters():
            p.normal_(0, 0.2)
    x = torch.randn(9, cfg.hidden_size)
    got = mlp(x)

    # manual per-token reference
    probs = torch.softmax(x.float() @ mlp.router.float().t(), -1)
    want = torch.zeros_like(x)
    for t in range(x.shape[0]):
        topv, topi = probs[t].topk(cfg.num_experts_per_tok)
        topv = topv / topv.sum()
        acc = torch.zeros(cfg.hidden_size)
        for w, e in zip(topv.tolist(), topi.tolist()):
            g, u = (x[t] @ mlp.gate_up[e].t()).chunk(2)
            act = torch.nn.functional.silu(g) * u
            acc += w * (act @ mlp.down[e].t())
        want[t] = acc
    assert torch.allclose(got.float(), want.float(), atol=1e-4), \
        (got - want).abs().max()


def test_moe_static_dispatch_matches_exact():
    """The graph-capturable static-capacity decode dispatch must produce
    the same output as the exact per-expert loop (routing parity)."""
    cfg = CONFIGS["tiny-moe"]
    torch.manual_seed(3)
    mlp = MoEMLP(cfg)
    with torch.no_grad():
        for p in mlp.parameters():
            p.normal_(0, 0.2)
    for T in (1, 4, 33):
        x = torch.randn(T, cfg.hidden_size)
        exact = mlp(x, static=False)
        static = mlp(x, static=True)
        assert torch.allclose(static.float(), exact.float(), atol=1e-4),             (static - exact).abs().max()


def test_moe_router_uses_all_experts():
    cfg = CONFIGS["tiny-moe"]
    torch.manual_seed(1)
    mlp = MoEMLP(cfg)
    with torch.no_grad():
        for p in mlp.parameters():
            p.normal_(0, 0.2)
    x = torch.randn(256, cfg.hidden_size)
    probs = torch.softmax(x.float() @ mlp.router.float().t(), -1)
    _, topi = probs.topk(cfg.num_experts_per_tok, -1)
    assert len(set(topi.flatten().tolist())) == cfg.num_experts


def test_moe_engine_generates_deterministic():
    cfg = CONFIGS["tiny-moe"]
    prompts = [[1, 5, 9, 20], [3, 7, 2]]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    outs = []
    for _ in range(2):
        eng = LLMEngine(cfg, device="cpu", dtype=torch.float32, page_size=4,
                        num_pages=64, max_num_seqs=4, enable_graphs=False)
        outs.append(eng.generate(prompts, sp))
    assert outs[0] == outs[1]
    assert all(len(o) == 6 for o in outs[0])
    assert all(0 <= t < cfg.vocab_size for o in outs[0] for t in o)


def test_moe_norm_folding_exact():
    cfg = CONFIGS["tiny-moe"]
    T = 7
    torch.manual_seed(2)
    m = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32).init_random(4)
    with torch.no_grad():
        for layer in m.layers:
            layer.input_norm.normal_(1.0, 0.1)
            layer.post_norm.normal_(1.0, 0.1)
        m.final_norm.normal_(1.0, 0.1)
    ids = torch.randint(0, cfg.vocab_size, (T,), dtype=torch.int32)
    pos = torch.arange(T, dtype=torch.int32)
    base = m(ids, pos, KVCache(cfg, 64, 16, "cpu", torch.float32),
             _md_prefill(T)).float()
    m.fold_norm_weights()  # folds post_norm into experts AND router
    folded = m(ids, pos, KVCache(cfg, 64, 16, "cpu", torch.float32),
               _md_prefill(T)).float()
    assert torch.allclose(base, folded, atol=1e-3), \
        (base - folded).abs().max()


def test_moe_checkpoint_roundtrip(tmp_path):
    from agentfield_amd.models.checkpoint import (config_from_dir,
                                                  load_checkpoint,
                                                  save_checkpoint)
    cfg = CONFIGS["tiny-moe"]
    m1 = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32).init_random(9)
    save_checkpoint(m1, str(tmp_path / "moe"))
    rc = config_from_dir(str(tmp_path / "moe"))
    assert rc.num_experts == 4 and rc.num_experts_per_tok == 2
    m2 = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32)
    load_checkpoint(m2, str(tmp_path / "moe"))
    for (n1, p1), (_n2, p2) in zip(m1.named_parameters(),
                                   m2.named_parameters()):
        assert torch.equal(p1, p2), n1
    # Mixtral HF names present in the file
    from safetensors import safe_open
    with safe_open(str(tmp_path / "moe" / "model.safetensors"),
                   framework="pt") as h:
        keys = set(h.keys())
    assert "model.layers.0.block_sparse_moe.gate.weight" in keys
    assert "model.layers.1.block_sparse_moe.experts.3.w2.weight" in keys
    assert not any("mlp.gate_proj" in k for k in keys)
