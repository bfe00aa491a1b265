"""Model checkpoint save/load (safetensors -> HBM).

Weight names use the HF Llama convention so real Llama-3 checkpoints load
directly; sharded-directory checkpoints (model-0000x-of-0000y.safetensors +
index json) are supported.  TP ranks load their slice without materializing
the full tensor on device.
"""
from __future__ import annotations

import json
from pathlib import Path

import torch
from safetensors import safe_open
from safetensors.torch import save_file

from .llama import LlamaConfig, LlamaForCausalLM

# our param name -> HF name pattern
_HF_MAP = {
    "embed": "model.embed_tokens.weight",
    "final_norm": "model.norm.weight",
    "lm_head": "lm_head.weight",
}


def _hf_layer_names(i: int) -> dict:
    p = f"model.layers.{i}."
    return {
        f"layers.{i}.input_norm": p + "input_layernorm.weight",
        f"layers.{i}.post_norm": p + "post_attention_layernorm.weight",
        f"layers.{i}.attn.qkv": (p + "self_attn.q_proj.weight",
                                 p + "self_attn.k_proj.weight",
                                 p + "self_attn.v_proj.weight"),
        f"layers.{i}.attn.o": p + "self_attn.o_proj.weight",
        f"layers.{i}.mlp.gate_up": (p + "mlp.gate_proj.weight",
                                    p + "mlp.up_proj.weight"),
        f"layers.{i}.mlp.down": p + "mlp.down_proj.weight",
    }


def save_checkpoint(model: LlamaForCausalLM, path: str) -> None:
    """Save in HF-compatible safetensors (single file + config.json)."""
    cfg = model.cfg
    if cfg.lm_vocab_rows is not None:
        raise ValueError("save_checkpoint expects an unsharded model; "
                         "TP ranks hold weight slices, not the full model")
    root = Path(path)
    root.mkdir(parents=True, exist_ok=True)
    tensors = {}
    sd = {k: v for k, v in model.state_dict().items()}
    for ours, hf in _HF_MAP.items():
        tensors[hf] = sd[ours].cpu().contiguous()
    moe = cfg.num_experts > 1
    for i in range(cfg.num_layers):
        for ours, hf in _hf_layer_names(i).items():
            if moe and (".mlp.gate_up" in ours or ".mlp.down" in ours):
                continue  # emitted in Mixtral layout below
            t = sd[ours].cpu()
            if isinstance(hf, tuple):
                if "qkv" in ours:
                    q, k, v = t.split([cfg.q_size, cfg.kv_size, cfg.kv_size], 0)
                    for name, part in zip(hf, (q, k, v)):
                        tensors[name] = part.contiguous()
                else:  # gate_up
                    g, u = t.chunk(2, 0)
                    tensors[hf[0]] = g.contiguous()
                    tensors[hf[1]] = u.contiguous()
            else:
                tensors[hf] = t.contiguous()
        if moe:  # HF Mixtral layout: block_sparse_moe.gate + experts.N.w1/2/3
            p = f"model.layers.{i}.block_sparse_moe."
            tensors[p + "gate.weight"] = \
                sd[f"layers.{i}.mlp.router"].cpu().contiguous()
            gu = sd[f"layers.{i}.mlp.gate_up"].cpu()
            dn = sd[f"layers.{i}.mlp.down"].cpu()
            for e in range(cfg.num_experts):
                g, u = gu[e].chunk(2, 0)
                tensors[p + f"experts.{e}.w1.weight"] = g.contiguous()
                tensors[p + f"experts.{e}.w3.weight"] = u.contiguous()
                tensors[p + f"experts.{e}.w2.weight"] = dn[e].contiguous()
    save_file(tensors, str(root / "model.safetensors"))
    cfg_json = {
        "architectures": ["MixtralForCausalLM" if moe else
                          "LlamaForCausalLM"],
        "hidden_size": cfg.hidden_size,
        "intermediate_size": cfg.intermediate_size,
        "num_hidden_layers": cfg.num_layers,
        "num_attention_heads": cfg.num_heads,
        "num_key_value_heads": cfg.num_kv_heads,
        "head_dim": cfg.head_dim,
        "vocab_size": cfg.vocab_size,
        "rope_theta": cfg.rope_theta,
        "max_position_embeddings": cfg.max_position,
        "rms_norm_eps": cfg.rms_eps,
    }
    if moe:
        cfg_json["num_local_experts"] = cfg.num_experts
        cfg_json["num_experts_per_tok"] = cfg.num_experts_per_tok
    (root / "config.json").write_text(json.dumps(cfg_json, indent=2))


def config_from_dir(path: str) -> LlamaConfig:
    c = json.loads((Path(path) / "config.json").read_text())
    return LlamaConfig(
        name=Path(path).name,
        hidden_size=c["hidden_size"],
        intermediate_size=c["intermediate_size"],
        num_layers=c["num_hidden_layers"],
        num_heads=c["num_attention_heads"],
        num_kv_heads=c.get("num_key_value_heads", c["num_attention_heads"]),
        head_dim=c.get("head_dim",
                       c["hidden_size"] // c["num_attention_heads"]),
        vocab_size=c["vocab_size"],
        rope_theta=c.get("rope_theta", 500000.0),
        max_position=c.get("max_position_embeddings", 8192),
        rms_eps=c.get("rms_norm_eps", 1e-5),
        num_experts=c.get("num_local_experts", 1),
        num_experts_per_tok=c.get("num_experts_per_tok", 2))


class _ShardedReader:
    """Reads tensors across one or many safetensors files."""

    def __init__(self, path: Path):
        idx = path / "model.safetensors.index.json"
        self.path = path
        if idx.exists():
            self.weight_map = json.loads(idx.read_text())["weight_map"]
        else:
            files = sorted(path.glob("*.safetensors"))
            self.weight_map = {}
            for f in files:
                with safe_open(str(f), framework="pt") as h:
                    for k in h.keys():
                        self.weight_map[k] = f.name
        self._open = {}

    def get(self, name: str, sl=None) -> torch.Tensor:
        f = self.weight_map[name]
        if f not in self._open:
            self._open[f] = safe_open(str(self.path / f), framework="pt")
        h = self._open[f]
        return h.get_slice(name)[sl] if sl is not None else h.get_tensor(name)


def load_checkpoint(model: LlamaForCausalLM, path: str,
                    tp: int = 1, rank: int = 0) -> LlamaForCausalLM:
    """Load HF-layout weights into the (possibly TP-sharded) model."""
    cfg_full = config_from_dir(path)
    cfg = model.cfg
    rd = _ShardedReader(Path(path))
    D = cfg_full.head_dim
    hq = cfg_full.num_heads // tp
    hk = cfg_full.num_kv_heads // tp
    inter = cfg_full.intermediate_size // tp
    dev = model.embed.device

    def rows(name, lo, hi):
        return rd.get(name, slice(lo, hi))

    with torch.no_grad():
        model.embed.copy_(rd.get(_HF_MAP["embed"]).to(dev, model.embed.dtype))
        model.final_norm.copy_(rd.get(_HF_MAP["final_norm"]).to(dev))
        lm = _HF_MAP["lm_head"]
        vs_rows = model.lm_head.shape[0]
        lm_sl = (slice(rank * vs_rows, (rank + 1) * vs_rows)
                 if vs_rows != cfg_full.vocab_size else None)
        if lm not in rd.weight_map:  # tied embeddings
            src = model.embed[lm_sl] if lm_sl is not None else model.embed
            model.lm_head.copy_(src)
        else:
            model.lm_head.copy_(
                rd.get(lm, lm_sl).to(dev, model.lm_head.dtype))
        for i in range(cfg.num_layers):
            p = f"model.layers.{i}."
            L = model.layers[i]
            L.input_norm.copy_(rd.get(p + "input_layernorm.weight").to(dev))
            L.post_norm.copy_(rd.get(p + "post_attention_layernorm.weight").to(dev))
            q = rows(p + "self_attn.q_proj.weight", rank * hq * D, (rank + 1) * hq * D)
            k = rows(p + "self_attn.k_proj.weight", rank * hk * D, (rank + 1) * hk * D)
            v = rows(p + "self_attn.v_proj.weight", rank * hk * D, (rank + 1) * hk * D)
            L.attn.qkv.copy_(torch.cat([q, k, v], 0).to(dev, L.attn.qkv.dtype))
            o = rd.get(p + "self_attn.o_proj.weight")[:, rank * hq * D:(rank + 1) * hq * D]
            L.attn.o.copy_(o.to(dev, L.attn.o.dtype))
            if cfg_full.num_experts > 1:
                m = p + "block_sparse_moe."
                L.mlp.router.copy_(
                    rd.get(m + "gate.weight").to(dev, L.mlp.router.dtype))
                for e in range(cfg_full.num_experts):
                    g = rows(m + f"experts.{e}.w1.weight",
                             rank * inter, (rank + 1) * inter)
                    u = rows(m + f"experts.{e}.w3.weight",
                             rank * inter, (rank + 1) * inter)
                    L.mlp.gate_up[e].copy_(
                        torch.cat([g, u], 0).to(dev, L.mlp.gate_up.dtype))
                    d = rd.get(m + f"experts.{e}.w2.weight")[
                        :, rank * inter:(rank + 1) * inter]
                    L.mlp.down[e].copy_(d.to(dev, L.mlp.down.dtype))
            else:
                g = rows(p + "mlp.gate_proj.weight", rank * inter, (rank + 1) * inter)
                u = rows(p + "mlp.up_proj.weight", rank * inter, (rank + 1) * inter)
                L.mlp.gate_up.copy_(torch.cat([g, u], 0).to(dev, L.mlp.gate_up.dtype))
                d = rd.get(p + "mlp.down_proj.weight")[:, rank * inter:(rank + 1) * inter]
                L.mlp.down.copy_(d.to(dev, L.mlp.down.dtype))
    return model
