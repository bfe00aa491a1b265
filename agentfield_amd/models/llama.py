"""Llama-family decoder built directly on the MI355X-native op set.

Layout decisions (MI355X-first):
  * activations bf16 [T, H] with fp32 accumulation inside kernels
  * QKV and gate/up projections are fused single GEMMs (hipBLASLt via
    torch.matmul for the plain GEMMs — see ops.gemm_bf16 for the in-house
    MFMA kernel; fused hot ops are hand-written HIP)
  * attention reads/writes a paged KV cache sized for 288 GB HBM3E
  * prefill runs varlen-packed, decode runs one row per sequence and is
    hipGraph-capturable (no host-side data-dependent control flow)
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
from torch import nn

from .. import ops


@dataclass
class LlamaConfig:
    name: str = "custom"
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    max_position: int = 8192
    rms_eps: float = 1e-5
    tie_embeddings: bool = False
    # vocab-parallel lm_head: rows this rank holds (None -> full vocab).
    # Embedding stays replicated (lookup is local); logits shards are
    # all-gathered along the vocab dim so sampling sees full logits.
    lm_vocab_rows: int | None = None
    # sparse MoE (Mixtral-style): >1 experts replaces the dense FFN with a
    # softmax-routed top-k mixture; intermediate_size is PER EXPERT.
    num_experts: int = 1
    num_experts_per_tok: int = 2
    # sliding-window attention (Mistral): 0 = full causal; >0 = each token
    # attends to at most the last `sliding_window` positions (all layers)
    sliding_window: int = 0

    @property
    def q_size(self):
        return self.num_heads * self.head_dim

    @property
    def kv_size(self):
        return self.num_kv_heads * self.head_dim

    def shard(self, tp: int) -> "LlamaConfig":
        """Per-rank config under tensor parallelism."""
        assert self.num_heads % tp == 0 and self.num_kv_heads % tp == 0 \
            and self.intermediate_size % tp == 0
        return LlamaConfig(
            name=f"{self.name}-tp{tp}", hidden_size=self.hidden_size,
            intermediate_size=self.intermediate_size // tp,
            num_layers=self.num_layers, num_heads=self.num_heads // tp,
            num_kv_heads=self.num_kv_heads // tp, head_dim=self.head_dim,
            vocab_size=self.vocab_size, rope_theta=self.rope_theta,
            max_position=self.max_position, rms_eps=self.rms_eps,
            tie_embeddings=self.tie_embeddings,
            lm_vocab_rows=(self.vocab_size // tp
                           if self.vocab_size % tp == 0 else None),
            num_experts=self.num_experts,
            num_experts_per_tok=self.num_experts_per_tok,
            sliding_window=self.sliding_window)


CONFIGS = {
    # Reference parity target configs (BASELINE.json): Llama-3-8B / 70B.
    "llama-3-8b": LlamaConfig(name="llama-3-8b", hidden_size=4096,
                              intermediate_size=14336, num_layers=32,
                              num_heads=32, num_kv_heads=8, vocab_size=128256),
    "llama-3-70b": LlamaConfig(name="llama-3-70b", hidden_size=8192,
                               intermediate_size=28672, num_layers=80,
                               num_heads=64, num_kv_heads=8, vocab_size=128256),
    # same architecture, different shapes: loadable from HF checkpoints
    "llama-2-7b": LlamaConfig(name="llama-2-7b", hidden_size=4096,
                              intermediate_size=11008, num_layers=32,
                              num_heads=32, num_kv_heads=32,
                              vocab_size=32000, rope_theta=10000.0,
                              max_position=4096),
    "llama-2-13b": LlamaConfig(name="llama-2-13b", hidden_size=5120,
                               intermediate_size=13824, num_layers=40,
                               num_heads=40, num_kv_heads=40,
                               vocab_size=32000, rope_theta=10000.0,
                               max_position=4096),
    # Mistral v0.1: 4096-token sliding-window attention (both kernels take
    # the band; KV pages behind the window stay allocated — rolling-buffer
    # page reuse is a scheduler change queued in docs/ROADMAP.md)
    "mistral-7b": LlamaConfig(name="mistral-7b", hidden_size=4096,
                              intermediate_size=14336, num_layers=32,
                              num_heads=32, num_kv_heads=8,
                              vocab_size=32000, rope_theta=10000.0,
                              max_position=8192, sliding_window=4096),
    # small configs for tests / smoke
    "tiny": LlamaConfig(name="tiny", hidden_size=256, intermediate_size=512,
                        num_layers=2, num_heads=2, num_kv_heads=1,
                        vocab_size=512, max_position=512),
    "debug-1b": LlamaConfig(name="debug-1b", hidden_size=2048,
                            intermediate_size=8192, num_layers=16,
                            num_heads=16, num_kv_heads=8, vocab_size=32000,
                            max_position=8192),
    # sparse MoE family (Mixtral-style; intermediate_size is per expert)
    "mixtral-8x7b": LlamaConfig(name="mixtral-8x7b", hidden_size=4096,
                                intermediate_size=14336, num_layers=32,
                                num_heads=32, num_kv_heads=8,
                                vocab_size=32000, rope_theta=1e6,
                                max_position=8192, num_experts=8,
                                num_experts_per_tok=2),
    "tiny-moe": LlamaConfig(name="tiny-moe", hidden_size=256,
                            intermediate_size=512, num_layers=2,
                            num_heads=2, num_kv_heads=1, vocab_size=512,
                            max_position=512, num_experts=4,
                            num_experts_per_tok=2),
}


@dataclass
class AttnMetadata:
    """Describes the batch for the attention kernels.

    Prefill: cu_seqlens [B+1] i32 (chunk rows), seq_lens list[int] (chunk
             lens), q_start [B] i32 (absolute chunk start; 0 unless chunked),
             block_table [B,maxp] i32, slots [T] i64.
    Decode:  block_table [B,maxp] i32, seq_lens_t [B] i32, slots [B] i64,
             nsplit chosen by the engine.
    """
    is_prefill: bool
    slots: torch.Tensor
    cu_seqlens: torch.Tensor | None = None
    seq_lens: list[int] | None = None
    q_start: torch.Tensor | None = None
    block_table: torch.Tensor | None = None
    seq_lens_t: torch.Tensor | None = None
    nsplit: int = 1
    decode_scratch: tuple | None = None


class KVCache:
    """Paged KV cache for all layers: [L][2][npages, Hk, page, D] bf16."""

    def __init__(self, cfg: LlamaConfig, num_pages: int, page_size: int,
                 device, dtype=torch.bfloat16):
        self.page_size = page_size
        self.num_pages = num_pages
        shape = (num_pages, cfg.num_kv_heads, page_size, cfg.head_dim)
        self.k = [torch.zeros(shape, dtype=dtype, device=device)
                  for _ in range(cfg.num_layers)]
        self.v = [torch.zeros(shape, dtype=dtype, device=device)
                  for _ in range(cfg.num_layers)]

    @staticmethod
    def bytes_per_page(cfg: LlamaConfig, page_size: int) -> int:
        return 2 * cfg.num_layers * cfg.num_kv_heads * page_size * cfg.head_dim * 2


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, layer_idx: int):
        super().__init__()
        self.cfg = cfg
        self.layer_idx = layer_idx
        H, D = cfg.hidden_size, cfg.head_dim
        self.qkv = nn.Parameter(torch.empty(cfg.q_size + 2 * cfg.kv_size, H))
        self.o = nn.Parameter(torch.empty(H, cfg.q_size))
        self.scale = 1.0 / math.sqrt(D)

    def attend(self, qkv, positions, rope_tab, kv: KVCache, md: AttnMetadata,
               scale: tuple | None = None):
        """RoPE + cache append + attention over the fused QKV buffer."""
        cfg = self.cfg
        # strided views straight into the fused projection (no copies)
        q = qkv[:, :cfg.q_size]
        k = qkv[:, cfg.q_size:cfg.q_size + cfg.kv_size]
        v = qkv[:, cfg.q_size + cfg.kv_size:]
        ops.rope_cache(q, k, v, positions, rope_tab,
                       kv.k[self.layer_idx], kv.v[self.layer_idx], md.slots,
                       scale=scale)
        if md.is_prefill:
            return ops.attn_prefill(q, kv.k[self.layer_idx],
                                    kv.v[self.layer_idx], md.block_table,
                                    md.q_start, md.cu_seqlens, md.seq_lens,
                                    self.scale, head_dim=cfg.head_dim,
                                    window=cfg.sliding_window)
        return ops.attn_decode(q, kv.k[self.layer_idx], kv.v[self.layer_idx],
                               md.block_table, md.seq_lens_t, self.scale,
                               nsplit=md.nsplit, scratch=md.decode_scratch,
                               window=cfg.sliding_window)

    def forward(self, x, positions, rope_tab, kv: KVCache, md: AttnMetadata):
        qkv = ops.linear(x, self.qkv)
        o = self.attend(qkv, positions, rope_tab, kv, md)
        return ops.linear(o, self.o)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_up = nn.Parameter(torch.empty(2 * cfg.intermediate_size,
                                                cfg.hidden_size))
        self.down = nn.Parameter(torch.empty(cfg.hidden_size,
                                             cfg.intermediate_size))

    def forward(self, x):
        return ops.linear(ops.linear(x, self.gate_up, silu_fuse=True),
                          self.down)


class MoEMLP(nn.Module):
    """Mixtral-style sparse FFN (reference family scope: the upstream
    framework serves arbitrary provider models through LiteLLM; in-process
    we add MoE as a second model family next to dense Llama).

    Softmax router over num_experts, top-k per token with renormalized
    gates.  Two dispatch modes:

    * prefill (exact, eager): each expert runs the fused gate_up/down on
      its routed token subset — FLOP-proportional, shapes data-dependent.
    * decode (static, hipGraph-capturable): every expert runs on every
      token with routing weights zeroed for unrouted pairs.  Decode
      GEMMs are weight-streaming-bound and a top-2-of-8 batch >= ~16
      touches essentially every expert anyway, so streaming all expert
      weights costs what exact dispatch costs — while every shape stays
      static: one grouped gate_up GEMM [T, E*2I], one fused SwiGLU over
      [T*E, 2I], one batched down GEMM, one weighted sum."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        E, H, inter = cfg.num_experts, cfg.hidden_size, cfg.intermediate_size
        self.top_k = cfg.num_experts_per_tok
        self.router = nn.Parameter(torch.empty(E, H))
        self.gate_up = nn.Parameter(torch.empty(E, 2 * inter, H))
        self.down = nn.Parameter(torch.empty(E, H, inter))

    def forward(self, x, static: bool = False):
        # fp32 routing (bf16 softmax near-ties destabilize expert choice)
        E = self.router.shape[0]
        probs = torch.softmax(x.float() @ self.router.float().t(), dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        topv = topv / topv.sum(dim=-1, keepdim=True)
        if static:
            T, H = x.shape
            I2 = self.gate_up.shape[1]
            w = torch.zeros(T, E, dtype=torch.float32, device=x.device)
            w.scatter_(1, topi, topv)
            gu = ops.linear(x, self.gate_up.reshape(E * I2, H))
            act = ops.silu_and_mul(gu.reshape(T * E, I2))
            act = act.reshape(T, E, I2 // 2).transpose(0, 1)  # [E, T, I]
            ye = torch.bmm(act, self.down.transpose(1, 2))    # [E, T, H]
            out = (ye.float() * w.t().unsqueeze(-1)).sum(0)
            return out.to(x.dtype)
        out = torch.zeros(x.shape, dtype=torch.float32, device=x.device)
        for e in range(E):
            sel, slot = (topi == e).nonzero(as_tuple=True)
            if sel.numel() == 0:
                continue
            xe = x[sel]
            ye = ops.linear(ops.linear(xe, self.gate_up[e], silu_fuse=True),
                            self.down[e])
            out.index_add_(0, sel, ye.float() * topv[sel, slot].unsqueeze(1))
        return out.to(x.dtype)


class LlamaLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, layer_idx: int):
        super().__init__()
        self.input_norm = nn.Parameter(torch.empty(cfg.hidden_size))
        self.post_norm = nn.Parameter(torch.empty(cfg.hidden_size))
        self.attn = LlamaAttention(cfg, layer_idx)
        self.mlp = MoEMLP(cfg) if cfg.num_experts > 1 else LlamaMLP(cfg)
        self.eps = cfg.rms_eps

    def forward(self, h, residual, positions, rope_tab, kv, md):
        if residual is None:
            residual = h
            h = ops.rmsnorm(h, self.input_norm, self.eps)
        else:
            h, residual = ops.rmsnorm(h, self.input_norm, self.eps, residual)
        h = self.attn(h, positions, rope_tab, kv, md)
        h, residual = ops.rmsnorm(h, self.post_norm, self.eps, residual)
        if isinstance(self.mlp, MoEMLP):
            # decode always uses the static-capacity dispatch: measured on
            # MI355X it beats the exact per-expert loop at every decode
            # batch up to 256 (the loop's nonzero() host syncs dominate),
            # and it is the only hipGraph-capturable form
            h = self.mlp(h, static=not md.is_prefill)
        else:
            h = self.mlp(h)
        return h, residual

    def forward_decode_fused(self, residual, ss, ss2, positions, rope_tab,
                             kv, md):
        """Decode fast path: RMSNorms fold into the skinny GEMMs — input
        normalization happens in the GEMM's X staging (from per-row
        sum-of-squares stats), residual-add + next-norm stats in the GEMM
        combine.  Zero standalone norm kernels per layer."""
        at, mlp = self.attn, self.mlp
        M = residual.shape[0]
        K = residual.shape[1]
        small = M <= 64  # measured crossover vs hipBLASLt (BENCHMARKS.md)
        if small:
            qkv = ops.linear_skinny(residual, at.qkv, scale=(ss, self.eps))
            o = at.attend(qkv, positions, rope_tab, kv, md)
        else:
            # blas GEMM on the un-normalized rows; the RMSNorm scalar rides
            # along in rope_cache (RoPE is linear, so this is exact)
            qkv = residual @ at.qkv.t()
            o = at.attend(qkv, positions, rope_tab, kv, md,
                          scale=(ss, K, self.eps))
        ops.linear_skinny(o, at.o, mode=4, residual=residual, ss_out=ss2)
        if small:
            act = ops.linear_skinny(residual, mlp.gate_up, mode=2,
                                    scale=(ss2, self.eps))
        else:
            act = ops.silu_and_mul(residual @ mlp.gate_up.t(),
                                   scale=(ss2, K, self.eps))
        ops.linear_skinny(act, mlp.down, mode=4, residual=residual, ss_out=ss)
        return residual, ss


class LlamaForCausalLM(nn.Module):
    """The model proper.  TP-sharded variants are built by parallel.tp."""

    def __init__(self, cfg: LlamaConfig, device="cpu", dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        factory = dict(device=device, dtype=dtype)
        with torch.device(device):
            self.embed = nn.Parameter(torch.empty(cfg.vocab_size, cfg.hidden_size))
            self.layers = nn.ModuleList(
                [LlamaLayer(cfg, i) for i in range(cfg.num_layers)])
            self.final_norm = nn.Parameter(torch.empty(cfg.hidden_size))
            self.lm_head = nn.Parameter(torch.empty(
                cfg.lm_vocab_rows or cfg.vocab_size, cfg.hidden_size))
        self.to(dtype=dtype)
        self.register_buffer(
            "rope_tab",
            ops.rope_table(cfg.max_position, cfg.head_dim, cfg.rope_theta,
                           device=device), persistent=False)

    @torch.no_grad()
    def init_random(self, seed: int = 0):
        dev = self.embed.device
        g = torch.Generator(device=dev).manual_seed(seed)
        for name, p in self.named_parameters():
            if "norm" in name:
                p.fill_(1.0)
            else:
                p.normal_(0.0, 0.02, generator=g)
        return self

    @torch.no_grad()
    def fold_norm_weights(self):
        """Fold each RMSNorm's elementwise weight into the projection that
        consumes it (qkv <- input_norm, gate_up <- post_norm, lm_head <-
        final_norm), leaving unit norm weights.  Mathematically identical;
        lets the fused decode path apply the norm as a per-row output
        scalar in the GEMM combine, keeping the async weight stream."""
        if getattr(self, "_norms_folded", False):
            return self
        for layer in self.layers:
            layer.attn.qkv.copy_(
                (layer.attn.qkv.float() * layer.input_norm.float()
                 ).to(layer.attn.qkv.dtype))
            layer.input_norm.fill_(1.0)
            layer.mlp.gate_up.copy_(
                (layer.mlp.gate_up.float() * layer.post_norm.float()
                 ).to(layer.mlp.gate_up.dtype))  # broadcasts over MoE [E,2I,H]
            if isinstance(layer.mlp, MoEMLP):
                # the router consumes the same post-norm output: fold there too
                layer.mlp.router.copy_(
                    (layer.mlp.router.float() * layer.post_norm.float()
                     ).to(layer.mlp.router.dtype))
            layer.post_norm.fill_(1.0)
        self.lm_head.copy_((self.lm_head.float() * self.final_norm.float()
                            ).to(self.lm_head.dtype))
        self.final_norm.fill_(1.0)
        self._norms_folded = True
        return self

    def _can_fuse_decode(self, T: int) -> bool:
        import os
        if os.environ.get("AF_NO_FUSED_DECODE") == "1":
            return False
        if not getattr(self, "_norms_folded", False):
            return False
        cfg = self.cfg
        return (not getattr(self, "no_fused_decode", False) and T <= 128
                and cfg.num_experts == 1  # MoE routes through the eager path
                and cfg.hidden_size % 64 == 0
                and cfg.intermediate_size % 64 == 0
                and (cfg.q_size + 2 * cfg.kv_size) % 64 == 0
                and cfg.q_size % 64 == 0)

    def forward(self, ids, positions, kv: KVCache, md: AttnMetadata,
                logit_rows: torch.Tensor | None = None):
        """ids/positions [T] -> logits [T or len(logit_rows), vocab]."""
        T = ids.numel()
        if not md.is_prefill and ids.is_cuda and self._can_fuse_decode(T):
            ss = torch.empty(T, 8, dtype=torch.float32, device=ids.device)
            ss2 = torch.empty_like(ss)
            residual = ops.embedding(ids, self.embed, ss=ss)
            for layer in self.layers:
                residual, ss = layer.forward_decode_fused(
                    residual, ss, ss2, positions, self.rope_tab, kv, md)
            h = ops.rmsnorm(residual, self.final_norm, self.cfg.rms_eps)
            return self._project_logits(h)
        h = ops.embedding(ids, self.embed)
        residual = None
        for layer in self.layers:
            h, residual = layer(h, residual, positions, self.rope_tab, kv, md)
        h, _ = ops.rmsnorm(h, self.final_norm, self.cfg.rms_eps, residual)
        if logit_rows is not None:
            h = ops.gather_rows(h, logit_rows)
        return self._project_logits(h)

    def _project_logits(self, h: torch.Tensor) -> torch.Tensor:
        """lm_head projection.  Under vocab-parallel TP each rank computes
        its [T, V/tp] shard and the shards are all-gathered along the vocab
        dim (rank r owns global vocab rows [r*V/tp, (r+1)*V/tp)), so every
        rank samples from identical full logits."""
        logits = ops.linear(h, self.lm_head)
        if getattr(self, "_tp_vocab_parallel", False):
            import torch.distributed as dist
            group = getattr(self, "tp_logits_group", None)
            world = dist.get_world_size(group)
            shards = [torch.empty_like(logits) for _ in range(world)]
            dist.all_gather(shards, logits.contiguous(), group=group)
            logits = torch.cat(shards, dim=-1)
        return logits
