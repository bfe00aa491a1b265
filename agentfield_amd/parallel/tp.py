"""Tensor parallelism over RCCL/xGMI (one process per GPU).

Sharding (MI355X-first — sized so the only collective per layer is one
all-reduce after attn-O and one after MLP-down, the pattern that maps onto
xGMI's 7 p2p links):
  * qkv:    column-parallel — each rank keeps Hq/tp query heads and Hk/tp
            KV heads (whole heads, so RoPE/attention stay local)
  * o:      row-parallel    — partial sums all-reduced
  * gate_up: column-parallel (gate and up halves sharded separately so the
            fused silu_mul kernel sees a local [gate|up] layout)
  * down:   row-parallel    — partial sums all-reduced
  * embed:  replicated (the lookup is local and cheap)
  * lm_head: vocab-parallel when vocab % tp == 0 — rank r holds global
            vocab rows [r*V/tp, (r+1)*V/tp), computes its logits shard and
            all-gathers along the vocab dim (saves the ~1-2 GB replicated
            head AND tp x the lm_head GEMM FLOPs; sampling still sees full
            logits on every rank, so scheduling stays identical-decision)

Weights are deterministic per parameter NAME, so every rank can materialize
the full tensor (fp32, one at a time), slice its shard and free — no rank
ever holds the whole model.
"""
from __future__ import annotations

import hashlib

import torch
import torch.distributed as dist

from ..models.llama import LlamaConfig, LlamaForCausalLM


def _seed_for(name: str, base_seed: int) -> int:
    h = hashlib.sha256(f"{base_seed}:{name}".encode()).digest()
    return int.from_bytes(h[:8], "little") % (2**63)


def _full_param(name: str, shape, base_seed: int, device) -> torch.Tensor:
    if "norm" in name:
        return torch.ones(shape, dtype=torch.float32, device=device)
    g = torch.Generator(device=device).manual_seed(_seed_for(name, base_seed))
    t = torch.empty(shape, dtype=torch.float32, device=device)
    return t.normal_(0.0, 0.02, generator=g)


def shard_llama_weights(model: LlamaForCausalLM, full_cfg: LlamaConfig,
                        tp: int, rank: int, base_seed: int = 0) -> None:
    """Fill a shard-shaped model with its slice of deterministic full
    weights.  Replaces init_random for TP runs (rank-agnostic layout)."""
    cfg = full_cfg
    D = cfg.head_dim
    hq, hk = cfg.num_heads // tp, cfg.num_kv_heads // tp
    inter = cfg.intermediate_size // tp
    dev = model.embed.device
    with torch.no_grad():
        for name, p in model.named_parameters():
            if name == "embed" or name == "lm_head":
                full_shape = (cfg.vocab_size, cfg.hidden_size)
                vs_rows = p.shape[0]  # lm_head may be vocab-sharded
            elif name.endswith("attn.qkv"):
                full_shape = (cfg.q_size + 2 * cfg.kv_size, cfg.hidden_size)
            elif name.endswith("attn.o"):
                full_shape = (cfg.hidden_size, cfg.q_size)
            elif name.endswith("mlp.gate_up"):
                full_shape = (2 * cfg.intermediate_size, cfg.hidden_size)
                if p.dim() == 3:  # MoE: per-expert FFN, same col sharding
                    full_shape = (cfg.num_experts,) + full_shape
            elif name.endswith("mlp.down"):
                full_shape = (cfg.hidden_size, cfg.intermediate_size)
                if p.dim() == 3:
                    full_shape = (cfg.num_experts,) + full_shape
            else:  # norms, MoE router (replicated)
                full_shape = tuple(p.shape)
            full = _full_param(name, full_shape, base_seed, dev)
            if name.endswith("attn.qkv"):
                q, k, v = full.split([cfg.q_size, cfg.kv_size, cfg.kv_size], 0)
                shard = torch.cat([
                    q[rank * hq * D:(rank + 1) * hq * D],
                    k[rank * hk * D:(rank + 1) * hk * D],
                    v[rank * hk * D:(rank + 1) * hk * D]], dim=0)
            elif name.endswith("attn.o"):
                shard = full[:, rank * hq * D:(rank + 1) * hq * D]
            elif name.endswith("mlp.gate_up"):
                cdim = full.dim() - 2  # expert dim (if any) leads
                gate, up = full.chunk(2, dim=cdim)
                sl = slice(rank * inter, (rank + 1) * inter)
                shard = torch.cat([gate.narrow(cdim, sl.start, inter),
                                   up.narrow(cdim, sl.start, inter)], dim=cdim)
            elif name.endswith("mlp.down"):
                shard = full.narrow(full.dim() - 1, rank * inter, inter)
            elif name == "lm_head" and p.shape[0] != cfg.vocab_size:
                shard = full[rank * vs_rows:(rank + 1) * vs_rows]
            else:
                shard = full
            p.copy_(shard.to(p.dtype))
            del full


class _TPLayerHook:
    """Inserts the all-reduce after attn-O and MLP-down projections.

    Small bf16 decode tensors go through the one-shot xGMI kernel when
    its init self-test passed (parallel/oneshot.py); everything else —
    prefill-sized tensors, capture-time calls — uses RCCL."""

    def __init__(self, group, oneshot=None):
        self.group = group
        self.oneshot = oneshot

    def __call__(self, module, inputs, output):
        if self.oneshot is not None and self.oneshot.eligible(output):
            return self.oneshot.allreduce(output)
        dist.all_reduce(output, group=self.group)
        return output


def build_tp_model(full_cfg: LlamaConfig, tp: int, rank: int, device,
                   dtype=torch.bfloat16, group=None,
                   base_seed: int = 0,
                   oneshot=None) -> LlamaForCausalLM:
    """Construct this rank's shard of the model with collectives attached."""
    shard_cfg = full_cfg.shard(tp)
    model = LlamaForCausalLM(shard_cfg, device=device, dtype=dtype)
    shard_llama_weights(model, full_cfg, tp, rank, base_seed)
    if tp > 1:
        hook = _TPLayerHook(group, oneshot)
        for layer in model.layers:
            layer.attn.register_forward_hook(hook)
            layer.mlp.register_forward_hook(hook)
        model.no_fused_decode = True  # collectives attach to module forward
        if shard_cfg.lm_vocab_rows is not None:
            model._tp_vocab_parallel = True
            model.tp_logits_group = group
    return model


class TPEngineGroup:
    """Drives one LLMEngine across a TP process group.

    All ranks run the identical engine code; rank 0 is the request source
    and broadcasts each submitted request (prompt ids + sampling) before
    stepping, so every rank makes identical scheduling/sampling decisions
    and the only steady-state communication is the per-layer all-reduce.
    """

    def __init__(self, full_cfg: LlamaConfig, device, dtype=torch.bfloat16,
                 group=None, base_seed: int = 0, **engine_kw):
        import os
        from ..engine import LLMEngine
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.tp = dist.get_world_size(group) if dist.is_initialized() else 1
        if self.tp > 1:
            # RCCL-inside-hipGraph: AF_TP_GRAPHS=0/1 forces; otherwise a
            # runtime self-test captures+replays a real all-reduce in a
            # hipGraph and enables graphs only when every rank verifies
            # the replayed result (graph-capture support differs across
            # RCCL builds — probe, don't assume)
            env = os.environ.get("AF_TP_GRAPHS")
            if env is not None:
                engine_kw.setdefault("enable_graphs", env == "1")
            elif torch.device(device).type == "cuda":
                engine_kw.setdefault("enable_graphs",
                                     self._rccl_graph_selftest(device))
            else:
                engine_kw.setdefault("enable_graphs", False)
        oneshot = None
        if self.tp > 1 and torch.device(device).type == "cuda" and \
                os.environ.get("AF_ONESHOT_AR", "1") != "0":
            from .oneshot import OneShotAllReduce
            cand = OneShotAllReduce(group, device)
            oneshot = cand if cand.ok else None
        self.oneshot = oneshot
        model = build_tp_model(full_cfg, self.tp, self.rank, device, dtype,
                               group, base_seed, oneshot=oneshot)
        shard_cfg = model.cfg
        if self.tp > 1 and "num_pages" not in engine_kw and \
                torch.device(device).type == "cuda":
            # identical KV page budget on every rank: min of local estimates
            from ..models.llama import KVCache
            free, _ = torch.cuda.mem_get_info(device)
            page_size = engine_kw.get("page_size", 16)
            local = max(16, int(free * 0.80) //
                        KVCache.bytes_per_page(shard_cfg, page_size))
            t = torch.tensor([local], dtype=torch.int64, device=device)
            dist.all_reduce(t, op=dist.ReduceOp.MIN, group=group)
            engine_kw["num_pages"] = int(t.item())
        self.engine = LLMEngine(shard_cfg, device=device, dtype=dtype,
                                model=model, tp_group=group, **engine_kw)

    def _rccl_graph_selftest(self, device) -> bool:
        """Capture one RCCL all-reduce in a hipGraph and verify a replay
        on every rank; all ranks must agree (MIN-reduce of the verdict)
        before graphs are enabled."""
        ok = 0
        try:
            x = torch.ones(64, device=device)
            dist.all_reduce(x, group=self.group)  # warm RCCL comms
            torch.cuda.synchronize()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                dist.all_reduce(x, group=self.group)
            torch.cuda.current_stream().wait_stream(s)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                dist.all_reduce(x, group=self.group)
            x.fill_(1.0)
            g.replay()
            torch.cuda.synchronize()
            ok = int(abs(x[0].item() - self.tp) < 1e-3)
        except Exception:
            ok = 0
        verdict = torch.tensor([ok], dtype=torch.int32, device=device)
        dist.all_reduce(verdict, op=dist.ReduceOp.MIN, group=self.group)
        return bool(int(verdict.item()))

    # ---- rank-0 request API -------------------------------------------
    def submit(self, prompt_ids, sampling) -> int | None:
        """Called with identical args on every rank (bench/test mode), or
        on rank 0 only followed by sync_requests()."""
        return self.engine.add_request(prompt_ids, sampling)

    def _bcast_device(self):
        """Device RCCL payloads must live on (gloo broadcasts CPU)."""
        if dist.get_backend(self.group) == "nccl":
            return torch.device("cuda", torch.cuda.current_device())
        return torch.device("cpu")

    def broadcast_and_submit(self, requests: list | None):
        """Rank 0 passes its new requests; other ranks pass None.

        No pickle on the hot path: requests ride a flat
        int32 tensor (parallel/wire.py) — one length broadcast, and a
        payload broadcast only when there are new requests."""
        if self.tp == 1:
            out = []
            for (ids, sp) in requests or []:
                out.append(self.engine.add_request(ids, sp))
            return out
        from .wire import decode_requests, encode_requests
        dev = self._bcast_device()
        if self.rank == 0:
            payload = encode_requests(requests or []).to(dev)
            size = torch.tensor([payload.numel()], dtype=torch.int64,
                                device=dev)
        else:
            size = torch.zeros(1, dtype=torch.int64, device=dev)
        dist.broadcast(size, src=0, group=self.group)
        n = int(size.item())
        if n <= 1:  # encode([]) == [0]: nothing new this step
            return []
        if self.rank != 0:
            payload = torch.empty(n, dtype=torch.int32, device=dev)
        dist.broadcast(payload, src=0, group=self.group)
        rids = []
        for (ids, sp) in decode_requests(payload.cpu()):
            rids.append(self.engine.add_request(ids, sp))
        return rids

    def step(self):
        return self.engine.step()

    def has_work(self):
        return self.engine.has_work()

    def get_finished(self, rid):
        return self.engine.get_finished(rid)
