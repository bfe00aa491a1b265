"""Expert parallelism (EP) for the sparse-MoE family: experts are
sharded across ranks and tokens travel to their experts over an
all-to-all exchange (RCCL over xGMI on the GPU; gloo lacks all_to_all,
so the same exchange runs over batched isend/irecv there — identical
semantics, which is what the CPU tests pin).

Topology: rank r owns experts [r*E/ep, (r+1)*E/ep).  The router is
replicated (tiny), so every rank computes identical top-k choices for
ITS tokens; each forward is:

  1. route local tokens, sort the (token, expert-slot) pairs by owner
     rank, count per-destination
  2. all-to-all the counts, then the activations (one [sum, H] tensor
     split by destination — xGMI is point-to-point, so one large
     all_to_all_single beats per-peer sends)
  3. run the LOCAL experts' fused gate_up/SwiGLU/down on the received
     rows (grouped per expert — exact dispatch, FLOP-proportional)
  4. all-to-all the results back and combine with the renormalized
     gate weights

Capacity note: the exchange is exact (variable splits), not
static-capacity — hipGraph capture of EP decode would need the static
variant (engine MoE decode already has one for the single-rank path).
"""
from __future__ import annotations

import torch
import torch.distributed as dist
from torch import nn

from .. import ops


def _exchange(t: torch.Tensor, out_splits: list[int], in_splits: list[int],
              group) -> torch.Tensor:
    """all_to_all_single with a gloo fallback (batched isend/irecv)."""
    out = torch.empty(sum(out_splits), *t.shape[1:], dtype=t.dtype,
                      device=t.device)
    backend = dist.get_backend(group)
    if backend != "gloo":
        dist.all_to_all_single(out, t.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits, group=group)
        return out
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    in_off = [0]
    for s in in_splits:
        in_off.append(in_off[-1] + s)
    out_off = [0]
    for s in out_splits:
        out_off.append(out_off[-1] + s)
    reqs = []
    tc = t.contiguous()
    for peer in range(world):
        if peer == rank:
            out[out_off[peer]:out_off[peer + 1]] = \
                tc[in_off[peer]:in_off[peer + 1]]
            continue
        if in_splits[peer]:
            reqs.append(dist.isend(
                tc[in_off[peer]:in_off[peer + 1]].contiguous(), peer,
                group=group))
        if out_splits[peer]:
            buf = torch.empty(out_splits[peer], *t.shape[1:], dtype=t.dtype,
                              device=t.device)
            reqs.append((dist.irecv(buf, peer, group=group), peer, buf))
    for r in reqs:
        if isinstance(r, tuple):
            req, peer, buf = r
            req.wait()
            out[out_off[peer]:out_off[peer + 1]] = buf
        else:
            r.wait()
    return out


def build_ep_model(cfg, device="cpu", dtype=torch.float32,
                   base_seed: int = 0, group=None):
    """Full MoE model with every sparse FFN replaced by its EP shard.
    Every rank builds the SAME full model (same seed) and slices its
    experts — attention/embeddings stay replicated (EP-only layout; the
    reference placement for MoE where expert weights dominate memory).
    Returns a LlamaForCausalLM whose MoEMLP modules are EPMoE."""
    from ..models.llama import LlamaForCausalLM, MoEMLP
    model = LlamaForCausalLM(cfg, device=device,
                             dtype=dtype).init_random(base_seed)
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return model  # single rank: the plain MoE IS the EP=1 layout
    for layer in model.layers:
        if isinstance(layer.mlp, MoEMLP):
            layer.mlp = EPMoE.shard_from(layer.mlp, group=group).to(
                device=device, dtype=dtype)
    model.no_fused_decode = True  # collectives attach to module forward
    return model


class EPEngineGroup:
    """Drives one LLMEngine per rank with an expert-parallel MoE model:
    attention/embeddings replicated, experts sharded, tokens exchanged
    inside each MoE forward.  Request sync, stepping and the no-pickle
    wire codec are TPEngineGroup's (identical scheduling + seeds on all
    ranks -> identical outputs); only the model construction differs —
    the full config is kept (no head/FFN sharding) and the per-layer
    collective is the MoE all-to-all instead of TP all-reduces."""

    def __init__(self, cfg, device, dtype=torch.bfloat16, group=None,
                 base_seed: int = 0, **engine_kw):
        from ..engine import LLMEngine
        from .tp import TPEngineGroup
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.tp = dist.get_world_size(group) if dist.is_initialized() else 1
        if torch.device(device).type != "cuda":
            engine_kw.setdefault("enable_graphs", False)
        model = build_ep_model(cfg, device=device, dtype=dtype,
                               base_seed=base_seed, group=group)
        self.engine = LLMEngine(cfg, device=device, dtype=dtype,
                                model=model, tp_group=group, **engine_kw)

    def _bcast_device(self):
        from .tp import TPEngineGroup
        return TPEngineGroup._bcast_device(self)

    def broadcast_and_submit(self, requests=None):
        from .tp import TPEngineGroup
        return TPEngineGroup.broadcast_and_submit(self, requests)

    def submit(self, prompt_ids, sampling):
        return self.engine.add_request(prompt_ids, sampling)

    def step(self):
        return self.engine.step()

    def has_work(self):
        return self.engine.has_work()

    def get_finished(self, rid):
        return self.engine.get_finished(rid)


class EPMoE(nn.Module):
    """Expert-parallel Mixtral-style sparse FFN.  Construct from a full
    MoEMLP's weights via `shard_from` (each rank keeps its expert slice
    plus the replicated router)."""

    def __init__(self, num_experts: int, top_k: int, hidden: int,
                 inter: int, group=None):
        super().__init__()
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        assert num_experts % self.world == 0, \
            "num_experts must divide the EP group"
        self.E = num_experts
        self.e_local = num_experts // self.world
        self.top_k = top_k
        self.router = nn.Parameter(torch.empty(num_experts, hidden))
        self.gate_up = nn.Parameter(torch.empty(self.e_local, 2 * inter,
                                                hidden))
        self.down = nn.Parameter(torch.empty(self.e_local, hidden, inter))

    @classmethod
    def shard_from(cls, moe, group=None) -> "EPMoE":
        """Slice a full MoEMLP (models.llama) onto this rank."""
        E, H = moe.router.shape
        inter = moe.down.shape[2]
        ep = cls(E, moe.top_k, H, inter, group=group)
        lo = ep.rank * ep.e_local
        with torch.no_grad():
            ep.router.copy_(moe.router)
            ep.gate_up.copy_(moe.gate_up[lo:lo + ep.e_local])
            ep.down.copy_(moe.down[lo:lo + ep.e_local])
        return ep

    def forward(self, x: torch.Tensor, static: bool = False) -> torch.Tensor:
        if static:
            return self._forward_static(x)
        T, H = x.shape
        dev = x.device
        # 1) replicated fp32 routing (identical to MoEMLP.forward)
        probs = torch.softmax(x.float() @ self.router.float().t(), dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        topv = topv / topv.sum(dim=-1, keepdim=True)

        # flatten (token, slot) pairs and sort by owning rank (stable so
        # the inverse permutation reassembles results deterministically)
        flat_e = topi.reshape(-1)                      # [T*k]
        owner = flat_e // self.e_local
        order = torch.argsort(owner, stable=True)
        inv = torch.empty_like(order)
        inv[order] = torch.arange(order.numel(), device=dev)
        send_tok = (torch.arange(T, device=dev)
                    .repeat_interleave(self.top_k))[order]
        send_e = flat_e[order]
        send_x = x[send_tok]

        in_splits = torch.bincount(owner, minlength=self.world).tolist()
        # 2) exchange counts, then rows (+ expert ids alongside)
        counts = torch.tensor(in_splits, dtype=torch.int64)
        all_counts = [torch.zeros_like(counts) for _ in range(self.world)]
        dist.all_gather(all_counts, counts, group=self.group)
        out_splits = [int(c[self.rank]) for c in all_counts]
        rx = _exchange(send_x, out_splits, in_splits, self.group)
        re = _exchange(send_e.unsqueeze(1).to(torch.int64), out_splits,
                       in_splits, self.group).squeeze(1)

        # 3) local experts on received rows (exact grouped dispatch)
        ry = torch.zeros(rx.shape[0], H, dtype=torch.float32, device=dev)
        lo = self.rank * self.e_local
        for le in range(self.e_local):
            sel = (re == lo + le).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            xe = rx[sel]
            ye = ops.linear(ops.linear(xe, self.gate_up[le], silu_fuse=True),
                            self.down[le])
            ry[sel] = ye.float()

        # 4) results return to the sender; combine with gate weights
        back = _exchange(ry.to(x.dtype), in_splits, out_splits, self.group)
        back = back[inv].reshape(T, self.top_k, H)
        out = (back.float() * topv.unsqueeze(-1)).sum(dim=1)
        return out.to(x.dtype)

    def _forward_static(self, x: torch.Tensor) -> torch.Tensor:
        """Static-capacity exchange for decode: every rank sends a FIXED
        [world, T*top_k, H] block (slots padded with expert-id -1), so
        every tensor shape is data-INDEPENDENT — the property hipGraph
        capture needs.  Padding-only (capacity = T*top_k per
        destination), so unlike trained capacity-factor routing nothing
        is ever dropped and the result is EXACTLY the dynamic
        dispatch's.  Costs world x the exchange volume of the exact
        path; decode batches are small, so the static shape is the
        better trade (same reasoning as the single-rank static MoE)."""
        T, H = x.shape
        dev = x.device
        cap = T * self.top_k
        probs = torch.softmax(x.float() @ self.router.float().t(), dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        topv = topv / topv.sum(dim=-1, keepdim=True)

        flat_e = topi.reshape(-1)                       # [T*k]
        owner = flat_e // self.e_local
        # slot s of destination block d holds pair index s if owned by d
        send_x = torch.zeros(self.world * cap, H, dtype=x.dtype, device=dev)
        send_e = torch.full((self.world * cap,), -1, dtype=torch.int64,
                            device=dev)
        tok_of_pair = (torch.arange(T, device=dev)
                       .repeat_interleave(self.top_k))
        idx = owner * cap + torch.arange(cap, device=dev)
        send_x[idx] = x[tok_of_pair]
        send_e[idx] = flat_e

        splits = [cap] * self.world
        rx = _exchange(send_x, splits, splits, self.group)
        re = _exchange(send_e.unsqueeze(1), splits, splits,
                       self.group).squeeze(1)

        ry = torch.zeros(self.world * cap, H, dtype=torch.float32,
                         device=dev)
        lo = self.rank * self.e_local
        for le in range(self.e_local):
            sel = (re == lo + le).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            ye = ops.linear(ops.linear(rx[sel], self.gate_up[le],
                                       silu_fuse=True), self.down[le])
            ry[sel] = ye.float()

        back = _exchange(ry.to(x.dtype), splits, splits, self.group)
        # this rank's block d came back at block d; slot s = pair s
        mine = back[self.rank * cap:(self.rank + 1) * cap]             if False else None
        # reassemble: pair s was sent in block owner[s], slot s
        pair_rows = back.reshape(self.world, cap, H)[
            owner, torch.arange(cap, device=dev)]
        pair_rows = pair_rows.reshape(T, self.top_k, H)
        out = (pair_rows.float() * topv.unsqueeze(-1)).sum(dim=1)
        return out.to(x.dtype)
