"""Rejection-sampled speculative decoding (Leviathan et al.): extends
the greedy-exact verify path to sampling temperatures while keeping the
TARGET distribution exact.

For draft token d_j with draft distribution q_j and target distribution
p_j (both already temperature-scaled softmaxes):

  accept d_j with probability min(1, p_j[d_j] / q_j[d_j]);
  on the first rejection, emit a token from normalize(max(p_j - q_j, 0))
  and stop; if every draft token is accepted, emit a bonus token from
  p_k (the verify forward's last row).

Prompt-lookup drafts are deterministic proposals — a point mass at d_j —
so acceptance is with probability p_j[d_j] and the residual collapses to
p_j with d_j zeroed.

The marginal of each emitted token equals sampling p_j directly
(test_spec_sampling pins this statistically), so sampled outputs follow
the same distribution as non-speculative decoding; only the RNG stream
differs.
"""
from __future__ import annotations

import torch


def accept_resample(p_rows: torch.Tensor, draft: list[int],
                    q_rows: torch.Tensor | None,
                    gen: torch.Generator) -> list[int]:
    """p_rows [k+1, V] target probs (row j = after consuming draft[:j]);
    q_rows [k, V] draft probs or None (point-mass / prompt-lookup).
    Returns the emitted tokens: accepted prefix + (residual sample |
    bonus sample)."""
    k = len(draft)
    out: list[int] = []
    for j in range(k):
        p = p_rows[j]
        d = draft[j]
        q_d = float(q_rows[j, d]) if q_rows is not None else 1.0
        pd = float(p[d])
        r = float(torch.rand((), generator=gen, device=p.device))
        if q_d > 0.0 and r <= pd / q_d:
            out.append(d)
            continue
        # rejected: residual distribution max(p - q, 0)
        if q_rows is not None:
            adj = (p - q_rows[j]).clamp_(min=0.0)
        else:
            adj = p.clone()
            adj[d] = 0.0
        s = float(adj.sum())
        if s <= 0.0:  # p == q numerically: rejection was impossible; p wins
            adj = p
            s = 1.0
        tok = int(torch.multinomial(adj / s, 1, generator=gen))
        out.append(tok)
        return out
    out.append(int(torch.multinomial(p_rows[k], 1, generator=gen)))
    return out
