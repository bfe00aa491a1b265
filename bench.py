#!/usr/bin/env python3
"""Flagship serving benchmark: reasoner calls/sec on Llama-3-8B (bf16,
synthetic prompts, random-init weights).

One rank per GPU (DP replicas over RCCL when launched via torchrun).  A
"step" = each rank completes a fixed batch of `--calls` reasoner calls
end-to-end through the serving engine (continuous-batch prefill of
`--prompt-len` synthetic tokens + `--gen-len` decoded tokens each, sampling
included).  value = whole-job completed calls/sec across all N GPUs.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models import CONFIGS


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=4)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--calls", type=int, default=128,
                   help="reasoner calls per rank per step")
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--gen-len", type=int, default=64)
    p.add_argument("--max-num-seqs", type=int, default=128)
    p.add_argument("--no-graphs", action="store_true")
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree (requires WORLD_SIZE == tp; "
                        "ranks form one TP group instead of DP replicas)")
    p.add_argument("--device", default=None)
    return p.parse_args()


def run_step(eng: LLMEngine, rank: int, step: int, args) -> list[float]:
    """Submit `calls` requests and drive the engine until all finish.
    Returns per-call latencies (s)."""
    g = torch.Generator().manual_seed(1000 * rank + step)
    sp = SamplingParams(max_tokens=args.gen_len, ignore_eos=True)
    t_submit = {}
    lat = []
    for c in range(args.calls):
        prompt = torch.randint(0, eng.cfg.vocab_size, (args.prompt_len,),
                               generator=g).tolist()
        rid = eng.add_request(prompt, sp)
        assert rid is not None
        t_submit[rid] = time.perf_counter()
    pending = set(t_submit)
    while pending:
        eng.step()
        for rid in list(pending):
            if eng.get_finished(rid) is not None:
                lat.append(time.perf_counter() - t_submit[rid])
                pending.discard(rid)
    return lat


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo")
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    cfg = CONFIGS[args.model]
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    if args.tp > 1:
        if world != args.tp:
            raise SystemExit("--tp requires WORLD_SIZE == tp")
        from agentfield_amd.parallel import TPEngineGroup
        grp = TPEngineGroup(cfg, device, dtype=dtype,
                            max_num_seqs=args.max_num_seqs,
                            max_prefill_tokens=args.prompt_len * args.calls,
                            enable_graphs=not args.no_graphs and device == "cuda")
        eng = grp.engine
    else:
        eng = LLMEngine(cfg, device=device, dtype=dtype,
                        max_num_seqs=args.max_num_seqs,
                        max_prefill_tokens=args.prompt_len * args.calls,
                        enable_graphs=not args.no_graphs and device == "cuda",
                        seed=0)

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()

    for w in range(args.warmup):
        run_step(eng, rank, -1 - w, args)

    barrier_sync()
    t0 = time.perf_counter()
    lats = []
    for s in range(args.steps):
        lats.extend(run_step(eng, rank, s, args))
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device == "cuda" else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    n_replicas = 1 if args.tp > 1 else world
    total_calls = args.calls * args.steps * n_replicas
    value = total_calls / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    if rank == 0:
        out = {
            "metric": "reasoner_calls_per_sec",
            "value": round(value, 3),
            "unit": "calls/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if device == "cuda" else "fp32",
            "data": "synthetic",
            "config": {
                "model": cfg.name,
                "global_batch": args.calls * n_replicas,
                "seq_len": args.prompt_len + args.gen_len,
                "prompt_len": args.prompt_len,
                "gen_len": args.gen_len,
                "parallelism": f"tp{world}" if args.tp > 1 else f"dp{world}",
                "p50_call_ms": round(statistics.median(lats) * 1000, 1) if lats else None,
                "tokens_per_sec": round(total_calls * (args.prompt_len + args.gen_len) / elapsed, 1),
            },
        }
        print(json.dumps(out))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
