#!/usr/bin/env python3
"""Flagship serving benchmark: reasoner calls/sec + p50 REST latency on
Llama-3-8B (bf16, synthetic prompts, random-init weights).

Default mode (`--rest`, BASELINE config 2/4) measures the WHOLE stack:
REST client -> control plane (/api/v1/execute/:target) -> agent
@reasoner -> app.ai() -> in-process MI355X engine.  One rank per GPU
(DP replicas when launched via torchrun); rank 0 additionally hosts the
control-plane worker fleet (subprocess) and the load client (subprocess).
A "step" = `--calls` reasoner calls per rank completed end-to-end
(continuous-batch prefill of `--prompt-len` synthetic tokens + `--gen-len`
decoded tokens each, sampling included).  value = whole-job completed
calls/sec across all N GPUs; p50_call_ms is CLIENT-OBSERVED REST latency.

`--engine-only` drives LLMEngine directly (no HTTP), isolating GPU-side
throughput; `--tp N` shards one engine across N ranks instead.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import subprocess
import sys
import tempfile
import threading


def _die_with_parent():
    """preexec_fn for subprocesses (control plane, engine replica): take
    SIGTERM when the bench rank dies, so an aborted run can never leave
    a stale control plane squatting on the derived ports (an orphan
    there makes every subsequent run on the same ports fail with 404s)."""
    try:
        import ctypes
        ctypes.CDLL("libc.so.6").prctl(1, 15)  # PR_SET_PDEATHSIG, SIGTERM
    except Exception:
        pass
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
    p.add_argument("--rest", dest="rest", action="store_true", default=None,
                   help="measure through the REST path (default)")
    p.add_argument("--engine-only", dest="rest", action="store_false",
                   help="drive the engine directly, no HTTP")
    p.add_argument("--engine-proc", dest="engine_proc",
                   action="store_true", default=False,
                   help="run the engine as a per-rank serving-replica "
                        "subprocess (config-4 topology) instead of inside "
                        "the agent process.  Measured on MI355X: the extra "
                        "hop costs ~13%% on this stepped workload (71 vs 82 "
                        "calls/s) because each step ends with a 128-response "
                        "stampede through the replica's event loop, so "
                        "in-process remains the default")
    p.add_argument("--cp-workers", type=int, default=0,
                   help="control-plane worker processes (0 = auto)")
    return p.parse_args()


# --------------------------------------------------------------- REST mode
def _wait_http(url: str, timeout: float = 60.0) -> None:
    import httpx
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if httpx.get(url, timeout=2.0).status_code == 200:
                return
        except Exception:
            pass
        time.sleep(0.1)
    raise TimeoutError(f"{url} not up")


class _AgentServer:
    """uvicorn-in-a-thread hosting this rank's Agent (the engine stays in
    this process: app.ai() is served by the in-process runner)."""

    def __init__(self, app, port: int):
        import uvicorn
        self.config = uvicorn.Config(app, host="127.0.0.1", port=port,
                                     log_level="error", access_log=False,
                                     lifespan="off")
        self.server = uvicorn.Server(self.config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    def start(self):
        self.thread.start()
        deadline = time.time() + 30
        while time.time() < deadline and not self.server.started:
            time.sleep(0.02)
        if not self.server.started:
            raise TimeoutError("agent server did not start")
        return self


def run_rest(args, world: int, rank: int, dist, device: str, cfg) -> None:
    from agentfield_amd.sdk import Agent
    from agentfield_amd.sdk.ai import ByteTokenizer, EngineRunner, set_runner

    # the rank process hosts BOTH the engine driver thread and the agent's
    # HTTP threads; a larger switch interval cuts GIL churn on the
    # engine's per-step host work (measured: decode stays batch-128 but
    # steps run ~10% slower than engine-only under default 5 ms slices)
    sys.setswitchinterval(0.02)
    root = Path(__file__).resolve().parent

    # ports isolated per world size so the driver's back-to-back
    # N=1,2,4,8 SCALE runs can never collide with a straggler from the
    # previous run
    port_base = (17000 + int(os.environ.get("MASTER_PORT", "29500")) % 1000
                 + world * 23)
    cp_port = port_base
    # +500: clear of the CP worker ports (cp_port..+workers) and their
    # admin gRPC ports (cp_port+100..)
    agent_port = port_base + 500 + rank
    cp_url = f"http://127.0.0.1:{cp_port}"
    # control-plane worker fleet sized for the DP load: ~88 calls/s per
    # GPU, ~450+ calls/s per worker (measured) — one worker per 4 GPUs
    # would do; give headroom without oversubscribing small CI boxes
    auto = max(1, min(4, (os.cpu_count() or 8) // 8)) if world == 1 else \
        max(2, min(8, world, (os.cpu_count() or 8) // 4))
    cp_workers = args.cp_workers or auto

    eng_proc = None
    stats_url = None
    if args.engine_proc:
        # production-shaped topology: the engine is its own serving
        # replica process, so its driver thread never shares a GIL with
        # the agent's HTTP threads (in-proc mode measured ~8% slower)
        ep = port_base + 900 + rank
        stats_url = f"http://127.0.0.1:{ep}"
        eng_argv = [sys.executable, "-m", "agentfield_amd", "engine",
                    "--model", args.model, "--host", "127.0.0.1",
                    "--port", str(ep), "--device", device,
                    "--max-num-seqs", str(args.max_num_seqs),
                    "--max-prefill-tokens",
                    str(args.prompt_len * args.calls)]
        if args.no_graphs:
            eng_argv.append("--no-graphs")
        eng_proc = subprocess.Popen(
            eng_argv, env={**os.environ, "PYTHONPATH": str(root)}, cwd=root,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            preexec_fn=_die_with_parent)
        _wait_http(f"{stats_url}/v1/health", timeout=600.0)
        os.environ["AGENTFIELD_ENGINE_URLS"] = stats_url
        runner = None
    else:
        dtype = (torch.bfloat16 if device.startswith("cuda")
                 else torch.float32)
        kw = {}
        if not device.startswith("cuda"):
            kw = {"num_pages": 512, "max_num_seqs": 8,
                  "dtype": torch.float32}
        eng = LLMEngine(cfg, device=device,
                        max_num_seqs=kw.pop("max_num_seqs",
                                            args.max_num_seqs),
                        max_prefill_tokens=args.prompt_len * args.calls,
                        enable_graphs=not args.no_graphs and
                        device.startswith("cuda"),
                        dtype=kw.pop("dtype", dtype), seed=0, **kw)
        runner = EngineRunner(eng, ByteTokenizer(cfg.vocab_size))
        set_runner(args.model, runner)

    cp_proc = None
    if rank == 0:
        tmp = tempfile.mkdtemp(prefix="af-bench-")
        cp_proc = subprocess.Popen(
            [sys.executable, "-m", "agentfield_amd", "server",
             "--host", "127.0.0.1", "--port", str(cp_port),
             "--db", f"{tmp}/af.db", "--data-dir", tmp,
             "--workers", str(cp_workers)],
            env={**os.environ, "PYTHONPATH": str(root)}, cwd=root,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            preexec_fn=_die_with_parent)
        for i in range(cp_workers):
            _wait_http(f"http://127.0.0.1:{cp_port + i}/api/v1/health")
    if dist is not None:
        dist.barrier()

    from agentfield_amd.sdk.ai import AgentAI, AIConfig
    agent = Agent(f"gpu{rank}", agentfield_url=cp_url, auto_register=False,
                  base_url=f"http://127.0.0.1:{agent_port}",
                  ai_config=AIConfig(model=args.model))
    assert isinstance(agent.ai, AgentAI)

    @agent.reasoner()
    def reason(prompt: str, gen: int = 64):
        return {"text": agent.ai(prompt, max_tokens=gen, ignore_eos=True,
                                 temperature=0.0)}

    _AgentServer(agent, agent_port).start()
    if not agent.register():
        raise RuntimeError(f"rank {rank}: agent registration failed")
    if dist is not None:
        dist.barrier()

    if rank != 0:
        # serve: step barriers pace us with rank 0's load client
        for _ in range(args.warmup):
            dist.barrier()
        if device.startswith("cuda"):
            torch.cuda.synchronize()
        dist.barrier()  # timing start
        for _ in range(args.steps):
            dist.barrier()
        if device.startswith("cuda"):
            torch.cuda.synchronize()
        dist.barrier()  # timing end
        dist.barrier()  # teardown gate
        if eng_proc is not None:
            eng_proc.terminate()
            try:
                eng_proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                eng_proc.kill()
        return

    # ---- rank 0: drive the load via the client subprocess
    import httpx
    health = httpx.get(f"{cp_url}/api/v1/health", timeout=5.0).json()
    urls = health.get("workers") or [cp_url]
    targets = [f"gpu{r}.reason" for r in range(world)]
    client = subprocess.Popen(
        [sys.executable, str(root / "tools" / "rest_load.py")],
        stdin=subprocess.PIPE, stdout=subprocess.PIPE, text=True,
        env={**os.environ, "PYTHONPATH": str(root)}, cwd=root,
        preexec_fn=_die_with_parent)
    try:
        _drive_and_report(args, world, device, cfg, dist, client,
                          urls, targets, cp_workers, stats_url,
                          runner if stats_url is None else None)
    finally:
        client.terminate()
        for prc in (eng_proc, cp_proc):
            if prc is not None:
                prc.terminate()
                try:
                    prc.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    prc.kill()


def _drive_and_report(args, world, device, cfg, dist, client, urls,
                      targets, cp_workers, stats_url, runner):
    import httpx
    spec = {"calls": args.calls * world, "targets": targets, "urls": urls,
            "prompt_chars": max(8, args.prompt_len - 25),
            "gen": args.gen_len, "concurrency": args.calls * world}

    def one_step() -> dict:
        client.stdin.write(json.dumps(spec) + "\n")
        client.stdin.flush()
        out = json.loads(client.stdout.readline())
        if out["ok"] != spec["calls"]:
            raise RuntimeError(f"step had failures: {out['errors']}")
        return out

    for _ in range(args.warmup):
        one_step()
        if dist is not None:
            dist.barrier()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    if dist is not None:
        dist.barrier()
    t0 = time.perf_counter()
    lats: list[float] = []
    for _ in range(args.steps):
        out = one_step()
        lats.extend(out["latencies_ms"])
        if dist is not None:
            dist.barrier()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    if dist is not None:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    total_calls = args.calls * world * args.steps
    value = total_calls / elapsed
    if stats_url is not None:
        em = httpx.get(f"{stats_url}/v1/stats", timeout=10.0).json()
    else:
        em = runner.engine.metrics
    out = {
        "metric": "reasoner_calls_per_sec",
        "value": round(value, 3),
        "unit": "calls/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000.0, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if device.startswith("cuda") else "fp32",
        "data": "synthetic",
        "config": {
            "model": cfg.name,
            "global_batch": args.calls * world,
            "seq_len": args.prompt_len + args.gen_len,
            "prompt_len": args.prompt_len,
            "gen_len": args.gen_len,
            "parallelism": f"dp{world}",
            "path": "rest",
            "cp_workers": cp_workers,
            "p50_call_ms": round(statistics.median(lats), 1) if lats else None,
            "p95_call_ms": round(sorted(lats)[int(0.95 * len(lats))], 1)
            if lats else None,
            "tokens_per_sec": round(
                total_calls * (args.prompt_len + args.gen_len) / elapsed, 1),
            "decode_avg_batch": round(
                em["decode_tokens"] / max(1, em["decode_steps"]), 1),
            "spec_steps": em["spec_steps"],
            "engine_proc": stats_url is not None,
        },
    }
    print(json.dumps(out))
    client.stdin.close()
    if dist is not None:
        dist.barrier()  # teardown gate


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
    use_rest = (args.rest is None or args.rest) and args.tp <= 1
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        if use_rest:
            # DP-over-REST has no GPU collectives: gloo barriers pace the
            # steps without touching the engine's CUDA streams
            dist.init_process_group("gloo")
        else:
            dist.init_process_group(
                "nccl" if torch.cuda.is_available() else "gloo")
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    cfg = CONFIGS[args.model]
    if use_rest:
        if device == "cuda":
            device = f"cuda:{local_rank}"
        run_rest(args, world, rank, dist, device, cfg)
        if dist is not None:
            dist.destroy_process_group()
        return
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
