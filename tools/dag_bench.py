#!/usr/bin/env python3
"""BASELINE config 3: 3-agent workflow DAG with DID/VC audit, 1 GPU.

  planner.plan --app.call--> worker.work --app.call--> summarizer.sum

All three agents share one in-process engine (app.ai()); every reasoner
call repeats a common system preamble, so automatic prefix caching turns
the repeated prefill into a block-table lookup.  The control plane runs
as a subprocess with DID enabled and auto-VC issuance on; after the load
the script verifies one run's VC chain covers the whole DAG and reports
prefix-cache hit metrics from the engine.

  python tools/dag_bench.py --model tiny --calls 8        # CPU smoke
  python tools/dag_bench.py --model llama-3-8b --calls 48 # 1x MI355X
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import subprocess
import sys
import tempfile
import threading
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

import httpx
import torch

PREAMBLE = ("You are part of a three-stage agent workflow. Follow the "
            "instructions precisely, be terse, and never repeat the task "
            "text back. Stage discipline matters more than style. ")


def wait_http(url, timeout=60.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if httpx.get(url, timeout=2.0).status_code == 200:
                return
        except Exception:
            pass
        time.sleep(0.05)
    raise TimeoutError(url)


class AgentServer:
    def __init__(self, app, port):
        import uvicorn
        self.server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=port, log_level="error",
            access_log=False, lifespan="off"))
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    def start(self):
        self.thread.start()
        while not self.server.started:
            time.sleep(0.02)
        return self


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--calls", type=int, default=48)
    ap.add_argument("--gen", type=int, default=32)
    ap.add_argument("--cp-port", type=int, default=18210)
    args = ap.parse_args()

    from agentfield_amd.engine import LLMEngine
    from agentfield_amd.models import CONFIGS
    from agentfield_amd.sdk import Agent
    from agentfield_amd.sdk.ai import (AgentAI, AIConfig, ByteTokenizer,
                                       EngineRunner, set_runner)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = CONFIGS[args.model]
    kw = {"prefix_cache": True}
    if device == "cpu":
        kw.update(num_pages=512, max_num_seqs=8, dtype=torch.float32,
                  enable_graphs=False)
    eng = LLMEngine(cfg, device=device, **kw)
    runner = EngineRunner(eng, ByteTokenizer(cfg.vocab_size))
    set_runner(args.model, runner)

    tmp = tempfile.mkdtemp(prefix="af-dag-")
    cp_url = f"http://127.0.0.1:{args.cp_port}"
    cp = subprocess.Popen(
        [sys.executable, "-m", "agentfield_amd", "server",
         "--host", "127.0.0.1", "--port", str(args.cp_port),
         "--db", f"{tmp}/af.db", "--data-dir", tmp],
        env={**os.environ, "PYTHONPATH": str(ROOT),
             "AGENTFIELD_AUTO_VC": "1"},
        cwd=ROOT, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    wait_http(f"{cp_url}/api/v1/health")

    aicfg = AIConfig(model=args.model)
    agents = []
    for i, name in enumerate(("planner", "worker", "summarizer")):
        a = Agent(name, agentfield_url=cp_url, auto_register=False,
                  base_url=f"http://127.0.0.1:{args.cp_port + 10 + i}",
                  ai_config=aicfg, vc_enabled=True)
        agents.append(a)
    planner, worker, summarizer = agents

    @planner.reasoner()
    def plan(task: str):
        text = planner.ai(PREAMBLE, user=f"Stage 1 of 3. Plan: {task}",
                          max_tokens=args.gen, ignore_eos=True)
        sub = planner.call("worker.work", task=task)
        return {"plan": text[:60], "sub": sub}

    @worker.reasoner()
    def work(task: str):
        text = worker.ai(PREAMBLE, user=f"Stage 2 of 3. Execute: {task}",
                         max_tokens=args.gen, ignore_eos=True)
        sub = worker.call("summarizer.sum", task=task)
        return {"work": text[:60], "sub": sub}

    @summarizer.reasoner(name="sum")
    def summarize(task: str):
        text = summarizer.ai(PREAMBLE, user=f"Stage 3 of 3. Summarize: {task}",
                             max_tokens=args.gen, ignore_eos=True)
        return {"summary": text[:60]}

    for i, a in enumerate(agents):
        AgentServer(a, args.cp_port + 10 + i).start()
        assert a.register(), f"{a.node_id} registration failed"

    async def load():
        import aiohttp
        lat, run_ids = [], []
        async with aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=600)) as s:
            async def one(i):
                t0 = time.perf_counter()
                async with s.post(
                        f"{cp_url}/api/v1/execute/planner.plan",
                        json={"input": {"task": f"ship feature #{i}"}}) as r:
                    body = await r.json(content_type=None)
                assert body.get("status") == "completed", body
                lat.append((time.perf_counter() - t0) * 1e3)
                run_ids.append(body["run_id"])
            await asyncio.gather(*(one(i) for i in range(args.calls)))
        return lat, run_ids

    # warmup (fills the prefix cache) then timed run
    asyncio.run(load())
    t0 = time.perf_counter()
    lat, run_ids = asyncio.run(load())
    elapsed = time.perf_counter() - t0

    # DAG + VC-chain verification on one run
    rid = run_ids[0]
    dag = httpx.get(f"{cp_url}/api/ui/v1/workflows/{rid}/dag",
                    timeout=10).json()
    chain = httpx.get(f"{cp_url}/api/v1/did/workflow/{rid}/vc-chain",
                      timeout=10).json()
    sched = eng.sched
    out = {
        "metric": "dag_workflows_per_sec",
        "value": round(args.calls / elapsed, 3),
        "unit": "workflows/s",
        "reasoner_calls_per_sec": round(3 * args.calls / elapsed, 3),
        "p50_root_ms": round(statistics.median(lat), 1),
        "model": args.model,
        "device": device,
        "dag_nodes": len(dag.get("nodes", [])),
        "vc_chain_len": chain.get("count", 0),
        "vc_chain_valid": chain.get("all_valid", False),
        "prefix_cache": {
            "cached_tokens": getattr(sched, "cached_tokens", 0),
            "cache_hits": getattr(sched, "cache_hits", 0),
            "cache_pages": getattr(sched, "cache_pages", 0),
            "scheduler": type(sched).__name__,
        },
        "engine": {k: eng.metrics[k] for k in
                   ("prefill_tokens", "decode_tokens")},
    }
    print(json.dumps(out))
    assert out["dag_nodes"] >= 3, "DAG must span the 3-agent chain"
    assert out["vc_chain_len"] >= 3, "VC chain must cover the DAG"
    assert out["prefix_cache"]["cache_hits"] > 0, "prefix cache never hit"
    cp.terminate()


if __name__ == "__main__":
    main()
