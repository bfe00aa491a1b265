#!/usr/bin/env python3
"""BASELINE config 4 (single-node slice): async execution queue + HMAC
webhooks in front of the GPU engine at high concurrency.

Stands up a multi-worker control plane, one engine-backed agent
(app.ai() -> in-process LLMEngine), and an aiohttp webhook receiver that
VERIFIES every X-AgentField-Signature.  Submits --requests async
executions at --concurrency and reports end-to-end completions/s (submit
-> engine generate -> completion webhook delivered+verified).

  python tools/config4_bench.py --model llama-3-8b --requests 512 \
      --concurrency 512 --gen 64 --prompt-chars 487
"""
from __future__ import annotations

import argparse
import asyncio
import hashlib
import hmac
import json
import os
import statistics
import subprocess
import sys
import tempfile
import threading
import time
from pathlib import Path


def log(msg: str) -> None:
    print(f"[cfg4 +{time.monotonic() - T0:.1f}s] {msg}", file=sys.stderr,
          flush=True)


T0 = time.monotonic()

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

SECRET = "bench-hmac-secret"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--requests", type=int, default=512)
    ap.add_argument("--concurrency", type=int, default=512)
    ap.add_argument("--gen", type=int, default=64)
    ap.add_argument("--prompt-chars", type=int, default=487)
    ap.add_argument("--cp-workers", type=int, default=4)
    ap.add_argument("--port", type=int, default=18120)
    ap.add_argument("--warmup", type=int, default=64)
    args = ap.parse_args()

    import torch

    log("importing engine")
    from agentfield_amd.engine import LLMEngine
    from agentfield_amd.models import CONFIGS
    from agentfield_amd.sdk import Agent
    from agentfield_amd.sdk.ai import (AIConfig, ByteTokenizer, EngineRunner,
                                       set_runner)
    from bench import _AgentServer, _wait_http

    sys.setswitchinterval(0.02)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = CONFIGS[args.model]
    kw = {}
    if device == "cpu":
        kw = {"num_pages": 512, "max_num_seqs": 8, "dtype": torch.float32}
    log(f"building engine ({args.model}, {device})")
    eng = LLMEngine(cfg, device=device,
                    max_num_seqs=kw.pop("max_num_seqs", 128),
                    max_prefill_tokens=args.prompt_chars * 160,
                    enable_graphs=device == "cuda",
                    seed=0, **kw)
    runner = EngineRunner(eng, ByteTokenizer(cfg.vocab_size))
    set_runner(args.model, runner)
    log("engine ready; starting control plane")

    cp_port = args.port
    agent_port = args.port + 500
    hook_port = args.port + 600
    tmp = tempfile.mkdtemp(prefix="af-cfg4-")
    cp = subprocess.Popen(
        [sys.executable, "-m", "agentfield_amd", "server",
         "--host", "127.0.0.1", "--port", str(cp_port),
         "--db", f"{tmp}/af.db", "--data-dir", tmp,
         "--workers", str(args.cp_workers)],
        env={**os.environ, "PYTHONPATH": str(ROOT)}, cwd=ROOT,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        for i in range(args.cp_workers):
            _wait_http(f"http://127.0.0.1:{cp_port + i}/api/v1/health")
        cp_url = f"http://127.0.0.1:{cp_port}"

        agent = Agent("g0", agentfield_url=cp_url, auto_register=False,
                      base_url=f"http://127.0.0.1:{agent_port}",
                      ai_config=AIConfig(model=args.model))

        @agent.reasoner()
        def reason(prompt: str, gen: int = 64):
            return {"text": agent.ai(prompt, max_tokens=gen,
                                     ignore_eos=True, temperature=0.0)}

        log("control plane healthy; starting agent")
        _AgentServer(agent, agent_port).start()
        assert agent.register()
        log("agent registered; driving load")

        out = asyncio.run(drive(args, cp_url, hook_port))
        print(json.dumps(out))
    finally:
        cp.terminate()
        try:
            cp.wait(timeout=10)
        except subprocess.TimeoutExpired:
            cp.kill()
        runner.shutdown()


async def drive(args, cp_url: str, hook_port: int) -> dict:
    import aiohttp
    from aiohttp import web

    done: dict[str, float] = {}
    bad_sig = 0
    all_done = asyncio.Event()
    expected = {"n": 0}

    async def recv(request: web.Request):
        nonlocal bad_sig
        raw = await request.read()
        sig = request.headers.get("X-AgentField-Signature", "")
        want = "sha256=" + hmac.new(SECRET.encode(), raw,
                                    hashlib.sha256).hexdigest()
        if sig != want:
            bad_sig += 1
        ev = json.loads(raw)
        done[ev["execution_id"]] = time.perf_counter()
        if expected["n"] and len(done) >= expected["n"]:
            all_done.set()
        return web.json_response({"ok": True})

    app = web.Application()
    app.router.add_post("/hook", recv)
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", hook_port)
    await site.start()
    hook_url = f"http://127.0.0.1:{hook_port}/hook"

    prompt = "q" * args.prompt_chars
    sem = asyncio.Semaphore(args.concurrency)
    submit_t: dict[str, float] = {}

    async def submit(session, i: int) -> None:
        async with sem:
            t0 = time.perf_counter()
            async with session.post(
                    f"{cp_url}/api/v1/execute/async/g0.reason",
                    json={"input": {"prompt": prompt, "gen": args.gen},
                          "webhook": {"url": hook_url,
                                      "secret": SECRET}}) as r:
                assert r.status == 202, await r.text()
                body = await r.json()
                submit_t[body["execution_id"]] = t0

    stop_watch = threading.Event()

    def watchdog():
        while not stop_watch.wait(20.0):
            log(f"watchdog: submitted={len(submit_t)} done={len(done)} "
                f"expected={expected['n']}")

    threading.Thread(target=watchdog, daemon=True).start()
    async with aiohttp.ClientSession(
            connector=aiohttp.TCPConnector(limit=0),
            timeout=aiohttp.ClientTimeout(total=900)) as session:
        # warmup (untimed): fills graphs/caches
        expected["n"] = args.warmup
        log(f"warmup: submitting {args.warmup}")
        await asyncio.gather(*(submit(session, i)
                               for i in range(args.warmup)))
        log("warmup submitted; waiting for webhooks")
        await asyncio.wait_for(all_done.wait(), timeout=300)
        log("warmup complete")
        done.clear()
        submit_t.clear()
        all_done.clear()

        expected["n"] = args.requests
        t0 = time.perf_counter()
        log(f"timed: submitting {args.requests}")
        await asyncio.gather(*(submit(session, i)
                               for i in range(args.requests)))
        log("timed submitted; waiting for webhooks")
        await asyncio.wait_for(all_done.wait(), timeout=420)
        elapsed = time.perf_counter() - t0
        stop_watch.set()

    lats = sorted(done[e] - submit_t[e] for e in done if e in submit_t)
    await runner.cleanup()
    return {
        "bench": "config4_async_webhooks",
        "completions_per_sec": round(args.requests / elapsed, 2),
        "requests": args.requests,
        "concurrency": args.concurrency,
        "elapsed_s": round(elapsed, 2),
        "p50_s": round(statistics.median(lats), 3) if lats else None,
        "p95_s": round(lats[int(0.95 * len(lats))], 3) if lats else None,
        "webhooks_verified": len(done) - bad_sig,
        "bad_signatures": bad_sig,
    }


if __name__ == "__main__":
    main()
