#!/usr/bin/env python3
"""Control-plane load harness (reference parity: tools/perf/
nested_workflow_stress.py — async httpx driver with sync/async modes,
concurrency sweep, nested depth, latency percentiles, status histograms,
Prometheus deltas and backpressure verification).

  python tools/stress.py --url http://127.0.0.1:8520 --target echo.greet \
      --requests 200 --concurrency 16 [--mode async] [--payload-bytes 1024]

With --self-contained it spins up an in-process control plane + echo agent
(optionally --nested for a relay reasoner that fans out depth-2 calls).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import httpx


def pct(values, p):
    if not values:
        return None
    vs = sorted(values)
    return vs[min(len(vs) - 1, int(p * len(vs)))]


async def run_load(url: str, target: str, requests: int, concurrency: int,
                   mode: str, payload_bytes: int, depth: int) -> dict:
    import aiohttp
    sem = asyncio.Semaphore(concurrency)
    latencies: list[float] = []
    statuses: dict[str, int] = {}
    http_codes: dict[int, int] = {}
    backpressure = 0
    payload = {"name": "x" * max(1, payload_bytes)}
    if depth > 1:
        payload["depth"] = depth
    targets = target.split(",")  # round-robin across agent replicas
    # round-robin across control-plane workers when the plane advertises a
    # fleet (multi-worker mode)
    urls = [url]
    try:
        h = httpx.get(f"{url}/api/v1/health", timeout=2.0).json()
        if h.get("workers"):
            urls = h["workers"]
    except Exception:
        pass

    async with aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=120.0),
            connector=aiohttp.TCPConnector(limit=concurrency + 32)) as client:
        async def one(i: int):
            nonlocal backpressure
            async with sem:
                t0 = time.perf_counter()
                try:
                    if mode == "sync":
                        async with client.post(
                                f"{urls[i % len(urls)]}/api/v1/execute/{targets[i % len(targets)]}",
                                json={"input": payload}) as r:
                            http_codes[r.status] = http_codes.get(r.status, 0) + 1
                            body = (await r.json(content_type=None)
                                    if r.status == 200 else {})
                        st = body.get("status", f"http_{r.status}")
                    else:
                        async with client.post(
                                f"{urls[i % len(urls)]}/api/v1/execute/async/{targets[i % len(targets)]}",
                                json={"input": payload}) as r:
                            http_codes[r.status] = http_codes.get(r.status, 0) + 1
                            code = r.status
                            body = await r.json(content_type=None) \
                                if code in (200, 202) else {}
                        if code == 503:
                            backpressure += 1
                            st = "backpressure"
                        elif code == 202:
                            eid = body["execution_id"]
                            st = "queued"
                            for _ in range(600):
                                async with client.get(
                                        f"{urls[i % len(urls)]}/api/v1/executions/{eid}") as g:
                                    st = (await g.json(content_type=None)
                                          ).get("status", "unknown")
                                if st in ("completed", "failed", "timeout",
                                          "cancelled"):
                                    break
                                await asyncio.sleep(0.05)
                        else:
                            st = f"http_{code}"
                    statuses[st] = statuses.get(st, 0) + 1
                    latencies.append(time.perf_counter() - t0)
                except Exception as e:
                    statuses[f"error:{type(e).__name__}"] = \
                        statuses.get(f"error:{type(e).__name__}", 0) + 1

        t0 = time.perf_counter()
        await asyncio.gather(*(one(i) for i in range(requests)))
        wall = time.perf_counter() - t0

    return {
        "requests": requests,
        "concurrency": concurrency,
        "mode": mode,
        "wall_s": round(wall, 3),
        "req_per_s": round(requests / wall, 2),
        "p50_ms": round(pct(latencies, 0.50) * 1e3, 1) if latencies else None,
        "p95_ms": round(pct(latencies, 0.95) * 1e3, 1) if latencies else None,
        "p99_ms": round(pct(latencies, 0.99) * 1e3, 1) if latencies else None,
        "statuses": statuses,
        "http_codes": http_codes,
        "backpressure_503": backpressure,
    }


async def scrape_metrics(url: str) -> dict:
    wanted = ("agentfield_gateway_queue_depth",
              "agentfield_worker_inflight",
              "agentfield_waiters_inflight",
              "agentfield_gateway_backpressure_total")
    out = {}
    try:
        async with httpx.AsyncClient(timeout=5.0) as c:
            text = (await c.get(f"{url}/metrics")).text
        for line in text.splitlines():
            for w in wanted:
                if line.startswith(w + " "):
                    out[w] = float(line.split()[-1])
    except Exception:
        pass
    return out


def _wait_http(url: str, timeout: float = 20.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if httpx.get(url, timeout=1.0).status_code == 200:
                return
        except httpx.HTTPError:
            pass
        time.sleep(0.05)
    raise TimeoutError(f"{url} not up")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:8520")
    ap.add_argument("--target", default="echo.greet")
    ap.add_argument("--requests", type=int, default=200)
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--mode", choices=("sync", "async"), default="sync")
    ap.add_argument("--payload-bytes", type=int, default=1024)
    ap.add_argument("--depth", type=int, default=1,
                    help="nested call depth (needs the relay reasoner)")
    ap.add_argument("--self-contained", action="store_true",
                    help="spin up an in-process control plane + echo agent")
    ap.add_argument("--procs", action="store_true",
                    help="spin up the control plane and echo agent as "
                         "SEPARATE processes (no shared GIL — the real "
                         "deployment topology)")
    ap.add_argument("--cp-port", type=int, default=8520)
    ap.add_argument("--agent-port", type=int, default=8601)
    ap.add_argument("--agents", type=int, default=1,
                    help="number of echo agent processes (--procs mode); "
                         "load round-robins across them")
    ap.add_argument("--cp-workers", type=int, default=1,
                    help="control-plane worker processes (--procs mode)")
    args = ap.parse_args()

    servers = []
    procs = []
    if args.procs:
        import os
        import subprocess
        root = Path(__file__).resolve().parent.parent
        env = {**os.environ, "PYTHONPATH": str(root)}
        cp_url = f"http://127.0.0.1:{args.cp_port}"
        import tempfile
        tmp = tempfile.mkdtemp(prefix="af-stress-")
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "agentfield_amd", "server",
             "--host", "127.0.0.1", "--port", str(args.cp_port),
             "--db", f"{tmp}/af.db", "--data-dir", tmp,
             "--workers", str(args.cp_workers)],
            env=env, cwd=root, stdout=subprocess.DEVNULL,
            stderr=subprocess.DEVNULL))
        for i in range(args.cp_workers):
            _wait_http(f"http://127.0.0.1:{args.cp_port + i}/api/v1/health")
        names = (["echo"] if args.agents == 1
                 else [f"echo{i}" for i in range(args.agents)])
        for i, name in enumerate(names):
            procs.append(subprocess.Popen(
                [sys.executable, str(root / "tools" / "echo_agent.py"),
                 "--port", str(args.agent_port + i), "--cp", cp_url,
                 "--node-id", name],
                env=env, cwd=root, stdout=subprocess.DEVNULL,
                stderr=subprocess.DEVNULL))
        for i in range(args.agents):
            _wait_http(f"http://127.0.0.1:{args.agent_port + i}/health")
        # wait for registration
        for _ in range(200):
            r = httpx.get(f"{cp_url}/api/v1/nodes", timeout=2.0)
            ids = {n.get("id") for n in r.json().get("nodes", [])}
            if all(n in ids for n in names):
                break
            time.sleep(0.05)
        args.url = cp_url
        if args.agents > 1:
            args.target = ",".join(f"{n}.greet" for n in names)
    if args.self_contained:
        from agentfield_amd.controlplane import ControlPlane, create_app
        from agentfield_amd.controlplane.server import Config
        from agentfield_amd.sdk import Agent
        sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))
        from helpers import AppServer

        cp = ControlPlane(Config(background_services=False))
        cp_srv = AppServer(create_app(cp)).start().wait_healthy()
        agent = Agent("echo", agentfield_url=cp_srv.base_url,
                      auto_register=False)

        @agent.reasoner()
        def greet(name: str, depth: int = 1):
            if depth > 1:
                return {"nested": agent.call("echo.greet", name=name,
                                             depth=depth - 1)}
            return {"echo": len(name)}

        a_srv = AppServer(agent).start()
        agent.base_url = a_srv.base_url
        assert agent.register()
        args.url = cp_srv.base_url
        servers = [a_srv, cp_srv]

    before = asyncio.run(scrape_metrics(args.url))
    result = asyncio.run(run_load(args.url, args.target, args.requests,
                                  args.concurrency, args.mode,
                                  args.payload_bytes, args.depth))
    after = asyncio.run(scrape_metrics(args.url))
    result["metrics_before"] = before
    result["metrics_after"] = after
    print(json.dumps(result, indent=2))
    for s in servers:
        s.stop()
    for p in procs:
        p.terminate()
    for p in procs:
        try:
            p.wait(timeout=5)
        except Exception:
            p.kill()


if __name__ == "__main__":
    main()
