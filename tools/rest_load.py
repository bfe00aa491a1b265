#!/usr/bin/env python3
"""REST load generator for bench.py --rest.

Runs as a subprocess of the bench's rank 0 so the client's CPU work does
not share a GIL with the engine driver or the agent HTTP server.  Line
protocol on stdin/stdout:

  in : {"calls": N, "targets": [...], "urls": [...], "prompt_chars": C,
        "gen": G, "concurrency": K}
  out: {"ok": n, "errors": {...}, "latencies_ms": [...]}

One input line = one benchmark step (submit N calls round-robin over
targets and control-plane worker urls, wait for all).  EOF exits.
The aiohttp session persists across steps, so keep-alive connections
carry the whole run.
"""
from __future__ import annotations

import asyncio
import json
import sys
import time


async def run_step(session, spec) -> dict:
    calls = spec["calls"]
    targets = spec["targets"]
    urls = spec["urls"]
    payload_base = {"prompt": "x" * spec["prompt_chars"],
                    "gen": spec["gen"]}
    sem = asyncio.Semaphore(spec.get("concurrency") or calls)
    lat: list[float] = []
    errors: dict[str, int] = {}

    async def one(i: int):
        url = urls[i % len(urls)]
        target = targets[i % len(targets)]
        async with sem:
            t0 = time.perf_counter()
            try:
                async with session.post(
                        f"{url}/api/v1/execute/{target}",
                        json={"input": payload_base}) as r:
                    body = await r.json(content_type=None)
                    status = body.get("status") if r.status == 200 \
                        else f"http_{r.status}"
            except Exception as e:
                status = f"error:{type(e).__name__}"
            if status == "completed":
                lat.append((time.perf_counter() - t0) * 1e3)
            else:
                errors[status] = errors.get(status, 0) + 1

    await asyncio.gather(*(one(i) for i in range(calls)))
    return {"ok": len(lat), "errors": errors, "latencies_ms": lat}


async def main():
    import aiohttp
    session = aiohttp.ClientSession(
        timeout=aiohttp.ClientTimeout(total=600.0),
        connector=aiohttp.TCPConnector(limit=0))
    loop = asyncio.get_running_loop()
    reader = asyncio.StreamReader()
    await loop.connect_read_pipe(
        lambda: asyncio.StreamReaderProtocol(reader), sys.stdin)
    try:
        while True:
            line = await reader.readline()
            if not line:
                break
            spec = json.loads(line)
            out = await run_step(session, spec)
            sys.stdout.write(json.dumps(out) + "\n")
            sys.stdout.flush()
    finally:
        await session.close()


if __name__ == "__main__":
    asyncio.run(main())
