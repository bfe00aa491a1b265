#!/usr/bin/env python3
"""Standalone echo agent process for multi-process load testing.

  python tools/echo_agent.py --port 8601 --cp http://127.0.0.1:8520

Registers `echo.greet` (+ `echo.relay` for nested-depth runs) against the
control plane and serves until killed.  Used by tools/stress.py --procs and
bench.py --rest so the agent's Python work does not share a GIL with the
control plane or the load client.
"""
from __future__ import annotations

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8601)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--cp", default="http://127.0.0.1:8520")
    ap.add_argument("--node-id", default="echo")
    args = ap.parse_args()

    from agentfield_amd.sdk import Agent

    agent = Agent(args.node_id, agentfield_url=args.cp, auto_register=True,
                  base_url=f"http://{args.host}:{args.port}")

    @agent.reasoner()
    def greet(name: str, depth: int = 1):
        if depth > 1:
            return {"nested": agent.call(f"{args.node_id}.greet", name=name,
                                         depth=depth - 1)}
        return {"echo": len(name)}

    agent.serve(host=args.host, port=args.port)


if __name__ == "__main__":
    main()
