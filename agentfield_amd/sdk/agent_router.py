"""Modular agent composition (reference parity: SDK router/agent_registry —
agent.py:2042-2196 `include_router` with prefix rewriting).

An AgentRouter collects @reasoner/@skill declarations without owning a
FastAPI app; `Agent.include_router(router, prefix=...)` registers them on
the agent with the prefix dotted into the function name ("billing.report"),
which stays addressable through the control plane because execute targets
split on the FIRST dot only (node, then reasoner path).
"""
from __future__ import annotations


class AgentRouter:
    def __init__(self, prefix: str = ""):
        self.prefix = prefix.strip("./")
        self._pending: list[dict] = []

    def reasoner(self, name: str | None = None, tags=None, vc: bool = False):
        def deco(fn):
            self._pending.append({"kind": "reasoner",
                                  "name": name or fn.__name__,
                                  "tags": tags, "vc": vc, "fn": fn})
            return fn
        return deco

    def skill(self, name: str | None = None, tags=None,
              cache_results: bool = False):
        def deco(fn):
            self._pending.append({"kind": "skill",
                                  "name": name or fn.__name__,
                                  "tags": tags, "vc": False,
                                  "cache_results": cache_results, "fn": fn})
            return fn
        return deco

    def include_router(self, other: "AgentRouter", prefix: str = ""):
        """Routers nest; prefixes accumulate left-to-right."""
        pre = (prefix or other.prefix).strip("./")
        for item in other._pending:
            merged = dict(item)
            if pre:
                merged["name"] = f"{pre}.{item['name']}"
            self._pending.append(merged)
