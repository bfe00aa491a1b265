"""Execution context propagation (reference parity: execution_context.py
-> X-Run-ID / X-Execution-ID / X-Parent-Execution-ID / X-Session-ID /
X-Actor-ID headers + contextvars manager)."""
from __future__ import annotations

import contextvars
from dataclasses import dataclass, field, replace


@dataclass
class ExecutionContext:
    run_id: str | None = None
    execution_id: str | None = None
    parent_execution_id: str | None = None
    session_id: str | None = None
    actor_id: str | None = None
    agent_did: str | None = None
    # where to deliver the terminal status callback; set by the control
    # plane (X-AgentField-Callback) so callbacks land on the SAME control
    # plane worker that holds the sync waiter (sticky routing across a
    # multi-worker plane)
    callback_url: str | None = None
    extras: dict = field(default_factory=dict)

    def to_headers(self) -> dict[str, str]:
        h = {}
        if self.run_id:
            h["X-Run-ID"] = self.run_id
        if self.execution_id:
            h["X-Execution-ID"] = self.execution_id
        if self.parent_execution_id:
            h["X-Parent-Execution-ID"] = self.parent_execution_id
        if self.session_id:
            h["X-Session-ID"] = self.session_id
        if self.actor_id:
            h["X-Actor-ID"] = self.actor_id
        return h

    def child_headers(self) -> dict[str, str]:
        """Headers for a nested call: current execution becomes the parent."""
        h = self.to_headers()
        h.pop("X-Execution-ID", None)
        if self.execution_id:
            h["X-Parent-Execution-ID"] = self.execution_id
        return h

    @classmethod
    def from_headers(cls, headers) -> "ExecutionContext":
        get = headers.get
        return cls(
            run_id=get("x-run-id") or get("X-Run-ID"),
            execution_id=get("x-execution-id") or get("X-Execution-ID"),
            parent_execution_id=(get("x-parent-execution-id")
                                 or get("X-Parent-Execution-ID")),
            session_id=get("x-session-id") or get("X-Session-ID"),
            actor_id=get("x-actor-id") or get("X-Actor-ID"),
            callback_url=(get("x-agentfield-callback")
                          or get("X-AgentField-Callback")),
        )

    def child(self, execution_id: str | None = None) -> "ExecutionContext":
        return replace(self, parent_execution_id=self.execution_id,
                       execution_id=execution_id)


_current: contextvars.ContextVar[ExecutionContext | None] = \
    contextvars.ContextVar("af_execution_context", default=None)


def current_context() -> ExecutionContext | None:
    return _current.get()


def set_context(ctx: ExecutionContext | None):
    return _current.set(ctx)


def reset_context(token) -> None:
    _current.reset(token)
