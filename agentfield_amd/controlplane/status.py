"""Execution status vocabulary + alias normalization (reference parity:
pkg/types/status.go:6-40) and the agent-node state machine (status_manager.go)."""
from __future__ import annotations

PENDING = "pending"
QUEUED = "queued"
RUNNING = "running"
COMPLETED = "completed"
FAILED = "failed"
TIMEOUT = "timeout"
CANCELLED = "cancelled"

TERMINAL = {COMPLETED, FAILED, TIMEOUT, CANCELLED}

_ALIASES = {
    "success": COMPLETED, "succeeded": COMPLETED, "complete": COMPLETED,
    "ok": COMPLETED, "done": COMPLETED, "finished": COMPLETED,
    "error": FAILED, "failure": FAILED, "fail": FAILED,
    "timed_out": TIMEOUT, "canceled": CANCELLED, "in_progress": RUNNING,
    "started": RUNNING, "processing": RUNNING, "accepted": QUEUED,
}


def normalize(status: str) -> str:
    s = (status or "").strip().lower()
    return _ALIASES.get(s, s)


def is_terminal(status: str) -> bool:
    return normalize(status) in TERMINAL


# --- agent node state machine (status_manager.go:449-471) ------------------
NODE_STATES = ("registered", "starting", "active", "inactive", "unhealthy",
               "stopping", "stopped")

_NODE_TRANSITIONS = {
    "registered": {"starting", "active", "inactive", "stopped"},
    "starting": {"active", "unhealthy", "stopped", "inactive"},
    "active": {"inactive", "unhealthy", "stopping", "stopped", "active"},
    "inactive": {"active", "starting", "stopped", "unhealthy"},
    "unhealthy": {"active", "inactive", "stopped"},
    "stopping": {"stopped", "active"},
    "stopped": {"starting", "registered", "active"},
}


def valid_node_transition(cur: str, nxt: str) -> bool:
    if cur == nxt:
        return True
    return nxt in _NODE_TRANSITIONS.get(cur, set())
