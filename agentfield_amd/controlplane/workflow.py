"""Workflow DAG assembly + status aggregation (reference parity: C12/C13,
workflow_dag.go:268-365, workflowstatus/aggregator.go:11-49)."""
from __future__ import annotations

from . import status as st


def aggregate_status(statuses: list[str]) -> str:
    """Fold execution statuses into a run status.  Running overrides all
    (work still in flight); otherwise any failure fails the run."""
    if not statuses:
        return "unknown"
    norm = [st.normalize(s) for s in statuses]
    if any(s in (st.RUNNING, st.PENDING, st.QUEUED) for s in norm):
        return st.RUNNING
    if any(s == st.FAILED for s in norm):
        return st.FAILED
    if any(s == st.TIMEOUT for s in norm):
        return st.TIMEOUT
    if any(s == st.CANCELLED for s in norm):
        return st.CANCELLED
    if all(s == st.COMPLETED for s in norm):
        return st.COMPLETED
    return st.RUNNING


def build_dag(executions: list[dict], lightweight: bool = False) -> dict:
    """Assemble the run DAG from flat execution rows via parent links."""
    nodes = []
    ids = {e["id"] for e in executions}
    for e in executions:
        n = {
            "execution_id": e["id"],
            "reasoner_id": e.get("reasoner_id"),
            "node_id": e.get("node_id"),
            "status": e.get("status"),
            "parent_execution_id": e.get("parent_execution_id"),
            "depth": e.get("depth", 0),
            "started_at": e.get("started_at"),
            "finished_at": e.get("finished_at"),
            "duration_ms": e.get("duration_ms"),
        }
        if not lightweight:
            n["input"] = e.get("input")
            n["result"] = e.get("result")
            n["error_message"] = e.get("error_message")
        nodes.append(n)
    edges = [{"from": e["parent_execution_id"], "to": e["id"]}
             for e in executions
             if e.get("parent_execution_id") and e["parent_execution_id"] in ids]
    roots = [e["id"] for e in executions if not e.get("parent_execution_id")
             or e["parent_execution_id"] not in ids]
    return {
        "run_id": executions[0].get("run_id") if executions else None,
        "status": aggregate_status([e.get("status") for e in executions]),
        "nodes": nodes,
        "edges": edges,
        "roots": roots,
        "total": len(nodes),
    }
