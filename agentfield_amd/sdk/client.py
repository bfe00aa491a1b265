"""Control-plane REST client used by the SDK (reference parity: client.py
AgentFieldClient + async_execution_manager.py in compact form: submit,
sync execute, batch-status polling with adaptive backoff, SSE nudge)."""
from __future__ import annotations

import json
import time

import httpx


class AgentFieldClient:
    def __init__(self, base_url: str, timeout: float = 95.0):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self._client = httpx.Client(timeout=timeout)
        self._aclient: httpx.AsyncClient | None = None

    # -------------------------------------------------------- registration
    def register_agent(self, node: dict) -> dict:
        r = self._client.post(f"{self.base_url}/api/v1/nodes/register",
                              json=node)
        r.raise_for_status()
        return r.json()

    def claim_actions(self, node_id: str, lease_s: float = 30.0,
                      max_n: int = 16) -> list[dict]:
        r = self._client.post(
            f"{self.base_url}/api/v1/nodes/{node_id}/actions/claim",
            json={"lease_s": lease_s, "max": max_n})
        r.raise_for_status()
        return r.json().get("actions", [])

    def ack_action(self, node_id: str, action_id: int,
                   status: str = "done") -> bool:
        r = self._client.post(
            f"{self.base_url}/api/v1/nodes/{node_id}/actions/ack",
            json={"action_id": action_id, "status": status})
        return r.status_code == 200

    def heartbeat(self, node_id: str, payload: dict | None = None) -> bool:
        try:
            r = self._client.post(
                f"{self.base_url}/api/v1/nodes/{node_id}/heartbeat",
                json=payload or {"status": "active"})
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    def update_node_status(self, node_id: str, status: str) -> bool:
        try:
            r = self._client.post(
                f"{self.base_url}/api/v1/nodes/{node_id}/status",
                json={"status": status})
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    # ----------------------------------------------------------- execution
    def execute_sync(self, target: str, input: dict,
                     headers: dict | None = None,
                     webhook: dict | None = None) -> dict:
        body = {"input": input}
        if webhook:
            body["webhook"] = webhook
        r = self._client.post(f"{self.base_url}/api/v1/execute/{target}",
                              json=body, headers=headers or {})
        r.raise_for_status()
        return r.json()

    def execute_async(self, target: str, input: dict,
                      headers: dict | None = None,
                      webhook: dict | None = None) -> dict:
        body = {"input": input}
        if webhook:
            body["webhook"] = webhook
        r = self._client.post(f"{self.base_url}/api/v1/execute/async/{target}",
                              json=body, headers=headers or {})
        if r.status_code == 503:
            raise RuntimeError("control plane queue is full (backpressure)")
        r.raise_for_status()
        return r.json()

    def get_execution(self, execution_id: str) -> dict | None:
        r = self._client.get(f"{self.base_url}/api/v1/executions/{execution_id}")
        return r.json() if r.status_code == 200 else None

    def batch_status(self, ids: list[str]) -> dict:
        r = self._client.post(f"{self.base_url}/api/v1/executions/batch-status",
                              json={"execution_ids": ids})
        r.raise_for_status()
        return r.json()

    def wait_for_result(self, execution_id: str, timeout: float | None = None,
                        poll_initial: float | None = None,
                        poll_max: float | None = None,
                        use_sse: bool | None = None) -> dict:
        """Adaptive polling with an SSE nudge: a background listener on
        /api/ui/v1/executions/events wakes the poll loop the moment the
        terminal event fires (reference async_execution_manager.py:644)."""
        import os
        import threading
        # env-tunable like the reference's AsyncConfig.from_environment (P8)
        timeout = timeout if timeout is not None else float(
            os.environ.get("AGENTFIELD_RESULT_TIMEOUT", 300.0))
        poll_initial = poll_initial if poll_initial is not None else float(
            os.environ.get("AGENTFIELD_POLL_INITIAL", 0.05))
        poll_max = poll_max if poll_max is not None else float(
            os.environ.get("AGENTFIELD_POLL_MAX", 2.0))
        if use_sse is None:
            use_sse = os.environ.get("AGENTFIELD_SSE_NUDGE", "1") != "0"
        deadline = time.time() + timeout
        poll = poll_initial
        nudge = threading.Event()
        stop_sse = threading.Event()

        def sse_listener():
            try:
                with httpx.stream(
                        "GET", f"{self.base_url}/api/ui/v1/executions/events",
                        timeout=timeout) as resp:
                    for line in resp.iter_lines():
                        if stop_sse.is_set():
                            return
                        if line.startswith("data:") and execution_id in line:
                            nudge.set()
                            return
            except httpx.HTTPError:
                pass  # SSE is best-effort; polling still covers us

        if use_sse:
            threading.Thread(target=sse_listener, daemon=True).start()
        try:
            while time.time() < deadline:
                rec = self.get_execution(execution_id)
                if rec and rec.get("status") in ("completed", "failed",
                                                 "timeout", "cancelled"):
                    return rec
                if nudge.wait(poll):
                    nudge.clear()
                    poll = poll_initial  # event fired: check immediately
                else:
                    poll = min(poll * 1.5, poll_max)
        finally:
            stop_sse.set()
        raise TimeoutError(f"execution {execution_id} did not finish")

    def report_status(self, execution_id: str, status: str, result=None,
                      error: str | None = None,
                      duration_ms: float | None = None) -> bool:
        try:
            r = self._client.post(
                f"{self.base_url}/api/v1/executions/{execution_id}/status",
                json={"execution_id": execution_id, "status": status,
                      "result": result, "error": error,
                      "duration_ms": duration_ms})
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    def workflow_event(self, event: dict) -> bool:
        try:
            r = self._client.post(
                f"{self.base_url}/api/v1/workflow/executions/events", json=event)
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    # -------------------------------------------------------------- memory
    def memory_op(self, op: str, body: dict, headers: dict | None = None):
        r = self._client.post(f"{self.base_url}/api/v1/memory/{op}", json=body,
                              headers=headers or {})
        r.raise_for_status()
        return r.json()

    def memory_list(self, params: dict, headers: dict | None = None):
        r = self._client.get(f"{self.base_url}/api/v1/memory/list",
                             params=params, headers=headers or {})
        r.raise_for_status()
        return r.json()

    def lock_op(self, op: str, body: dict) -> dict:
        r = self._client.post(f"{self.base_url}/api/v1/locks/{op}", json=body)
        r.raise_for_status()
        return r.json()

    # --------------------------------------------------------------- DID/VC
    def did_register(self, node_id: str, reasoners: list[str],
                     skills: list[str]) -> dict:
        r = self._client.post(f"{self.base_url}/api/v1/did/register",
                              json={"node_id": node_id, "reasoners": reasoners,
                                    "skills": skills})
        r.raise_for_status()
        return r.json()

    def create_execution_vc(self, execution_id: str, **kw) -> dict:
        r = self._client.post(f"{self.base_url}/api/v1/execution/vc",
                              json={"execution_id": execution_id, **kw})
        r.raise_for_status()
        return r.json()

    def vc_chain(self, run_id: str) -> dict:
        r = self._client.get(
            f"{self.base_url}/api/v1/did/workflow/{run_id}/vc-chain")
        r.raise_for_status()
        return r.json()

    def close(self):
        self._client.close()
