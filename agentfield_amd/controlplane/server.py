"""The control-plane HTTP server: the full agent API + UI API surface
(reference parity: internal/server/server.go setupRoutes).

Execution hot path: parse target -> execution record ->
HTTP call to the agent node -> 200-sync or 202+status-callback -> event-bus
wake of sync waiters -> webhook + VC on terminal status.
"""
from __future__ import annotations

import asyncio
import json
import time
from contextlib import asynccontextmanager

import httpx
from fastapi import FastAPI, Request, Response, WebSocket, WebSocketDisconnect
from fastapi.responses import JSONResponse, StreamingResponse

from . import status as st
from .did import DIDService, Keystore, VCService
from .events import Buses
from .ids import new_execution_id, new_run_id
from .services import (ExecutionCleanup, HealthMonitor, Metrics,
                       PayloadStore, PresenceManager, WebhookDispatcher)
from .storage import Storage
from .workflow import aggregate_status, build_dag
from ..logging_setup import get_logger

log = get_logger("controlplane")


class Config:
    """Precedence: env (AGENTFIELD_*) > explicit kwargs (YAML) > defaults
    (reference parity: C27 viper precedence)."""

    _ENV = {
        "db_path": ("AGENTFIELD_DATABASE_URL", str),
        "sync_timeout": ("AGENTFIELD_SYNC_TIMEOUT", float),
        "agent_timeout": ("AGENTFIELD_AGENT_TIMEOUT", float),
        "step_retries": ("AGENTFIELD_STEP_RETRIES", int),
        "step_retry_backoff_s": ("AGENTFIELD_STEP_RETRY_BACKOFF", float),
        "async_workers": ("AGENTFIELD_EXEC_ASYNC_WORKERS", int),
        "async_queue_capacity": ("AGENTFIELD_EXEC_ASYNC_QUEUE_CAPACITY", int),
        "did_enabled": ("AGENTFIELD_DID_ENABLED", lambda v: v not in ("0", "false")),
        "auto_vc": ("AGENTFIELD_AUTO_VC", lambda v: v in ("1", "true")),
        "presence_ttl": ("AGENTFIELD_PRESENCE_TTL", float),
        "health_interval": ("AGENTFIELD_HEALTH_INTERVAL", float),
        "admin_grpc_port": ("AGENTFIELD_ADMIN_GRPC_PORT", int),
        "public_url": ("AGENTFIELD_PUBLIC_URL", str),
        "background_services": ("AGENTFIELD_BACKGROUND_SERVICES",
                                lambda v: v not in ("0", "false")),
        "worker_urls": ("AGENTFIELD_WORKER_URLS",
                        lambda v: [u for u in v.split(",") if u]),
    }

    def __init__(self, **kw):
        import os as _os
        for field, (env, cast) in self._ENV.items():
            v = _os.environ.get(env)
            if v is not None:
                kw[field] = cast(v)
        self.db_path = kw.get("db_path", ":memory:")
        self.payload_dir = kw.get("payload_dir")
        self.keystore_path = kw.get("keystore_path")
        self.sync_timeout = kw.get("sync_timeout", 90.0)
        self.agent_timeout = kw.get("agent_timeout", 90.0)
        # transient-failure step retries (502/503/504 + connect errors)
        self.step_retries = kw.get("step_retries", 2)
        self.step_retry_backoff_s = kw.get("step_retry_backoff_s", 0.2)
        self.async_workers = kw.get("async_workers", 8)
        self.async_queue_capacity = kw.get("async_queue_capacity", 1024)
        self.did_enabled = kw.get("did_enabled", True)
        self.auto_vc = kw.get("auto_vc", False)
        self.presence_ttl = kw.get("presence_ttl", 300.0)
        self.health_interval = kw.get("health_interval", 60.0)
        self.background_services = kw.get("background_services", True)
        self.admin_grpc_port = kw.get("admin_grpc_port")
        # this worker's own externally-reachable URL; sent to agents as
        # X-AgentField-Callback so status callbacks stick to the worker
        # holding the sync waiter (enables multi-worker planes)
        self.public_url = kw.get("public_url")
        # the whole worker fleet (advertised via /api/v1/health so load
        # clients can round-robin without a front proxy)
        self.worker_urls = kw.get("worker_urls") or []


class ControlPlane:
    def __init__(self, cfg: Config | None = None):
        self.cfg = cfg or Config()
        self.storage = Storage(self.cfg.db_path)
        self.buses = Buses()
        self.metrics = Metrics()
        self.payloads = PayloadStore(self.cfg.payload_dir)
        self.webhooks = WebhookDispatcher(self.storage, self.metrics)
        self.presence = PresenceManager(self.storage, self.buses,
                                        ttl=self.cfg.presence_ttl)
        self.health = HealthMonitor(self.storage, self.buses,
                                    interval=self.cfg.health_interval)
        self.cleanup = ExecutionCleanup(self.storage)
        self.dids = DIDService(self.storage, Keystore(self.cfg.keystore_path)) \
            if self.cfg.did_enabled else None
        self.vcs = VCService(self.storage, self.dids) if self.dids else None
        self.client = None  # aiohttp.ClientSession (created in start())
        self._async_q: asyncio.Queue | None = None
        self._workers: list[asyncio.Task] = []
        self._grpc_server = None
        self.started_at = time.time()

    async def start(self):
        log.info("control plane starting (db=%s, workers=%d)",
                 self.cfg.db_path, self.cfg.async_workers)
        # aiohttp for the agent-dispatch hot path: its C http stack costs a
        # fraction of httpx per request (the control plane makes one
        # outbound call per execution)
        import aiohttp
        self.client = aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=self.cfg.agent_timeout),
            connector=aiohttp.TCPConnector(limit=512))
        self._async_q = asyncio.Queue(self.cfg.async_queue_capacity)
        for _ in range(self.cfg.async_workers):
            self._workers.append(asyncio.create_task(self._async_worker()))
        await self.webhooks.start()
        if self.cfg.admin_grpc_port:
            from .admin_grpc import start_admin_grpc
            self._grpc_server = start_admin_grpc(self,
                                                 port=self.cfg.admin_grpc_port)
        if self.cfg.background_services:
            await self.presence.start()
            await self.health.start()
            await self.cleanup.start()

    async def stop(self):
        for t in self._workers:
            t.cancel()
        await self.webhooks.stop()
        await self.presence.stop()
        await self.health.stop()
        await self.cleanup.stop()
        if self._grpc_server is not None:
            self._grpc_server.stop(grace=0.5)
        if self.client:
            await self.client.close()

    # ----------------------------------------------------- execution core
    async def probe_node(self, node: dict) -> bool:
        """One active GET /health probe (C9's single-node variant)."""
        import httpx
        if not node.get("base_url"):
            return False
        try:
            async with httpx.AsyncClient(timeout=5.0) as client:
                r = await client.get(node["base_url"].rstrip("/") + "/health")
                return r.status_code == 200
        except Exception:
            return False

    @staticmethod
    def parse_target(target: str) -> tuple[str, str]:
        if "." not in target:
            raise ValueError("target must be node_id.reasoner_id")
        node_id, reasoner_id = target.split(".", 1)
        return node_id, reasoner_id

    @staticmethod
    def _validate_webhook(wh: dict) -> str | None:
        if not wh.get("url"):
            return "webhook.url required"
        if len(wh.get("secret") or "") > 4096:
            return "webhook secret too large"
        headers = wh.get("headers") or {}
        if len(headers) > 20:
            return "too many webhook headers"
        for k, v in headers.items():
            if len(str(v)) > 512 or len(str(k)) > 512:
                return "webhook header too large"
        return None

    def _target_type(self, node: dict, reasoner_id: str) -> str:
        skills = {s.get("id") for s in node.get("skills", [])}
        return "skill" if reasoner_id in skills else "reasoner"

    def prepare_execution(self, target: str, body: dict,
                          headers) -> tuple[dict, dict | None]:
        node_id, reasoner_id = self.parse_target(target)
        node = self.storage.get_node(node_id)
        if node is None:
            return None, {"status_code": 404,
                          "error": f"agent node '{node_id}' not found"}
        webhook = body.get("webhook")
        if webhook:
            err = self._validate_webhook(webhook)
            if err:
                return None, {"status_code": 400, "error": err}
        run_id = headers.get("x-run-id") or new_run_id()
        rec = {
            "id": new_execution_id(),
            "run_id": run_id,
            "parent_execution_id": headers.get("x-parent-execution-id"),
            "node_id": node_id,
            "reasoner_id": reasoner_id,
            "target_type": self._target_type(node, reasoner_id),
            "status": st.RUNNING,
            "input": body.get("input", {}),
            "session_id": headers.get("x-session-id"),
            "actor_id": headers.get("x-actor-id"),
            "webhook_registered": bool(webhook),
            "_node": node,
            "_context": body.get("context"),
        }
        inline, uri = self.payloads.maybe_offload(rec["id"], "input",
                                                  rec["input"])
        stored = {**rec, "input": {"$payload_uri": uri}} if uri else rec
        self.storage.create_execution_full(stored, run_id, webhook)
        return rec, None

    async def call_agent(self, rec: dict) -> tuple[int, dict | None, str | None]:
        node = rec["_node"]
        path = "skills" if rec["target_type"] == "skill" else "reasoners"
        url = f"{node['base_url'].rstrip('/')}/{path}/{rec['reasoner_id']}"
        headers = {
            "X-Execution-ID": rec["id"],
            "X-Run-ID": rec["run_id"],
            "Content-Type": "application/json",
        }
        if rec.get("parent_execution_id"):
            headers["X-Parent-Execution-ID"] = rec["parent_execution_id"]
        if rec.get("session_id"):
            headers["X-Session-ID"] = rec["session_id"]
        if rec.get("actor_id"):
            headers["X-Actor-ID"] = rec["actor_id"]
        if self.cfg.public_url:
            # sticky callback routing: the terminal status callback must
            # land on THIS worker (it holds the sync waiter's future)
            headers["X-AgentField-Callback"] = self.cfg.public_url
        payload = dict(rec.get("input") or {})
        # bounded step retry on TRANSIENT failures (connect errors, 502/
        # 503/504 — an agent restarting under the process manager): the
        # reference's worker retries steps and counts them
        # (agentfield_step_retries_total).  4xx and agent-side exceptions
        # (500) are NOT transient and fail immediately.
        last_err = "agent unreachable"
        for attempt in range(self.cfg.step_retries + 1):
            if attempt:
                self.metrics.step_retries.inc()
                await asyncio.sleep(self.cfg.step_retry_backoff_s * attempt)
            try:
                async with self.client.post(url, json=payload,
                                            headers=headers) as resp:
                    if resp.status == 202:
                        return 202, None, None
                    if resp.status == 200:
                        try:
                            return 200, \
                                await resp.json(content_type=None), None
                        except ValueError:
                            return 200, {"raw": await resp.text()}, None
                    text = await resp.text()
                    last_err = f"agent HTTP {resp.status}: {text[:300]}"
                    if resp.status not in (502, 503, 504):
                        return resp.status, None, last_err
            except Exception as e:
                last_err = f"agent unreachable: {e}"
        return 0, None, last_err

    def complete_execution(self, execution_id: str, status: str, result=None,
                           error: str | None = None,
                           duration_ms: float | None = None) -> dict | None:
        status = st.normalize(status)
        if result is not None:
            _, uri = self.payloads.maybe_offload(execution_id, "result", result)
            if uri:
                result = {"$payload_uri": uri}
        applied, rec = self.storage.finalize_execution(
            execution_id, status, result, error, duration_ms,
            aggregate=aggregate_status)
        if rec is None:
            return None
        if not applied:
            return rec  # already terminal: no duplicate events/webhooks/VCs
        if rec.get("duration_ms"):
            self.metrics.step_duration.observe(rec["duration_ms"] / 1000.0)
        self.buses.publish_execution({
            "execution_id": execution_id, "status": status,
            "terminal": st.is_terminal(status), "run_id": rec.get("run_id"),
        })
        if st.is_terminal(status):
            self.buses.reasoner.publish({
                "type": "reasoner.execution", "node_id": rec.get("node_id"),
                "reasoner_id": rec.get("reasoner_id"), "status": status,
                "duration_ms": rec.get("duration_ms"),
            })
            self.webhooks.notify(rec)
            if self.cfg.auto_vc and self.vcs:
                try:
                    target_did = None
                    if rec.get("node_id"):
                        d = self.storage.did_for_subject("agent", rec["node_id"])
                        target_did = d["did"] if d else None
                    self.vcs.issue_execution_vc(rec, target_did=target_did)
                except Exception:
                    pass
        return rec

    async def _async_worker(self):
        while True:
            rec = await self._async_q.get()
            self.metrics.queue_depth.set(self._async_q.qsize())
            self.metrics.worker_inflight.inc()
            try:
                code, result, err = await self.call_agent(rec)
                if code == 200:
                    self.complete_execution(rec["id"], st.COMPLETED,
                                            result=result)
                elif code != 202:
                    self.complete_execution(rec["id"], st.FAILED, error=err)
                # 202: agent will deliver the terminal status callback
            finally:
                self.metrics.worker_inflight.dec()

    def _resolve_payload(self, v):
        if isinstance(v, dict) and "$payload_uri" in v:
            try:
                return self.payloads.load(v["$payload_uri"])
            except Exception:
                return v
        return v

    # ------------------------------------------------- hot-path handlers
    # Plain async functions (no framework Request/Response) so both the
    # FastAPI routes and the raw-ASGI fast path share one implementation.
    async def h_execute_sync(self, target: str, body: dict,
                             headers) -> tuple[int, dict, dict]:
        rec, err = self.prepare_execution(target, body, headers)
        if err:
            return err["status_code"], {"error": err["error"]}, {}
        t0 = time.time()
        # register the waiter BEFORE dispatching so a fast agent callback
        # can't race past it (the terminal event would otherwise be lost
        # and the request would ride out the full sync timeout)
        fut = self.buses.register_waiter(rec["id"])
        try:
            code, result, errmsg = await self.call_agent(rec)
            if code == 200:
                final = self.complete_execution(
                    rec["id"], st.COMPLETED, result=result,
                    duration_ms=(time.time() - t0) * 1e3)
            elif code == 202:
                self.metrics.waiters_inflight.inc()
                try:
                    ev = await self.buses.wait_for_execution(
                        rec["id"], self.cfg.sync_timeout, fut=fut)
                finally:
                    self.metrics.waiters_inflight.dec()
                final = self.storage.get_execution(rec["id"])
                if ev is None and not st.is_terminal(final.get("status", "")):
                    final = self.complete_execution(
                        rec["id"], st.TIMEOUT, error="sync wait timed out")
            else:
                final = self.complete_execution(rec["id"], st.FAILED,
                                                error=errmsg)
        finally:
            self.buses.discard_waiter(rec["id"], fut)
        return 200, self.envelope(final), {"X-Execution-ID": rec["id"],
                                           "X-Run-ID": rec["run_id"]}

    async def h_execute_async(self, target: str, body: dict,
                              headers) -> tuple[int, dict, dict]:
        rec, err = self.prepare_execution(target, body, headers)
        if err:
            return err["status_code"], {"error": err["error"]}, {}
        try:
            self._async_q.put_nowait(rec)
        except asyncio.QueueFull:
            self.metrics.backpressure.inc()
            self.complete_execution(rec["id"], st.FAILED,
                                    error="async queue is full")
            return 503, {"error": "queue is full"}, {}
        self.metrics.queue_depth.set(self._async_q.qsize())
        ts = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        return 202, {
            "execution_id": rec["id"], "run_id": rec["run_id"],
            "workflow_id": rec["run_id"], "status": "queued",
            "target": target, "type": rec["target_type"],
            "created_at": ts, "enqueued_at": ts,
        }, {"X-Execution-ID": rec["id"], "X-Run-ID": rec["run_id"]}

    def h_status_callback(self, execution_id: str,
                          body: dict) -> tuple[int, dict]:
        status = st.normalize(body.get("status", ""))
        if status not in (st.RUNNING, st.PENDING) and \
                not st.is_terminal(status):
            return 400, {"error": f"unknown status '{body.get('status')}'"}
        if st.is_terminal(status):
            self.complete_execution(execution_id, status, body.get("result"),
                                    body.get("error"),
                                    body.get("duration_ms"))
        return 200, {"status": "ok"}

    def h_get_execution(self, execution_id: str) -> tuple[int, dict]:
        rec = self.storage.get_execution(execution_id)
        if rec is None:
            return 404, {"error": "not found"}
        return 200, self.envelope(rec)

    def envelope(self, rec: dict) -> dict:
        rec = {**rec, "input": self._resolve_payload(rec.get("input")),
               "result": self._resolve_payload(rec.get("result"))}
        return {
            "execution_id": rec["id"],
            "run_id": rec.get("run_id"),
            "status": rec.get("status"),
            "result": rec.get("result"),
            "error_message": rec.get("error_message"),
            "duration_ms": rec.get("duration_ms"),
            "finished_at": rec.get("finished_at"),
            "webhook_registered": bool(rec.get("webhook_registered")),
        }


def _sse(event: dict) -> str:
    return f"data: {json.dumps(event)}\n\n"


class FastPathASGI:
    """Raw-ASGI dispatch for the execute hot loop.

    Starlette matches routes by scanning ~50 compiled regexes per request
    and FastAPI adds dependency resolution on top (profiled: ~45% of
    control-plane CPU under load, tools/stress.py).  The four hot
    endpoints — sync/async execute, the agent status callback and
    execution GET — are dispatched here with exact prefix checks and
    hand-rolled request/response handling; everything else falls through
    to the full FastAPI app unchanged."""

    _EXEC = "/api/v1/execute/"
    _EXEC_ASYNC = "/api/v1/execute/async/"
    _EXECUTIONS = "/api/v1/executions/"

    def __init__(self, app, cp: "ControlPlane"):
        self.app = app
        self.cp = cp

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http":
            return await self.app(scope, receive, send)
        path, method = scope["path"], scope["method"]
        cp = self.cp
        if method == "POST" and path.startswith(self._EXEC):
            headers = {k.decode("latin1"): v.decode("latin1")
                       for k, v in scope["headers"]}
            body = await self._read_json(receive)
            if path.startswith(self._EXEC_ASYNC):
                target = path[len(self._EXEC_ASYNC):]
                code, payload, hdrs = await cp.h_execute_async(
                    target, body, headers)
            else:
                target = path[len(self._EXEC):]
                code, payload, hdrs = await cp.h_execute_sync(
                    target, body, headers)
            return await self._respond(send, code, payload, hdrs)
        if path.startswith(self._EXECUTIONS):
            rest = path[len(self._EXECUTIONS):]
            if method == "POST" and rest.endswith("/status") and \
                    "/" not in rest[:-7]:
                body = await self._read_json(receive)
                code, payload = cp.h_status_callback(rest[:-7], body)
                return await self._respond(send, code, payload)
            if method == "GET" and "/" not in rest and rest:
                code, payload = cp.h_get_execution(rest)
                return await self._respond(send, code, payload)
        await self.app(scope, receive, send)

    @staticmethod
    async def _read_json(receive) -> dict:
        chunks = []
        while True:
            msg = await receive()
            chunks.append(msg.get("body", b""))
            if not msg.get("more_body"):
                break
        try:
            out = json.loads(b"".join(chunks))
            return out if isinstance(out, dict) else {}
        except (ValueError, TypeError):
            return {}

    @staticmethod
    async def _respond(send, code: int, payload: dict,
                       extra_headers: dict | None = None):
        body = json.dumps(payload).encode()
        headers = [(b"content-type", b"application/json"),
                   (b"content-length", str(len(body)).encode())]
        for k, v in (extra_headers or {}).items():
            headers.append((k.encode("latin1"), str(v).encode("latin1")))
        await send({"type": "http.response.start", "status": code,
                    "headers": headers})
        await send({"type": "http.response.body", "body": body})


def create_app(cp: ControlPlane | None = None, **cfg_kw) -> FastAPI:
    cp = cp or ControlPlane(Config(**cfg_kw))

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        await cp.start()
        yield
        await cp.stop()

    app = FastAPI(title="agentfield-amd control plane", lifespan=lifespan)
    app.state.cp = cp

    # ------------------------------------------------------------- basics
    @app.get("/api/v1/health")
    async def health():
        out = {"status": "healthy", "uptime_s": time.time() - cp.started_at,
               "version": "0.1.0"}
        if cp.cfg.worker_urls:
            out["workers"] = cp.cfg.worker_urls
        return out

    @app.get("/metrics")
    async def metrics():
        return Response(cp.metrics.render(), media_type="text/plain")

    # -------------------------------------------------------------- nodes
    @app.post("/api/v1/nodes/register")
    async def register_node(req: Request):
        body = await req.json()
        node_id = body.get("id") or body.get("node_id")
        if not node_id:
            return JSONResponse({"error": "id required"}, status_code=400)
        body["id"] = node_id
        cp.storage.upsert_node(body)
        cp.storage.set_node_status(node_id, "active")
        cp.buses.node.publish({"type": "node.registered", "node_id": node_id})
        out = {"status": "registered", "node_id": node_id,
               "base_url": body.get("base_url")}
        if cp.dids:
            reasoners = [r.get("id") for r in body.get("reasoners", [])]
            skills = [s.get("id") for s in body.get("skills", [])]
            out["identity"] = cp.dids.register_node(node_id, reasoners, skills)
        return out

    @app.get("/api/v1/nodes")
    async def list_nodes():
        return {"nodes": cp.storage.list_nodes()}

    @app.get("/api/v1/nodes/{node_id}")
    async def get_node(node_id: str):
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return node

    @app.post("/api/v1/nodes/{node_id}/heartbeat")
    async def heartbeat(node_id: str, req: Request):
        try:
            body = await req.json()
        except Exception:
            body = {}
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not registered",
                                 "action": "re-register"}, status_code=404)
        status = st.normalize(body.get("status") or "active")
        cp.storage.touch_heartbeat(node_id, "active" if status in
                                   ("active", "healthy", "completed") else None)
        if "mcp_servers" in body:
            # per-node MCP server health rides the enhanced heartbeat
            # (reference: HeartbeatHandler nodes.go:646 + health_monitor
            # MCP polling); surfaces at /api/ui/v1/mcp
            meta = dict(node.get("metadata") or {})
            meta["mcp_servers"] = body["mcp_servers"]
            cp.storage.set_node_metadata(node_id, meta)
        if isinstance(body.get("engine"), dict):
            # agents hosting an in-process engine report it: feeds the
            # agentfield_engine_* gauges on /metrics, per node, and the
            # dashboard's engines panel (latest snapshot in node meta)
            cp.metrics.record_engine_heartbeat(node_id, body["engine"])
            meta = dict(node.get("metadata") or {})
            meta["engine"] = {k: body["engine"][k]
                              for k in ("running", "kv_free_pages")
                              if k in body["engine"]}
            cp.storage.set_node_metadata(node_id, meta)
        return {"status": "ok"}

    @app.post("/api/v1/nodes/{node_id}/status")
    async def node_status(node_id: str, req: Request):
        body = await req.json()
        from .status import valid_node_transition
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        nxt = body.get("status")
        if nxt not in ("registered", "starting", "active", "inactive",
                       "unhealthy", "stopping", "stopped"):
            return JSONResponse({"error": f"invalid status {nxt}"},
                                status_code=400)
        if not valid_node_transition(node["status"], nxt):
            return JSONResponse(
                {"error": f"invalid transition {node['status']} -> {nxt}"},
                status_code=409)
        cp.storage.set_node_status(node_id, nxt)
        cp.buses.node.publish({"type": "node.status", "node_id": node_id,
                               "status": nxt})
        return {"status": "ok"}

    @app.delete("/api/v1/nodes/{node_id}")
    async def delete_node(node_id: str):
        cp.storage.delete_node(node_id)
        cp.buses.node.publish({"type": "node.removed", "node_id": node_id})
        return {"status": "ok"}

    @app.post("/api/v1/nodes/status/bulk")
    async def nodes_status_bulk(req: Request):
        body = await req.json()
        out = {}
        for nid in body.get("ids", []):
            node = cp.storage.get_node(nid)
            out[nid] = ({"status": node["status"],
                         "last_heartbeat": node.get("last_heartbeat")}
                        if node else None)
        return {"statuses": out}

    @app.post("/api/v1/nodes/{node_id}/status/refresh")
    async def node_status_refresh(node_id: str):
        """Probe the agent's /health NOW and reconcile its status."""
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        healthy = await cp.probe_node(node)
        nxt = "active" if healthy else "unhealthy"
        if node["status"] != nxt:
            cp.storage.set_node_status(node_id, nxt)
            cp.buses.node.publish({"type": "node.status",
                                   "node_id": node_id, "status": nxt})
        return {"node_id": node_id, "healthy": healthy, "status": nxt}

    @app.get("/api/v1/nodes/{node_id}/lifecycle")
    async def node_lifecycle(node_id: str):
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return {"node_id": node_id, "status": node["status"],
                "registered_at": node.get("registered_at"),
                "last_heartbeat": node.get("last_heartbeat"),
                "last_status_change": node.get("last_status_change"),
                "pending_actions": cp.storage.pending_actions(node_id)}

    def _lifecycle_action(node_id: str, action: str, status: str | None):
        node = cp.storage.get_node(node_id)
        if node is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        aid = cp.storage.enqueue_action(node_id, action)
        if status:
            cp.storage.set_node_status(node_id, status)
            cp.buses.node.publish({"type": "node.status",
                                   "node_id": node_id, "status": status})
        return {"status": "ok", "action_id": aid}

    @app.post("/api/v1/nodes/{node_id}/start")
    async def node_start(node_id: str):
        return _lifecycle_action(node_id, "start", "starting")

    @app.post("/api/v1/nodes/{node_id}/stop")
    async def node_stop(node_id: str):
        return _lifecycle_action(node_id, "stop", "stopping")

    @app.post("/api/v1/nodes/{node_id}/shutdown")
    async def node_shutdown(node_id: str):
        return _lifecycle_action(node_id, "shutdown", "stopping")

    @app.post("/api/v1/nodes/{node_id}/actions/claim")
    async def actions_claim(node_id: str, req: Request):
        try:
            body = await req.json()
        except Exception:
            body = {}
        acts = cp.storage.claim_actions(
            node_id, lease_s=float(body.get("lease_s", 30.0)),
            limit=int(body.get("max", 16)))
        return {"actions": acts}

    @app.post("/api/v1/nodes/{node_id}/actions/ack")
    async def actions_ack(node_id: str, req: Request):
        body = await req.json()
        ok = cp.storage.ack_action(int(body["action_id"]),
                                   body.get("status", "done"))
        if not ok:
            return JSONResponse({"error": "unknown or unclaimed action"},
                                status_code=409)
        return {"status": "ok"}

    # ------------------------------------------------------------ execute
    @app.post("/api/v1/execute/{target}")
    async def execute_sync(target: str, req: Request):
        try:
            body = await req.json()
        except Exception:
            body = {}
        code, payload, hdrs = await cp.h_execute_sync(target, body,
                                                      req.headers)
        return JSONResponse(payload, status_code=code, headers=hdrs)

    # legacy direct-execution routes (reference: POST /reasoners/:id and
    # /skills/:id with workflow headers, reasoners.go:45+): same sync
    # pipeline, target addressed by path segments instead of node.reasoner
    @app.post("/api/v1/reasoners/{node_id}/{reasoner_id}")
    async def execute_reasoner_direct(node_id: str, reasoner_id: str,
                                      req: Request):
        return await execute_sync(f"{node_id}.{reasoner_id}", req)

    @app.post("/api/v1/skills/{node_id}/{skill_id}")
    async def execute_skill_direct(node_id: str, skill_id: str,
                                   req: Request):
        return await execute_sync(f"{node_id}.{skill_id}", req)

    @app.post("/api/v1/execute/async/{target}")
    async def execute_async(target: str, req: Request):
        try:
            body = await req.json()
        except Exception:
            body = {}
        code, payload, hdrs = await cp.h_execute_async(target, body,
                                                       req.headers)
        return JSONResponse(payload, status_code=code, headers=hdrs)

    @app.get("/api/v1/executions/{execution_id}")
    async def get_execution(execution_id: str):
        code, payload = cp.h_get_execution(execution_id)
        return JSONResponse(payload, status_code=code)

    @app.post("/api/v1/executions/batch-status")
    async def batch_status(req: Request):
        body = await req.json()
        ids = body.get("execution_ids") or body.get("ids") or []
        found = cp.storage.batch_status(ids)
        return {"executions": {i: cp.envelope(found[i]) for i in found},
                "missing": [i for i in ids if i not in found]}

    @app.post("/api/v1/executions/{execution_id}/cancel")
    async def cancel_execution(execution_id: str):
        rec = cp.storage.get_execution(execution_id)
        if rec is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        if st.is_terminal(rec.get("status", "")):
            return JSONResponse({"error": "already terminal",
                                 "status": rec["status"]}, status_code=409)
        final = cp.complete_execution(execution_id, st.CANCELLED,
                                      error="cancelled by client")
        return cp.envelope(final)

    @app.post("/api/v1/executions/{execution_id}/status")
    async def execution_status_callback(execution_id: str, req: Request):
        """Agent-side terminal/progress status ingestion (A.2)."""
        body = await req.json()
        code, payload = cp.h_status_callback(execution_id, body)
        return JSONResponse(payload, status_code=code)

    # ------------------------------------------------- workflow events/DAG
    @app.post("/api/v1/executions/{execution_id}/notes")
    async def add_execution_note(execution_id: str, req: Request):
        if cp.storage.get_execution(execution_id) is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        body = await req.json()
        nid = cp.storage.add_note(execution_id, str(body.get("note", "")),
                                  author=str(body.get("author", "")))
        return {"status": "ok", "note_id": nid}

    @app.get("/api/v1/executions/{execution_id}/notes")
    async def list_execution_notes(execution_id: str):
        return {"notes": cp.storage.notes_for(execution_id)}

    @app.post("/api/v1/workflow/executions/events")
    async def workflow_events(req: Request):
        ev = await req.json()
        if not ev.get("execution_id"):
            return JSONResponse({"error": "execution_id required"},
                                status_code=400)
        owns = cp.storage.upsert_workflow_event(ev)
        status = st.normalize(ev.get("status") or "")
        if owns and st.is_terminal(status):
            cp.complete_execution(ev["execution_id"], status,
                                  ev.get("result"), ev.get("error"),
                                  ev.get("duration_ms"))
        run_id = ev.get("run_id") or ev.get("workflow_id")
        if run_id:
            sibs = cp.storage.executions_by_run(run_id)
            cp.storage.upsert_run(run_id,
                                  aggregate_status([e["status"] for e in sibs]))
        return {"status": "ok"}

    # -------------------------------------------------------------- memory
    def _scope(body: dict, headers) -> tuple[str, str]:
        """Scope resolution (A.5): explicit wins, else header priority."""
        if body.get("scope"):
            return body["scope"], body.get("scope_id") or "global"
        for hdr, scope in (("x-workflow-id", "workflow"),
                           ("x-run-id", "workflow"),
                           ("x-session-id", "session"),
                           ("x-actor-id", "actor")):
            v = headers.get(hdr)
            if v:
                return scope, v
        return "global", "global"

    @app.post("/api/v1/locks/acquire")
    async def lock_acquire(req: Request):
        """Cross-process lease lock (SQLite-atomic UPSERT): reentrant for
        the same owner, steals only expired leases."""
        body = await req.json()
        ok = cp.storage.lock_acquire(body["name"], body["owner"],
                                     float(body.get("ttl_s", 30.0)))
        return {"acquired": ok,
                "holder": cp.storage.lock_holder(body["name"])}

    @app.post("/api/v1/locks/release")
    async def lock_release(req: Request):
        body = await req.json()
        return {"released": cp.storage.lock_release(body["name"],
                                                    body["owner"])}

    @app.post("/api/v1/locks/refresh")
    async def lock_refresh(req: Request):
        body = await req.json()
        return {"refreshed": cp.storage.lock_refresh(
            body["name"], body["owner"], float(body.get("ttl_s", 30.0)))}

    @app.post("/api/v1/memory/set")
    async def memory_set(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        cp.storage.memory_set(scope, sid, body["key"], body.get("value"))
        cp.buses.memory.publish({"op": "set", "scope": scope, "scope_id": sid,
                                 "key": body["key"], "value": body.get("value"),
                                 "at": time.time()})
        return {"status": "ok", "scope": scope, "scope_id": sid}

    @app.post("/api/v1/memory/get")
    async def memory_get(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        value = cp.storage.memory_get(scope, sid, body["key"])
        return {"key": body["key"], "value": value, "found": value is not None,
                "scope": scope, "scope_id": sid}

    @app.post("/api/v1/memory/delete")
    async def memory_delete(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        deleted = cp.storage.memory_delete(scope, sid, body["key"])
        if deleted:
            cp.buses.memory.publish({"op": "delete", "scope": scope,
                                     "scope_id": sid, "key": body["key"],
                                     "at": time.time()})
        return {"deleted": deleted}

    @app.get("/api/v1/memory/list")
    async def memory_list(req: Request):
        scope, sid = _scope(dict(req.query_params), req.headers)
        keys = cp.storage.memory_list(scope, sid,
                                      req.query_params.get("prefix", ""))
        return {"keys": keys, "scope": scope, "scope_id": sid}

    @app.post("/api/v1/memory/vector/set")
    async def vector_set(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        cp.storage.vector_set(scope, sid, body["key"], body["embedding"],
                              body.get("metadata"))
        return {"status": "ok"}

    @app.post("/api/v1/memory/vector/search")
    async def vector_search(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        res = cp.storage.vector_search(scope, sid, body["embedding"],
                                       body.get("top_k", 5),
                                       body.get("metric", "cosine"),
                                       body.get("filters"))
        return {"results": res}

    @app.post("/api/v1/memory/vector/delete")
    async def vector_delete(req: Request):
        body = await req.json()
        scope, sid = _scope(body, req.headers)
        return {"deleted": cp.storage.vector_delete(scope, sid, body["key"])}

    @app.get("/api/v1/memory/events/history")
    async def memory_history(req: Request):
        since = float(req.query_params.get("since", 0))
        return {"events": cp.storage.memory_events_since(
            since, req.query_params.get("scope"))}

    @app.websocket("/api/v1/memory/events/ws")
    async def memory_ws(ws: WebSocket):
        await ws.accept()
        sid, q = cp.buses.memory.subscribe()
        try:
            while True:
                ev = await q.get()
                await ws.send_json(ev)
        except (WebSocketDisconnect, Exception):
            pass
        finally:
            cp.buses.memory.unsubscribe(sid)

    @app.get("/api/v1/memory/events/sse")
    async def memory_sse():
        sid, q = cp.buses.memory.subscribe()

        async def gen():
            try:
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), 15.0)
                        yield _sse(ev)
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                cp.buses.memory.unsubscribe(sid)
        return StreamingResponse(gen(), media_type="text/event-stream")

    # ------------------------------------------------------------- DID/VC
    @app.post("/api/v1/did/register")
    async def did_register(req: Request):
        if not cp.dids:
            return JSONResponse({"error": "DID disabled"}, status_code=400)
        body = await req.json()
        node_id = body.get("node_id") or body.get("agent_id")
        identity = cp.dids.register_node(
            node_id, body.get("reasoners", []), body.get("skills", []))
        return {"status": "ok", "identity": identity}

    @app.get("/api/v1/did/resolve/{did}")
    async def did_resolve(did: str):
        doc = cp.dids.resolve(did) if cp.dids else None
        if doc is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return {"did": did, "document": doc}

    @app.get("/api/v1/did/document/{did}")
    async def did_document_ep(did: str):
        doc = cp.dids.resolve(did) if cp.dids else None
        if doc is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return doc

    @app.get("/api/v1/did/status")
    async def did_status():
        if not cp.dids:
            return {"enabled": False}
        return {"enabled": True, "root_did": cp.dids.root_did,
                "agents": len(cp.storage.list_dids("agent")),
                "components": len(cp.storage.list_dids("component"))}

    @app.post("/api/v1/did/verify")
    async def did_verify(req: Request):
        body = await req.json()
        return VCService.verify_document(body.get("credential") or body)

    @app.post("/api/v1/execution/vc")
    async def create_execution_vc(req: Request):
        if not cp.vcs:
            return JSONResponse({"error": "DID disabled"}, status_code=400)
        body = await req.json()
        rec = cp.storage.get_execution(body.get("execution_id", ""))
        if rec is None:
            return JSONResponse({"error": "execution not found"},
                                status_code=404)
        doc = cp.vcs.issue_execution_vc(rec, body.get("caller_did"),
                                        body.get("target_did"))
        return {"status": "ok", "vc": doc}

    @app.get("/api/v1/executions/{execution_id}/vc")
    async def get_execution_vc(execution_id: str):
        rec = cp.storage.vc_for_execution(execution_id)
        if rec is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return {"vc": rec["document"],
                "verification": cp.vcs.verify_execution(execution_id)}

    @app.post("/api/ui/v1/executions/{execution_id}/verify-vc")
    async def verify_vc_comprehensive(execution_id: str):
        """Comprehensive integrity/security/compliance verification report
        (reference: VerifyExecutionVCComprehensive, vc_service.go:926,
        route server.go:767)."""
        if not cp.vcs:
            return JSONResponse({"error": "DID disabled"}, status_code=400)
        return cp.vcs.verify_execution_comprehensive(execution_id)

    @app.get("/api/v1/did/workflow/{run_id}/vc-chain/verify")
    async def verify_chain_comprehensive(run_id: str):
        if not cp.vcs:
            return JSONResponse({"error": "DID disabled"}, status_code=400)
        return cp.vcs.verify_chain_comprehensive(run_id)

    @app.get("/api/v1/did/workflow/{run_id}/vc-chain")
    async def vc_chain(run_id: str):
        if not cp.vcs:
            return JSONResponse({"error": "DID disabled"}, status_code=400)
        return cp.vcs.workflow_chain(run_id)

    @app.get("/api/v1/did/export/vcs")
    async def export_vcs(req: Request):
        run_id = req.query_params.get("workflow_id")
        vcs = cp.storage.vcs_for_run(run_id) if run_id else []
        return {"credentials": [v["document"] for v in vcs]}

    # ------------------------------------------------------------- UI API
    @app.get("/api/ui/v1/dashboard/summary")
    async def dashboard():
        nodes = cp.storage.list_nodes()
        recents = cp.storage.list_executions(limit=200)
        by_status: dict[str, int] = {}
        for e in recents:
            by_status[e["status"]] = by_status.get(e["status"], 0) + 1
        engines = {n["id"]: (n.get("metadata") or {}).get("engine")
                   for n in nodes
                   if (n.get("metadata") or {}).get("engine")}
        return {
            "nodes": {"total": len(nodes),
                      "active": sum(1 for n in nodes if n["status"] == "active")},
            "executions": {"recent": len(recents), "by_status": by_status},
            "engines": engines,  # latest heartbeat snapshot per GPU node
            "uptime_s": time.time() - cp.started_at,
        }

    @app.get("/api/ui/v1/nodes")
    async def ui_nodes():
        return {"nodes": cp.storage.list_nodes()}

    @app.get("/api/ui/v1/executions")
    async def ui_executions(req: Request):
        return {"executions": cp.storage.list_executions(
            limit=int(req.query_params.get("limit", 100)),
            node_id=req.query_params.get("node_id"),
            status=req.query_params.get("status"))}

    @app.get("/api/ui/v1/reasoners")
    async def ui_reasoners():
        out = []
        for n in cp.storage.list_nodes():
            for r in n.get("reasoners", []):
                out.append({"node_id": n["id"], **(r if isinstance(r, dict)
                                                  else {"id": r})})
        return {"reasoners": out}

    @app.get("/api/ui/v1/executions/events")
    async def execution_sse():
        sid, q = cp.buses.execution.subscribe()

        async def gen():
            try:
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), 15.0)
                        yield _sse(ev)
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                cp.buses.execution.unsubscribe(sid)
        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/api/ui/v1/nodes/events")
    async def node_sse():
        sid, q = cp.buses.node.subscribe()

        async def gen():
            try:
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), 15.0)
                        yield _sse(ev)
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                cp.buses.node.unsubscribe(sid)
        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/api/ui/v1/config")
    async def ui_config():
        """Sanitized server configuration (reference C33 env/config
        handlers): secrets and key material are never exposed."""
        cfg = cp.cfg
        hidden = {"keystore_key", "webhook_secret"}
        out = {}
        for k, v in vars(cfg).items():
            if k.startswith("_") or any(s in k.lower() for s in
                                        ("secret", "key", "token", "pass")):
                continue
            if isinstance(v, (str, int, float, bool, type(None))):
                out[k] = v
        return {"config": out, "hidden_fields": sorted(
            k for k in vars(cfg) if k not in out and not k.startswith("_"))}

    @app.get("/api/ui/v1/reasoners/events")
    async def reasoner_sse():
        sid, q = cp.buses.reasoner.subscribe()

        async def gen():
            try:
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), 15.0)
                        yield _sse(ev)
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                cp.buses.reasoner.unsubscribe(sid)
        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/api/ui/v1/mcp")
    async def ui_mcp():
        """MCP visibility: servers advertised by registered nodes'
        metadata (the SDK's MCPManager reports them at registration)."""
        servers = []
        for node in cp.storage.list_nodes():
            for s in (node.get("metadata") or {}).get("mcp_servers", []):
                servers.append({"node_id": node["id"], **(
                    s if isinstance(s, dict) else {"name": s})})
        return {"servers": servers}

    @app.get("/api/ui/v1/workflows/{run_id}/dag")
    async def workflow_dag(run_id: str, req: Request):
        execs = cp.storage.executions_by_run(run_id)
        if not execs:
            return JSONResponse({"error": "not found"}, status_code=404)
        light = req.query_params.get("lightweight") in ("1", "true")
        return build_dag(execs, lightweight=light)

    @app.get("/api/ui/v2/workflow-runs")
    async def workflow_runs(req: Request):
        return {"runs": cp.storage.list_runs(
            int(req.query_params.get("limit", 50)))}

    @app.get("/api/ui/v2/workflow-runs/{run_id}")
    async def workflow_run(run_id: str):
        run = cp.storage.get_run(run_id)
        if run is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        execs = cp.storage.executions_by_run(run_id)
        return {"run": run, "dag": build_dag(execs, lightweight=True)}

    @app.get("/api/ui/v1/executions/{execution_id}/timeline")
    async def execution_timeline(execution_id: str):
        """Per-execution timeline: the execution, its ancestors and children
        (C33 execution_timeline.go)."""
        rec = cp.storage.get_execution(execution_id)
        if rec is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        chain = []
        cur = rec
        seen = set()
        while cur and cur["id"] not in seen:  # ancestors
            seen.add(cur["id"])
            chain.append(cur)
            pid = cur.get("parent_execution_id")
            cur = cp.storage.get_execution(pid) if pid else None
        siblings = cp.storage.executions_by_run(rec.get("run_id")) \
            if rec.get("run_id") else []
        children = [e for e in siblings
                    if e.get("parent_execution_id") == execution_id]
        events = [{"at": rec.get("started_at"), "event": "started"},
                  {"at": rec.get("finished_at"), "event": rec.get("status")}]
        return {"execution": cp.envelope(rec),
                "ancestors": [cp.envelope(e) for e in chain[1:]],
                "children": [cp.envelope(e) for e in children],
                "events": [e for e in events if e["at"]],
                "webhook_history": cp.storage.webhook_history(execution_id)}

    @app.get("/api/ui/v1/activity/recent")
    async def recent_activity(req: Request):
        """Recent cross-entity activity feed (C33 recent_activity.go)."""
        limit = int(req.query_params.get("limit", 30))
        items = []
        for e in cp.storage.list_executions(limit=limit):
            items.append({
                "type": "execution", "at": e.get("finished_at")
                or e.get("created_at"), "id": e["id"],
                "summary": f"{e.get('node_id')}.{e.get('reasoner_id')} "
                           f"{e.get('status')}"})
        for n in cp.storage.list_nodes():
            items.append({"type": "node", "at": n.get("last_status_change"),
                          "id": n["id"],
                          "summary": f"node {n['id']} {n['status']}"})
        items = [i for i in items if i["at"]]
        items.sort(key=lambda x: -(x["at"] or 0))
        return {"activity": items[:limit]}

    @app.get("/api/ui/v1/reasoners/{node_id}/{reasoner_id}/metrics")
    async def reasoner_metrics(node_id: str, reasoner_id: str):
        """Per-reasoner performance metrics from execution history
        (reference GetReasonerPerformanceMetrics, storage.go:117)."""
        execs = [e for e in cp.storage.list_executions(limit=500,
                                                       node_id=node_id)
                 if e.get("reasoner_id") == reasoner_id]
        durs = sorted(e["duration_ms"] for e in execs
                      if e.get("duration_ms") is not None)
        n = len(execs)
        ok = sum(1 for e in execs if e["status"] == "completed")

        def pct(p):
            return durs[min(len(durs) - 1, int(p * len(durs)))] if durs else None
        return {"node_id": node_id, "reasoner_id": reasoner_id,
                "executions": n, "success_rate": (ok / n) if n else None,
                "p50_ms": pct(0.50), "p95_ms": pct(0.95), "p99_ms": pct(0.99)}

    # --------------------------------------------------- embedded web UI
    from pathlib import Path as _Path
    _ui = _Path(__file__).parent / "ui" / "index.html"

    @app.get("/")
    async def ui_index():
        return Response(_ui.read_text(), media_type="text/html")

    wrapped = FastPathASGI(app, cp)
    wrapped.state = app.state  # convenience passthrough (app.state.cp)
    return wrapped
