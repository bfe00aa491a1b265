"""Background services: webhook dispatcher (HMAC + DB-backed retry),
presence leases, health monitor, execution cleanup, payload store, metrics.

Reference parity map: C5 webhook_dispatcher, C6 cleanup,
C8-C10 status/health/presence, C11 payload store, C26 Prometheus metrics.
"""
from __future__ import annotations

import asyncio
import hashlib
import hmac
import json
import time
from pathlib import Path

import httpx
from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest)

from . import status as st
from ..logging_setup import get_logger

log = get_logger("controlplane.webhooks")


class Metrics:
    """Prometheus metrics with the reference's metric names (C26)."""

    def __init__(self):
        self.registry = CollectorRegistry()
        self.queue_depth = Gauge("agentfield_gateway_queue_depth",
                                 "async execution queue depth",
                                 registry=self.registry)
        self.worker_inflight = Gauge("agentfield_worker_inflight",
                                     "in-flight async executions",
                                     registry=self.registry)
        self.waiters_inflight = Gauge("agentfield_waiters_inflight",
                                      "sync waiters blocked on completion",
                                      registry=self.registry)
        self.backpressure = Counter("agentfield_gateway_backpressure_total",
                                    "rejected submissions (queue full)",
                                    registry=self.registry)
        self.step_duration = Histogram("agentfield_step_duration_seconds",
                                       "execution duration", registry=self.registry)
        self.step_retries = Counter("agentfield_step_retries_total",
                                    "execution retries", registry=self.registry)
        # engine-side metrics, aggregated from agent heartbeats (each
        # agent with an in-process EngineRunner reports its engine's
        # counters; labeled by node so DP fleets don't clobber each other)
        self.engine_tokens = Counter("agentfield_engine_tokens_total",
                                     "tokens generated", ["node", "kind"],
                                     registry=self.registry)
        self.engine_batch = Gauge("agentfield_engine_batch_occupancy",
                                  "sequences in the running batch",
                                  ["node"], registry=self.registry)
        self.engine_kv_pages = Gauge("agentfield_engine_kv_free_pages",
                                     "free KV pages", ["node"],
                                     registry=self.registry)

    def record_engine_heartbeat(self, node_id: str, eng: dict) -> None:
        """Fold an agent heartbeat's engine snapshot into the gauges;
        token counters advance by the reported deltas."""
        for kind in ("prefill", "decode"):
            d = eng.get(f"{kind}_tokens_delta")
            if d:
                self.engine_tokens.labels(node=node_id, kind=kind).inc(d)
        if "running" in eng:
            self.engine_batch.labels(node=node_id).set(eng["running"])
        if "kv_free_pages" in eng:
            self.engine_kv_pages.labels(node=node_id).set(
                eng["kv_free_pages"])

    def render(self) -> bytes:
        return generate_latest(self.registry)


class PayloadStore:
    """Large input/result payloads on disk; executions hold URIs (C11)."""

    def __init__(self, root: str | None, inline_limit: int = 32 * 1024):
        self.root = Path(root) if root else None
        self.inline_limit = inline_limit
        if self.root:
            self.root.mkdir(parents=True, exist_ok=True)

    def maybe_offload(self, execution_id: str, kind: str, payload) -> tuple:
        """Returns (inline_payload, uri)."""
        raw = json.dumps(payload).encode()
        if self.root is None or len(raw) <= self.inline_limit:
            return payload, None
        p = self.root / f"{execution_id}.{kind}.json"
        p.write_bytes(raw)
        return None, str(p)

    def load(self, uri: str):
        return json.loads(Path(uri).read_bytes())


def sign_payload(secret: str, body: bytes) -> str:
    mac = hmac.new(secret.encode(), body, hashlib.sha256).hexdigest()
    return f"sha256={mac}"


class WebhookDispatcher:
    """HMAC-SHA256-signed webhook delivery with durable DB-backed retries
    (C5): immediate dispatch on Notify, plus a poller that re-drives due
    webhooks every poll_interval (warm start re-drives after restart)."""

    def __init__(self, storage, metrics: Metrics, *, workers: int = 4,
                 timeout: float = 10.0, max_attempts: int = 5,
                 backoff_base: float = 5.0, backoff_max: float = 300.0,
                 poll_interval: float = 5.0):
        self.storage = storage
        self.metrics = metrics
        self.timeout = timeout
        self.max_attempts = max_attempts
        self.backoff_base = backoff_base
        self.backoff_max = backoff_max
        self.poll_interval = poll_interval
        self._queue: asyncio.Queue = asyncio.Queue(256)
        self._workers = workers
        self._tasks: list[asyncio.Task] = []
        self._client: httpx.AsyncClient | None = None

    async def start(self):
        self._client = httpx.AsyncClient(timeout=self.timeout)
        for _ in range(self._workers):
            self._tasks.append(asyncio.create_task(self._worker()))
        self._tasks.append(asyncio.create_task(self._poller()))

    async def stop(self):
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        if self._client:
            await self._client.aclose()

    def build_payload(self, execution: dict) -> dict:
        """Wire shape per SURVEY.md A.4."""
        status = execution.get("status")
        event = ("execution.completed" if status == st.COMPLETED
                 else "execution.failed")
        payload = {
            "event": event,
            "execution_id": execution["id"],
            "workflow_id": execution.get("run_id"),
            "status": status,
            "target": f"{execution.get('node_id')}.{execution.get('reasoner_id')}",
            "type": execution.get("target_type", "reasoner"),
            "duration_ms": execution.get("duration_ms"),
            "timestamp": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
        }
        if status == st.COMPLETED:
            payload["result"] = execution.get("result")
        else:
            payload["error_message"] = execution.get("error_message")
        return payload

    def notify(self, execution: dict) -> None:
        """Called when an execution reaches a terminal state."""
        wh = self.storage.get_webhook(execution["id"])
        if not wh or wh["status"] in ("delivered", "failed", "inflight"):
            return
        self.storage.stage_webhook_payload(execution["id"],
                                           self.build_payload(execution))
        if self.storage.try_mark_webhook_inflight(execution["id"]):
            try:
                self._queue.put_nowait(execution["id"])
            except asyncio.QueueFull:
                # poller will pick it up: reset to retry
                self.storage.webhook_attempted(
                    execution["id"], False, 0, "dispatch queue full", 0.0,
                    self.max_attempts + 1)

    async def _worker(self):
        while True:
            execution_id = await self._queue.get()
            try:
                await self._deliver(execution_id)
            except Exception as e:  # a worker must never die w/ claims
                log.warning("webhook delivery crashed",
                            execution_id=execution_id, error=str(e))

    async def _deliver(self, execution_id: str):
        wh = self.storage.get_webhook(execution_id)
        if not wh:
            return
        if not wh.get("payload"):
            # claimed before the execution completed: release, the
            # completion-time notify() re-drives it
            self.storage.release_webhook_claim(execution_id)
            return
        body = wh["payload"].encode() if isinstance(wh["payload"], str) \
            else json.dumps(wh["payload"]).encode()
        headers = {"Content-Type": "application/json", **(wh["headers"] or {})}
        if wh.get("secret"):
            headers["X-AgentField-Signature"] = sign_payload(wh["secret"], body)
        backoff = min(self.backoff_max,
                      self.backoff_base * (2 ** wh["attempts"]))
        try:
            resp = await self._client.post(wh["url"], content=body,
                                           headers=headers)
            ok = 200 <= resp.status_code < 300
            self.storage.webhook_attempted(execution_id, ok, resp.status_code,
                                           None if ok else f"HTTP {resp.status_code}",
                                           backoff, self.max_attempts)
        except Exception as e:
            self.storage.webhook_attempted(execution_id, False, 0, str(e)[:500],
                                           backoff, self.max_attempts)

    async def _poller(self):
        while True:
            await asyncio.sleep(self.poll_interval)
            try:
                for wh in self.storage.due_webhooks():
                    if self.storage.try_mark_webhook_inflight(wh["execution_id"]):
                        await self._deliver(wh["execution_id"])
            except Exception:
                pass


class PresenceManager:
    """Heartbeat lease table: nodes turn inactive after ttl, evicted after
    hard_evict (C10)."""

    def __init__(self, storage, buses, *, ttl: float = 300.0,
                 sweep_interval: float = 30.0, hard_evict: float = 1800.0):
        self.storage = storage
        self.buses = buses
        self.ttl = ttl
        self.sweep_interval = sweep_interval
        self.hard_evict = hard_evict
        self._task: asyncio.Task | None = None

    async def start(self):
        self._task = asyncio.create_task(self._loop())

    async def stop(self):
        if self._task:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass

    def sweep_once(self, now: float | None = None) -> list[str]:
        now = now or time.time()
        changed = []
        for node in self.storage.list_nodes():
            hb = node.get("last_heartbeat") or 0
            if node["status"] in ("active", "starting") and now - hb > self.ttl:
                self.storage.set_node_status(node["id"], "inactive")
                changed.append(node["id"])
                self.buses.node.publish({"type": "node.inactive",
                                         "node_id": node["id"]})
            elif now - hb > self.hard_evict and node["status"] != "stopped":
                self.storage.set_node_status(node["id"], "stopped")
                changed.append(node["id"])
        return changed

    async def _loop(self):
        while True:
            await asyncio.sleep(self.sweep_interval)
            try:
                self.sweep_once()
            except Exception:
                pass


class HealthMonitor:
    """Active GET /health probes of registered agents (C9)."""

    def __init__(self, storage, buses, *, interval: float = 60.0,
                 timeout: float = 5.0):
        self.storage = storage
        self.buses = buses
        self.interval = interval
        self.timeout = timeout
        self._task: asyncio.Task | None = None

    async def start(self):
        self._task = asyncio.create_task(self._loop())

    async def stop(self):
        if self._task:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass

    async def probe_all(self):
        async with httpx.AsyncClient(timeout=self.timeout) as client:
            for node in self.storage.list_nodes():
                if node["status"] in ("stopped", "stopping") or not node["base_url"]:
                    continue
                try:
                    r = await client.get(node["base_url"].rstrip("/") + "/health")
                    healthy = r.status_code == 200
                except Exception:
                    healthy = False
                if not healthy and node["status"] == "active":
                    self.storage.set_node_status(node["id"], "unhealthy")
                    self.buses.node.publish({"type": "node.unhealthy",
                                             "node_id": node["id"]})
                elif healthy and node["status"] in ("unhealthy", "inactive"):
                    self.storage.set_node_status(node["id"], "active")
                    self.buses.node.publish({"type": "node.active",
                                             "node_id": node["id"]})

    async def _loop(self):
        while True:
            await asyncio.sleep(self.interval)
            try:
                await self.probe_all()
            except Exception:
                pass


class ExecutionCleanup:
    """Deletes terminal executions past retention; marks stale running ones
    failed (C6)."""

    def __init__(self, storage, *, retention_s: float = 24 * 3600,
                 stale_s: float = 1800, interval: float = 3600,
                 batch: int = 100):
        self.storage = storage
        self.retention_s = retention_s
        self.stale_s = stale_s
        self.interval = interval
        self.batch = batch
        self._task: asyncio.Task | None = None

    def run_once(self) -> dict:
        stale = self.storage.mark_stale_running(self.stale_s)
        deleted = self.storage.delete_old_executions(self.retention_s, self.batch)
        return {"stale_marked": stale, "deleted": deleted}

    async def start(self):
        async def loop():
            while True:
                await asyncio.sleep(self.interval)
                try:
                    self.run_once()
                except Exception:
                    pass
        self._task = asyncio.create_task(loop())

    async def stop(self):
        if self._task:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
