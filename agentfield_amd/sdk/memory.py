"""Hierarchical shared memory client (reference parity: memory.py — KV +
vector + change events, scoped global/workflow/session/actor with default
scope resolved from the current execution context)."""
from __future__ import annotations

import fnmatch
import json
import threading
import time

from .execution_context import current_context


class ScopedMemory:
    def __init__(self, iface: "MemoryInterface", scope: str,
                 scope_id: str | None = None):
        self._iface = iface
        self._scope = scope
        self._scope_id = scope_id

    def _body(self, **kw) -> dict:
        body = dict(kw)
        body["scope"] = self._scope
        sid = self._scope_id or self._iface._scope_id_for(self._scope)
        if sid:
            body["scope_id"] = sid
        return body

    def set(self, key: str, value):
        return self._iface.client.memory_op("set", self._body(key=key, value=value))

    def get(self, key: str, default=None):
        r = self._iface.client.memory_op("get", self._body(key=key))
        return r["value"] if r.get("found") else default

    def delete(self, key: str) -> bool:
        return self._iface.client.memory_op("delete", self._body(key=key))["deleted"]

    def keys(self, prefix: str = "") -> list[str]:
        body = self._body()
        return self._iface.client.memory_list(
            {"prefix": prefix, **body})["keys"]

    # vector ops share the scope
    def vector_set(self, key: str, embedding, metadata=None):
        return self._iface.client.memory_op(
            "vector/set", self._body(key=key, embedding=list(embedding),
                                     metadata=metadata))

    def vector_search(self, embedding, top_k: int = 5, metric: str = "cosine",
                      filters=None):
        return self._iface.client.memory_op(
            "vector/search", self._body(embedding=list(embedding), top_k=top_k,
                                        metric=metric, filters=filters))["results"]

    def vector_delete(self, key: str) -> bool:
        return self._iface.client.memory_op("vector/delete",
                                            self._body(key=key))["deleted"]


class DistributedLock:
    """Cross-process lease lock served by the control plane (storage-level
    atomic UPSERT; reference: the storage provider's lock table).  Usable
    as a context manager; the lease auto-refreshes on `refresh()` and
    expires server-side if the holder dies."""

    def __init__(self, client, name: str, owner: str, ttl_s: float = 30.0,
                 timeout_s: float = 60.0, poll_s: float = 0.2):
        self.client = client
        self.name = name
        self.owner = owner
        self.ttl_s = ttl_s
        self.timeout_s = timeout_s
        self.poll_s = poll_s

    def acquire(self, block: bool = True) -> bool:
        deadline = time.time() + self.timeout_s
        while True:
            r = self.client.lock_op("acquire", {"name": self.name,
                                                "owner": self.owner,
                                                "ttl_s": self.ttl_s})
            if r.get("acquired"):
                return True
            if not block or time.time() >= deadline:
                return False
            time.sleep(self.poll_s)

    def refresh(self) -> bool:
        return self.client.lock_op("refresh", {
            "name": self.name, "owner": self.owner,
            "ttl_s": self.ttl_s}).get("refreshed", False)

    def release(self) -> bool:
        return self.client.lock_op("release", {
            "name": self.name, "owner": self.owner}).get("released", False)

    def __enter__(self) -> "DistributedLock":
        if not self.acquire():
            raise TimeoutError(f"lock {self.name!r} not acquired "
                               f"within {self.timeout_s}s")
        return self

    def __exit__(self, *exc):
        self.release()
        return False


class MemoryInterface:
    """app.memory — default scope follows the current execution context
    (workflow > session > actor > global)."""

    def __init__(self, client, node_id: str):
        self.client = client
        self.node_id = node_id
        self._watchers: list[tuple[str, callable]] = []
        self._watch_thread: threading.Thread | None = None
        self._stop = threading.Event()

    def _scope_id_for(self, scope: str) -> str | None:
        ctx = current_context()
        if ctx is None:
            return None
        return {"workflow": ctx.run_id, "session": ctx.session_id,
                "actor": ctx.actor_id}.get(scope)

    def _default_scope(self) -> ScopedMemory:
        ctx = current_context()
        if ctx and ctx.run_id:
            return self.workflow
        if ctx and ctx.session_id:
            return self.session
        if ctx and ctx.actor_id:
            return self.actor
        return self.globals

    @property
    def workflow(self) -> ScopedMemory:
        return ScopedMemory(self, "workflow")

    @property
    def session(self) -> ScopedMemory:
        return ScopedMemory(self, "session")

    @property
    def actor(self) -> ScopedMemory:
        return ScopedMemory(self, "actor")

    @property
    def globals(self) -> ScopedMemory:
        return ScopedMemory(self, "global", "global")

    # default-scope conveniences
    def set(self, key: str, value):
        return self._default_scope().set(key, value)

    def get(self, key: str, default=None):
        return self._default_scope().get(key, default)

    def delete(self, key: str) -> bool:
        return self._default_scope().delete(key)

    def keys(self, prefix: str = "") -> list[str]:
        return self._default_scope().keys(prefix)

    # ------------------------------------------------------ change events
    def on_change(self, pattern: str = "*"):
        """Decorator: watch memory keys (glob pattern).  Events arrive over
        the server's SSE stream (/api/v1/memory/events/sse, push latency);
        if the stream drops, missed events are recovered from
        /api/v1/memory/events/history before reconnecting."""
        def deco(fn):
            self._watchers.append((pattern, fn))
            self._ensure_watch_thread()
            return fn
        return deco

    def _dispatch_event(self, ev: dict, since: float) -> float:
        since = max(since, ev.get("at", since))
        for pattern, fn in self._watchers:
            if fnmatch.fnmatch(ev.get("key", ""), pattern):
                try:
                    val = ev.get("value")
                    if isinstance(val, str):
                        try:
                            val = json.loads(val)
                        except ValueError:
                            pass
                    fn({"key": ev.get("key"), "op": ev.get("op"),
                        "value": val, "scope": ev.get("scope")})
                except Exception:
                    pass
        return since

    def _ensure_watch_thread(self):
        if self._watch_thread is not None:
            return
        def loop():
            since = time.time()
            base = self.client.base_url
            import httpx
            while not self._stop.is_set():
                try:
                    with httpx.stream(
                            "GET", f"{base}/api/v1/memory/events/sse",
                            timeout=httpx.Timeout(5.0, read=30.0)) as resp:
                        # catch-up AFTER the stream is subscribed: anything
                        # published while disconnected comes from history,
                        # anything newer is already buffered on the stream
                        # (no window where an event can miss both).
                        try:
                            r = httpx.get(
                                f"{base}/api/v1/memory/events/history",
                                params={"since": since}, timeout=5.0)
                            for ev in r.json().get("events", []):
                                since = self._dispatch_event(ev, since)
                        except Exception:
                            pass
                        for line in resp.iter_lines():
                            if self._stop.is_set():
                                return
                            if not line.startswith("data:"):
                                continue  # keepalive comments etc.
                            try:
                                ev = json.loads(line[5:].strip())
                            except ValueError:
                                continue
                            if ev.get("at", 0) <= since:
                                continue  # already delivered via catch-up
                            since = self._dispatch_event(ev, since)
                except Exception:
                    pass
                self._stop.wait(0.5)
        self._watch_thread = threading.Thread(target=loop, daemon=True,
                                              name="af-memory-watch")
        self._watch_thread.start()

    def stop(self):
        self._stop.set()
