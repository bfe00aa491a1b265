"""DP replica routing: place reasoner model calls across N single-GPU
engine servers (config 4's topology — the GPU-side analog of the control
plane's async worker pool).

Routing is least-loaded by (queued + running) from /v1/stats with a cached
snapshot, falling back to round-robin; unhealthy replicas are skipped and
retried on the next refresh.
"""
from __future__ import annotations

import itertools
import json
import threading
import time

import httpx


class RemoteRunner:
    """Drop-in for sdk.ai.EngineRunner that talks to a remote engine server
    (or a DPRouter of them) instead of the in-process engine."""

    def __init__(self, router: "DPRouter"):
        self.router = router

    def generate_text(self, prompt: str, cfg) -> str:
        body = dict(prompt=prompt,
                    max_tokens=cfg.max_tokens,
                    temperature=cfg.temperature,
                    stop=list(cfg.stop or ()),
                    model=cfg.model,
                    json_mode=getattr(cfg, "json_only", False),
                    ignore_eos=getattr(cfg, "ignore_eos", False),
                    timeout=cfg.timeout)
        schema = getattr(cfg, "json_schema", None)
        if schema is not None:
            body["json_schema"] = schema
        return self.router.generate(**body)["text"]

    def stream_text(self, prompt: str, cfg):
        yield from self.router.stream(prompt=prompt, max_tokens=cfg.max_tokens,
                                      temperature=cfg.temperature,
                                      stop=list(cfg.stop or ()),
                                      model=cfg.model,
                                      timeout=cfg.timeout)


class DPRouter:
    def __init__(self, urls: list[str], refresh_s: float = 2.0,
                 timeout: float = 600.0):
        self.urls = [u.rstrip("/") for u in urls]
        self.refresh_s = refresh_s
        self.timeout = timeout
        self._rr = itertools.cycle(range(len(self.urls)))
        self._loads: dict[str, float] = {}
        self._models: dict[str, str | None] = {}
        self._healthy: dict[str, bool] = {u: True for u in self.urls}
        self._last_refresh = 0.0
        self._lock = threading.Lock()
        # hundreds of agent threads share this client under DP load; the
        # httpx default pool (100) would throttle them
        self._client = httpx.Client(
            timeout=timeout,
            limits=httpx.Limits(max_connections=1024,
                                max_keepalive_connections=1024))

    def _refresh(self):
        nowt = time.time()
        with self._lock:
            if nowt - self._last_refresh < self.refresh_s:
                return
            self._last_refresh = nowt
        for u in self.urls:
            try:
                r = self._client.get(u + "/v1/stats", timeout=1.0)
                s = r.json()
                self._loads[u] = s.get("queued", 0) + s.get("running", 0)
                self._models[u] = s.get("model")
                self._healthy[u] = True
            except Exception:
                self._healthy[u] = False

    def pick(self, model: str | None = None) -> str:
        """Least-loaded healthy replica; with `model`, restricted to the
        replicas that report serving it (heterogeneous fleets), falling
        back to the whole fleet when no replica advertises the model."""
        self._refresh()
        healthy = [u for u in self.urls if self._healthy.get(u, True)]
        if not healthy:
            healthy = self.urls  # try anyway
        if model is not None:
            serving = [u for u in healthy if self._models.get(u) == model]
            if serving:
                healthy = serving
        if self._loads:
            return min(healthy, key=lambda u: self._loads.get(u, 0))
        return healthy[next(self._rr) % len(healthy)]

    def generate(self, **body) -> dict:
        last_err = None
        for _ in range(min(3, len(self.urls))):
            url = self.pick(body.get("model"))
            try:
                r = self._client.post(url + "/v1/generate", json=body)
                if r.status_code == 200:
                    self._loads[url] = self._loads.get(url, 0) + 1
                    return r.json()
                last_err = f"{url}: HTTP {r.status_code}"
            except Exception as e:
                last_err = f"{url}: {e}"
                self._healthy[url] = False
        raise RuntimeError(f"all engine replicas failed: {last_err}")

    def stream(self, **body):
        body["stream"] = True
        url = self.pick(body.get("model"))
        with self._client.stream("POST", url + "/v1/generate",
                                 json=body) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:"):
                    ev = json.loads(line[5:])
                    if ev.get("text"):
                        yield ev["text"]
                    if ev.get("done"):
                        return

    def stats(self) -> dict:
        self._refresh()
        return {"replicas": self.urls, "healthy": self._healthy,
                "loads": self._loads}
