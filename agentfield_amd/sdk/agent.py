"""Agent(FastAPI): decorated Python functions become control-plane-routed
reasoners/skills (reference parity: sdk/python/agentfield/agent.py P1-P5).

Key behaviors reproduced:
  * @app.reasoner()/@app.skill() auto-generate POST /reasoners/{name} and
    /skills/{name} endpoints with signature-derived input validation
  * the 202+callback pattern: when the control plane sends X-Execution-ID,
    the endpoint returns 202 immediately and the reasoner runs in the
    background, reporting the terminal status to
    POST /api/v1/executions/{id}/status (agent.py:1182-1199)
  * app.ai() -> in-process MI355X engine (ai.py)
  * app.call() -> nested cross-agent call through the control plane with
    context-header propagation (builds the workflow DAG)
  * workflow start/complete events for in-process calls (P12)
  * registration + 30 s heartbeat thread with re-register on 404 (P10)
"""
from __future__ import annotations

import asyncio
import inspect
import os
import threading
import time
import traceback

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from .ai import AgentAI, AIConfig
from .client import AgentFieldClient
from .execution_context import (ExecutionContext, current_context,
                                reset_context, set_context)
from .memory import DistributedLock, MemoryInterface
from .resilience import ResultCache, StatelessRateLimiter


class _FunctionMeta:
    def __init__(self, fn, name: str, kind: str, tags=None, vc: bool = False):
        self.fn = fn
        self.name = name
        self.kind = kind
        self.tags = list(tags or [])
        self.vc = vc
        sig = inspect.signature(fn)
        self.params = [p for p in sig.parameters.values()
                       if p.kind in (p.POSITIONAL_OR_KEYWORD, p.KEYWORD_ONLY)]
        self.required = [p.name for p in self.params
                        if p.default is p.empty]
        self.is_async = asyncio.iscoroutinefunction(fn)

    def input_schema(self) -> dict:
        props = {}
        for p in self.params:
            t = {int: "integer", float: "number", bool: "boolean",
                 str: "string", list: "array", dict: "object"}.get(
                     p.annotation, "string")
            props[p.name] = {"type": t}
        return {"type": "object", "properties": props,
                "required": self.required}

    def descriptor(self) -> dict:
        return {"id": self.name, "input_schema": self.input_schema(),
                "output_schema": {"type": "object"}, "tags": self.tags,
                "vc_enabled": self.vc}


class Agent(FastAPI):
    def __init__(self, node_id: str,
                 agentfield_url: str | None = None,
                 team_id: str | None = None,
                 version: str = "0.1.0",
                 ai_config: AIConfig | None = None,
                 auto_register: bool = True,
                 heartbeat_interval: float = 30.0,
                 vc_enabled: bool = False,
                 base_url: str | None = None,
                 **fastapi_kw):
        super().__init__(title=f"agent:{node_id}", **fastapi_kw)
        self.node_id = node_id
        self.team_id = team_id
        self.version = version
        self.vc_enabled = vc_enabled
        self.agentfield_url = (agentfield_url
                               or os.environ.get("AGENTFIELD_URL",
                                                 "http://127.0.0.1:8520"))
        self.base_url = base_url  # resolved at serve() if None
        self.client = AgentFieldClient(self.agentfield_url)
        self.ai = AgentAI(ai_config)
        self.ai._agent = self  # skills-as-tools loop (ai(tools=...))
        self.memory = MemoryInterface(self.client, node_id)
        self.rate_limiter = StatelessRateLimiter()
        self.result_cache = ResultCache()
        self.auto_register = auto_register
        self.heartbeat_interval = heartbeat_interval
        self.identity: dict = {}
        self._reasoners: dict[str, _FunctionMeta] = {}
        self._skills: dict[str, _FunctionMeta] = {}
        self._hb_thread: threading.Thread | None = None
        self._hb_stop = threading.Event()
        self._action_handlers: dict[str, callable] = {}
        self._registered = False
        self._setup_builtin_routes()

    # ------------------------------------------------------------ routes
    def _setup_builtin_routes(self):
        @self.get("/health")
        async def health():
            return {"status": "healthy", "node_id": self.node_id}

        @self.get("/reasoners")
        async def reasoners():
            return {"reasoners": [m.descriptor()
                                  for m in self._reasoners.values()]}

        @self.get("/skills")
        async def skills():
            return {"skills": [m.descriptor() for m in self._skills.values()]}

        @self.get("/node-info")
        async def node_info():
            return self.node_descriptor()

        @self.get("/status")
        async def status():
            return {"node_id": self.node_id,
                    "registered": self._registered,
                    "reasoners": list(self._reasoners),
                    "skills": list(self._skills)}

        @self.post("/shutdown")
        async def shutdown():
            self._hb_stop.set()
            return {"status": "shutting down"}

    def node_descriptor(self) -> dict:
        return {
            "id": self.node_id,
            "team_id": self.team_id,
            "base_url": self.base_url,
            "version": self.version,
            "deployment_type": "long_running",
            "reasoners": [m.descriptor() for m in self._reasoners.values()],
            "skills": [m.descriptor() for m in self._skills.values()],
            "metadata": {"sdk": "agentfield_amd", "vc_enabled": self.vc_enabled},
        }

    # --------------------------------------------------------- decorators
    def reasoner(self, name: str | None = None, tags=None, vc: bool = False):
        return self._register_fn("reasoner", name, tags, vc)

    def skill(self, name: str | None = None, tags=None,
              cache_results: bool = False):
        return self._register_fn("skill", name, tags, False,
                                 cache_results=cache_results)

    def _register_fn(self, kind: str, name, tags, vc, cache_results=False):
        def deco(fn):
            fname = name or fn.__name__
            meta = _FunctionMeta(fn, fname, kind, tags, vc)
            meta.cache_results = cache_results
            table = self._reasoners if kind == "reasoner" else self._skills
            table[fname] = meta
            path = f"/{kind}s/{fname}"

            async def endpoint(request: Request):
                try:
                    body = await request.json()
                except Exception:
                    body = {}
                if isinstance(body, dict) and "input" in body and \
                        set(body) <= {"input", "context"}:
                    body = body.get("input") or {}
                ctx = ExecutionContext.from_headers(request.headers)
                missing = [r for r in meta.required if r not in body]
                if missing:
                    return JSONResponse(
                        {"error": f"missing required input: {missing}"},
                        status_code=422)
                kwargs = {p.name: body[p.name] for p in meta.params
                          if p.name in body}
                if ctx.execution_id:
                    # control-plane invocation: 202 + background + callback
                    asyncio.get_running_loop().create_task(
                        self._run_with_callback(meta, kwargs, ctx))
                    return JSONResponse({"status": "accepted",
                                         "execution_id": ctx.execution_id},
                                        status_code=202)
                result = await self._invoke(meta, kwargs, ctx)
                return {"result": result}

            self.add_api_route(path, endpoint, methods=["POST"],
                               name=f"{kind}_{fname}")

            # local call wrapper with workflow tracking (+ optional cache)
            def local(*args, **kwargs):
                bound = self._bind_args(meta, args, kwargs)
                if getattr(meta, "cache_results", False):
                    key = ResultCache.key_for(self.node_id, meta.name, bound)
                    hit = self.result_cache.get(key)
                    if hit is not None:
                        return hit
                    out = self._run_tracked_sync(meta, bound)
                    self.result_cache.put(key, out)
                    return out
                return self._run_tracked_sync(meta, bound)
            local.__name__ = fn.__name__
            local.__wrapped__ = fn
            local.meta = meta
            return local
        return deco

    def include_router(self, router, prefix: str = ""):
        """Register an AgentRouter's reasoners/skills on this agent with the
        prefix dotted into each name (reference: include_router prefix
        rewriting).  Returns {rewritten_name: local_callable}."""
        pre = (prefix or router.prefix).strip("./")
        out = {}
        for item in router._pending:
            name = f"{pre}.{item['name']}" if pre else item["name"]
            if item["kind"] == "reasoner":
                deco = self.reasoner(name=name, tags=item.get("tags"),
                                     vc=item.get("vc", False))
            else:
                deco = self.skill(name=name, tags=item.get("tags"),
                                  cache_results=item.get("cache_results",
                                                         False))
            out[name] = deco(item["fn"])
        return out

    @staticmethod
    def _bind_args(meta: _FunctionMeta, args, kwargs) -> dict:
        out = dict(kwargs)
        for p, a in zip(meta.params, args):
            out[p.name] = a
        return out

    # ---------------------------------------------------------- execution
    def _ensure_executor(self):
        """Widen the loop's default thread pool (asyncio.to_thread default of
        ~12 workers serializes concurrent sync reasoners + callbacks)."""
        if not getattr(self, "_executor_set", False):
            from concurrent.futures import ThreadPoolExecutor
            asyncio.get_running_loop().set_default_executor(
                ThreadPoolExecutor(max_workers=128,
                                   thread_name_prefix="af-worker"))
            self._executor_set = True

    async def _invoke(self, meta: _FunctionMeta, kwargs: dict,
                      ctx: ExecutionContext):
        self._ensure_executor()
        token = set_context(ctx)
        # Control-plane-invoked executions (ctx carries the CP-minted
        # execution id) already have their record created by
        # prepare_execution and their terminal state delivered via the
        # status callback — SDK workflow events would be duplicate rows
        # and double the CP's HTTP load.  Events fire only for in-process
        # nested calls, which are otherwise invisible to the DAG (A.3).
        cp_invoked = bool(ctx.execution_id)
        if not cp_invoked:
            self._workflow_event(meta, ctx, "start", kwargs)
        t0 = time.time()
        try:
            if meta.is_async:
                result = await meta.fn(**kwargs)
            else:
                result = await asyncio.to_thread(meta.fn, **kwargs)
            if not cp_invoked:
                self._workflow_event(meta, ctx, "complete", kwargs,
                                     result=result,
                                     duration_ms=(time.time() - t0) * 1e3)
            return result
        except Exception as e:
            if not cp_invoked:
                self._workflow_event(meta, ctx, "error", kwargs, error=str(e),
                                     duration_ms=(time.time() - t0) * 1e3)
            raise
        finally:
            reset_context(token)

    def _run_tracked_sync(self, meta: _FunctionMeta, kwargs: dict):
        """Direct in-process call (still traced in the workflow DAG, P12)."""
        parent = current_context()
        ctx = (parent.child() if parent else ExecutionContext())
        token = set_context(ctx)
        t0 = time.time()
        self._workflow_event(meta, ctx, "start", kwargs)
        try:
            if meta.is_async:
                result = asyncio.run(meta.fn(**kwargs))
            else:
                result = meta.fn(**kwargs)
            self._workflow_event(meta, ctx, "complete", kwargs, result=result,
                                 duration_ms=(time.time() - t0) * 1e3)
            return result
        except Exception as e:
            self._workflow_event(meta, ctx, "error", kwargs, error=str(e),
                                 duration_ms=(time.time() - t0) * 1e3)
            raise
        finally:
            reset_context(token)

    def _callback_session(self):
        """Shared aiohttp session for status callbacks: posting on the loop
        avoids a thread hop per callback, and aiohttp's C http stack costs
        a fraction of the sync-httpx-in-thread path under load."""
        sess = getattr(self, "_cb_session", None)
        if sess is None or sess.closed:
            import aiohttp
            sess = self._cb_session = aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=30),
                connector=aiohttp.TCPConnector(limit=256))
        return sess

    async def _report_status(self, callback_url: str | None,
                             execution_id: str, status: str, result,
                             err, duration_ms) -> bool:
        base = (callback_url or self.agentfield_url).rstrip("/")
        try:
            async with self._callback_session().post(
                    f"{base}/api/v1/executions/{execution_id}/status",
                    json={"execution_id": execution_id, "status": status,
                          "result": result, "error": err,
                          "duration_ms": duration_ms}) as r:
                return r.status == 200
        except Exception:
            return False

    async def _run_with_callback(self, meta: _FunctionMeta, kwargs: dict,
                                 ctx: ExecutionContext):
        t0 = time.time()
        try:
            result = await self._invoke(meta, kwargs, ctx)
            status, payload, err = "completed", result, None
        except Exception as e:
            status, payload, err = "failed", None, \
                f"{e}\n{traceback.format_exc(limit=3)}"
        duration = (time.time() - t0) * 1e3
        for attempt in range(4):  # a lost callback strands the execution
            ok = await self._report_status(ctx.callback_url,
                                           ctx.execution_id, status, payload,
                                           err, duration)
            if ok:
                break
            await asyncio.sleep(0.2 * (2 ** attempt))
        if self.vc_enabled or meta.vc:
            try:
                await asyncio.to_thread(self.client.create_execution_vc,
                                        ctx.execution_id)
            except Exception:
                pass

    def _workflow_event(self, meta, ctx: ExecutionContext, kind: str,
                        kwargs, result=None, error=None, duration_ms=None):
        """Fire-and-forget nested-call tracing (A.3)."""
        if not ctx.run_id and not ctx.execution_id:
            return
        ev = {
            "execution_id": ctx.execution_id or
            f"local_{self.node_id}_{meta.name}_{int(time.time()*1e6)}",
            "run_id": ctx.run_id,
            "workflow_id": ctx.run_id,
            "reasoner_id": meta.name,
            "agent_node_id": self.node_id,
            "type": meta.kind,
            "parent_execution_id": ctx.parent_execution_id,
            "status": {"start": "running", "complete": "completed",
                       "error": "failed"}[kind],
        }
        if kind == "complete":
            ev["result"] = result if isinstance(result, dict) else \
                {"value": result}
            ev["duration_ms"] = duration_ms
        elif kind == "error":
            ev["error"] = error
            ev["duration_ms"] = duration_ms
        ctx.execution_id = ev["execution_id"]
        self._event_queue().put(ev)

    def _event_queue(self):
        """Single background dispatcher for fire-and-forget workflow events
        (a thread per event churns under load)."""
        q = getattr(self, "_wf_queue", None)
        if q is None:
            import queue as _queue
            q = self._wf_queue = _queue.Queue(maxsize=4096)

            def drain():
                while True:
                    ev = q.get()
                    try:
                        self.client.workflow_event(ev)
                    except Exception:
                        pass

            threading.Thread(target=drain, daemon=True,
                             name="af-wf-events").start()
        return q

    # ------------------------------------------------------- cross-agent
    def lock(self, name: str, ttl_s: float = 30.0,
             timeout_s: float = 60.0) -> DistributedLock:
        """Cross-process lease lock via the control plane:
        `with app.lock("migrate-db"): ...`"""
        return DistributedLock(self.client, name, owner=self.node_id,
                               ttl_s=ttl_s, timeout_s=timeout_s)

    def call(self, target: str, _async: bool = False, _webhook=None, **kwargs):
        """Nested cross-agent call via the control plane (builds the DAG)."""
        ctx = current_context()
        headers = ctx.child_headers() if ctx else {}
        if _async:
            return self.client.execute_async(target, kwargs, headers, _webhook)
        resp = self.rate_limiter.call(
            self.client.execute_sync, target, kwargs, headers, _webhook,
            retries=2)
        if resp.get("status") == "completed":
            result = resp.get("result")
            if isinstance(result, dict) and set(result) == {"result"}:
                return result["result"]
            return result
        raise RuntimeError(
            f"call {target} {resp.get('status')}: {resp.get('error_message')}")

    # ---------------------------------------------------------------- MCP
    def use_mcp(self, project_dir: str = ".", config: dict | None = None):
        """Start the project's MCP servers and expose their tools as
        auto-generated skills (reference P17/dynamic_skills)."""
        from ..mcp import MCPManager
        if not hasattr(self, "mcp"):
            self.mcp = MCPManager()
        if config:
            for name, spec in config.items():
                self.mcp.start_server(name, spec)
        else:
            self.mcp.start_all(project_dir)
        return self.mcp.register_as_skills(self)

    # ------------------------------------------------- lifecycle / serve
    def register(self) -> bool:
        try:
            resp = self.client.register_agent(self.node_descriptor())
            self.identity = resp.get("identity", {})
            self._registered = True
            return True
        except Exception:
            self._registered = False
            return False

    def _heartbeat_loop(self):
        while not self._hb_stop.wait(self.heartbeat_interval):
            payload = {"status": "active"}
            mcp = getattr(self, "mcp", None)
            if mcp is not None:
                # enhanced heartbeat: per-server MCP health rides along
                # (reference P10 agent_field_handler.py:227-264)
                payload["mcp_servers"] = mcp.status()
            eng = self._engine_heartbeat()
            if eng:
                payload["engine"] = eng
            ok = self.client.heartbeat(self.node_id, payload)
            if not ok:
                self.register()  # resilient re-register (P10)
            else:
                self._drain_actions()

    def _engine_heartbeat(self) -> dict | None:
        """Snapshot of the in-process engine (if this agent hosts one)
        for the control plane's agentfield_engine_* metrics: token
        counters as DELTAS since the previous heartbeat."""
        from .ai import _runners
        eng = None
        for r in _runners.values():
            eng = getattr(r, "engine", None)
            if eng is not None:
                break
        if eng is None:
            return None
        m = eng.metrics
        prev = getattr(self, "_hb_engine_prev", {})
        out = {
            "running": eng.sched.num_running(),
            "kv_free_pages": eng.sched.alloc.num_free,
            "prefill_tokens_delta":
                m["prefill_tokens"] - prev.get("prefill_tokens", 0),
            "decode_tokens_delta":
                m["decode_tokens"] - prev.get("decode_tokens", 0),
        }
        self._hb_engine_prev = {"prefill_tokens": m["prefill_tokens"],
                                "decode_tokens": m["decode_tokens"]}
        return out

    def _drain_actions(self):
        """Claim and execute pending control-plane lifecycle actions
        (reference C7 claim/ack lease protocol).  Stop/shutdown actions
        end the heartbeat loop; custom actions go to @on_action handlers."""
        try:
            acts = self.client.claim_actions(self.node_id)
        except Exception:
            return
        for a in acts:
            status = "done"
            try:
                if a["action"] in ("stop", "shutdown"):
                    self._hb_stop.set()
                elif a["action"] in self._action_handlers:
                    self._action_handlers[a["action"]](a.get("payload", {}))
                else:
                    status = "ignored"
            except Exception as e:
                status = f"error: {e}"
            try:
                self.client.ack_action(self.node_id, a["id"], status)
            except Exception:
                pass

    def on_action(self, name: str):
        """Decorator: handle a named control-plane action delivered via the
        claim/ack lease queue."""
        def deco(fn):
            self._action_handlers[name] = fn
            return fn
        return deco

    def start_background(self):
        if self.auto_register:
            self.register()
        if self._hb_thread is None:
            self._hb_thread = threading.Thread(target=self._heartbeat_loop,
                                               daemon=True, name="af-heartbeat")
            self._hb_thread.start()

    def _deferred_start(self):
        """Register once our own HTTP endpoint is up (runs in a thread;
        newer starlette removed add_event_handler)."""
        import httpx
        for _ in range(200):
            try:
                httpx.get(self.base_url + "/health", timeout=0.5)
                break
            except httpx.HTTPError:
                time.sleep(0.1)
        self.start_background()

    def serve(self, host: str = "127.0.0.1", port: int = 8600, **uvicorn_kw):
        import uvicorn
        if self.base_url is None:
            self.base_url = f"http://{host}:{port}"
        threading.Thread(target=self._deferred_start, daemon=True).start()
        uvicorn_kw.setdefault("access_log", False)
        uvicorn.run(self, host=host, port=port, log_level="warning",
                    **uvicorn_kw)

    run = serve
