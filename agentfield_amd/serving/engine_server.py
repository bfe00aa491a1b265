"""Standalone engine server: one process per GPU serving the local engine
over HTTP.  The DP topology for config 4 (8 replicas on one node) is eight
of these behind serving.DPRouter.

  python -m agentfield_amd.serving.engine_server --model llama-3-8b \
      --device cuda:0 --port 8710
"""
from __future__ import annotations

import argparse
import json
import queue

import torch
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..engine import LLMEngine, SamplingParams
from ..models import CONFIGS
from ..sdk.ai import ByteTokenizer, EngineRunner, load_tokenizer


def _json_ready(runner: EngineRunner) -> bool:
    """Grammar-constrained JSON: byte tokenizers mask bytes; HF
    tokenizers get a token-level grammar compiled on first use."""
    try:
        runner.ensure_json_support()
        return True
    except Exception:
        return False


def create_engine_app(runner: EngineRunner, model_name: str,
                      extra_models: dict[str, EngineRunner] | None = None
                      ) -> FastAPI:
    """One HTTP server over one or more in-process engines.  288 GB of
    HBM3E comfortably co-resides several model families (8B + 70B fit
    together), so a replica can serve multiple models; requests route by
    their `model` field, defaulting to the primary."""
    runners: dict[str, EngineRunner] = {model_name: runner,
                                        **(extra_models or {})}
    app = FastAPI(title=f"agentfield-engine:{model_name}")
    app.state.runner = runner

    @app.on_event("startup")
    async def _widen_thread_pool():
        # every in-flight /v1/generate parks a blocking wait on a worker
        # thread; anyio's default 40-token limiter would serialize large
        # decode batches (measured: 128 concurrent calls ran in waves)
        import anyio.to_thread
        anyio.to_thread.current_default_thread_limiter().total_tokens = 1024

    def pick(body: dict) -> tuple[EngineRunner, str]:
        m = body.get("model")
        if m and m in runners:
            return runners[m], m
        return runners[model_name], model_name

    def clamp_prompt(rnr: EngineRunner, ids: list[int],
                     max_tokens: int) -> list[int]:
        """Token-aware head trim so prompt+max_tokens always fits the
        context window (mirrors the reference's client-side trimming,
        agent_ai.py:267) — an oversized prompt must degrade, not 500."""
        limit = rnr.engine.cfg.max_position - 8
        if len(ids) + max_tokens > limit:
            keep = max(1, limit - max_tokens)
            ids = ids[-keep:]
        return ids

    @app.get("/v1/health")
    async def health():
        return {"status": "healthy", "model": model_name,
                "models": list(runners)}

    @app.get("/metrics")
    async def metrics():
        from fastapi.responses import Response
        from prometheus_client import (CollectorRegistry, Gauge,
                                       generate_latest)
        eng = runner.engine
        reg = CollectorRegistry()
        vals = {
            "agentfield_engine_queued": eng.sched.num_queued(),
            "agentfield_engine_running": eng.sched.num_running(),
            "agentfield_engine_kv_free_pages": eng.sched.alloc.num_free,
            "agentfield_engine_kv_total_pages": eng.sched.alloc.num_pages,
            "agentfield_engine_prefill_tokens_total": eng.metrics["prefill_tokens"],
            "agentfield_engine_decode_tokens_total": eng.metrics["decode_tokens"],
            "agentfield_engine_steps_total": eng.metrics["steps"],
        }
        for name, v in vals.items():
            Gauge(name, name, registry=reg).set(v)
        return Response(generate_latest(reg), media_type="text/plain")

    @app.get("/v1/stats")
    async def stats():
        eng = runner.engine
        out = {
            "model": model_name,
            "queued": eng.sched.num_queued(),
            "running": eng.sched.num_running(),
            "kv_free_pages": eng.sched.alloc.num_free,
            "kv_total_pages": eng.sched.alloc.num_pages,
            **eng.metrics,
        }
        ct = getattr(eng.sched, "cached_tokens", None)
        if ct is not None:  # prefix-cache-capable scheduler
            out["prefix_cached_tokens"] = ct
            out["prefix_cache_pages"] = getattr(eng.sched, "cache_pages", 0)
        if len(runners) > 1:
            out["models"] = {
                name: {"queued": r.engine.sched.num_queued(),
                       "running": r.engine.sched.num_running(),
                       **r.engine.metrics}
                for name, r in runners.items()}
        return out

    @app.post("/v1/generate")
    async def generate(req: Request):
        body = await req.json()
        runner, _mname = pick(body)
        schema = body.get("json_schema")
        jm = bool(body.get("json_mode", False)) or schema is not None
        if jm and not _json_ready(runner):
            return JSONResponse(
                {"error": "json_mode unsupported for this tokenizer"},
                status_code=400)
        if schema is not None and not isinstance(schema, dict):
            return JSONResponse(
                {"error": "json_schema must be an object"}, status_code=400)
        sp = SamplingParams(
            max_tokens=int(body.get("max_tokens", 128)),
            temperature=float(body.get("temperature", 0.0)),
            top_k=int(body.get("top_k", 0) or 0),
            top_p=float(body.get("top_p", 1.0) or 1.0),
            ignore_eos=bool(body.get("ignore_eos", False)),
            logprobs=min(int(body.get("logprobs", 0) or 0), 8),
            json_mode=jm, json_schema=schema)
        if "prompt_ids" in body:
            ids = [int(x) for x in body["prompt_ids"]]
        else:
            ids = runner.tokenizer.encode(body.get("prompt", ""))
        ids = clamp_prompt(runner, ids, sp.max_tokens)
        stops = tuple(s for s in body.get("stop", []) if s)
        stream = bool(body.get("stream", False))
        if not stream:
            import anyio
            w = runner.submit(ids, sp)

            def wait():
                w["done"].wait(float(body.get("timeout", 600)))
                return w
            w = await anyio.to_thread.run_sync(wait)
            if not w["done"].is_set():
                runner.cancel(w)  # free KV pages + batch slot
                return JSONResponse({"error": "timeout"}, status_code=504)
            if w["error"]:
                return JSONResponse({"error": w["error"]}, status_code=503)
            text = runner.tokenizer.decode(w["output"])
            for s in stops:
                idx = text.find(s)
                if idx >= 0:
                    text = text[:idx]
            out = {"output_ids": w["output"], "text": text}
            if w.get("logprobs") is not None:
                out["logprobs"] = [
                    {"logprob": e["logprob"],
                     "top": [[int(t), float(v)] for t, v in e["top"]]}
                    for e in w["logprobs"]]
            return out

        sq: queue.Queue = queue.Queue()
        w = runner.submit(ids, sp, stream_q=sq)

        async def gen():
            import anyio
            if not stops:
                while True:
                    tok, done = await anyio.to_thread.run_sync(sq.get)
                    piece = (runner.tokenizer.decode([tok])
                             if tok is not None else "")
                    yield f"data: {json.dumps({'token': tok, 'text': piece, 'done': done})}\n\n"
                    if done:
                        return
            # stop-sequence mode: holdback window so a stop split across
            # token pieces never reaches the client (events coalesce text;
            # token ids are omitted)
            hold = max(len(s) for s in stops) - 1
            acc = ""
            while True:
                tok, done = await anyio.to_thread.run_sync(sq.get)
                if tok is not None:
                    acc += runner.tokenizer.decode([tok])
                cuts = [i for i in (acc.find(s) for s in stops) if i >= 0]
                if cuts:
                    head = acc[:min(cuts)]
                    yield f"data: {json.dumps({'token': None, 'text': head, 'done': True})}\n\n"
                    runner.cancel(w)
                    return
                if done:
                    yield f"data: {json.dumps({'token': None, 'text': acc, 'done': True})}\n\n"
                    return
                if len(acc) > hold:
                    emit, acc = acc[:len(acc) - hold], acc[len(acc) - hold:]
                    if emit:
                        yield f"data: {json.dumps({'token': None, 'text': emit, 'done': False})}\n\n"
        return StreamingResponse(gen(), media_type="text/event-stream")

    # ------------------------------------------------- OpenAI-compatible
    # /v1/chat/completions and /v1/completions speak the OpenAI wire
    # protocol (including SSE chunk streaming + [DONE]), so any OpenAI
    # client library can point at an engine replica directly — the serving
    # analog of the reference's LiteLLM provider seam.
    def _assemble_chat_prompt(messages, mname: str = None,
                              rnr: EngineRunner = None) -> str:
        from ..sdk.multimodal import require_text
        flat = []
        for m in messages:
            flat.append({"role": m.get("role", "user"),
                         "content": require_text(m.get("content", ""),
                                                 mname or model_name)})
        tok = (rnr or runner).tokenizer
        if hasattr(tok, "apply_chat_template"):
            # HF tokenizer: the model's own chat template (Llama-3 header
            # tokens etc.) instead of the generic fallback format
            return tok.apply_chat_template(flat)
        parts = [f"<|{m['role']}|>\n{m['content']}" for m in flat]
        parts.append("<|assistant|>\n")
        return "\n".join(parts)

    def _finish_reason(n_out: int, max_tokens: int, stopped: bool) -> str:
        if stopped:
            return "stop"
        return "length" if n_out >= max_tokens else "stop"

    async def _oai_generate(body: dict, prompt: str, kind: str):
        import time as _time
        import uuid

        import anyio
        runner, _mname = pick(body)
        max_tokens = int(body.get("max_tokens", 128))
        # chat: logprobs=true + top_logprobs=N; completions: logprobs=N
        if kind == "chat":
            lp_n = (min(int(body.get("top_logprobs", 1) or 1), 8)
                    if body.get("logprobs") else 0)
        else:
            lp_n = min(int(body.get("logprobs", 0) or 0), 8)
        rf = body.get("response_format") or {}
        schema = None
        if rf.get("type") == "json_schema":
            # OpenAI structured outputs: {"type": "json_schema",
            #   "json_schema": {"name": ..., "schema": {...}, "strict": ...}}
            js = rf.get("json_schema") or {}
            schema = js.get("schema") if isinstance(js, dict) else None
            if not isinstance(schema, dict):
                return JSONResponse(
                    {"error": {"message": "response_format.json_schema"
                               ".schema must be an object",
                               "type": "invalid_request_error"}},
                    status_code=400)
        # OpenAI tool calling: with tool_choice "required" or a named
        # function, the call is GUARANTEED valid by schema-constrained
        # decoding (anyOf over the tool schemas).  "auto"/"none" generate
        # unconstrained (random-init models have no trained tool tokens;
        # real checkpoints would use their chat template's tool markers).
        tools = body.get("tools") or []
        tool_choice = body.get("tool_choice", "auto" if tools else "none")
        tool_mode = False
        if kind == "chat" and tools and tool_choice not in ("none", "auto"):
            chosen = tools
            if isinstance(tool_choice, dict):
                want = tool_choice.get("function", {}).get("name")
                chosen = [t for t in tools
                          if t.get("function", {}).get("name") == want]
                if not chosen:
                    return JSONResponse(
                        {"error": {"message": f"unknown tool {want!r}",
                                   "type": "invalid_request_error"}},
                        status_code=400)
            alts = []
            for t in chosen:
                fn = t.get("function", {})
                alts.append({
                    "type": "object",
                    "properties": {
                        "name": {"enum": [fn.get("name", "fn")]},
                        "arguments": fn.get("parameters") or {},
                    },
                    "required": ["name", "arguments"]})
            schema = alts[0] if len(alts) == 1 else {"anyOf": alts}
            tool_mode = True
            # surface the tool signatures to the model (a real checkpoint
            # would get these through its chat template's tool section)
            prompt = ("Available tools:\n" +
                      json.dumps([t.get("function", {}) for t in chosen]) +
                      "\nRespond with a single JSON tool call.\n" + prompt)
        jm = rf.get("type") in ("json_object", "json_schema") or tool_mode
        if jm and not _json_ready(runner):
            return JSONResponse(
                {"error": {"message": f"response_format {rf.get('type')} "
                           "unsupported for this tokenizer",
                           "type": "invalid_request_error"}},
                status_code=400)
        sp = SamplingParams(max_tokens=max_tokens,
                            temperature=float(body.get("temperature", 0.0)),
                            top_p=float(body.get("top_p", 1.0) or 1.0),
                            top_k=int(body.get("top_k", 0) or 0),
                            logprobs=lp_n,
                            json_mode=jm, json_schema=schema)
        stop_in = body.get("stop") or []
        stops = tuple(s for s in ([stop_in] if isinstance(stop_in, str)
                                  else stop_in) if s)
        ids = clamp_prompt(runner, runner.tokenizer.encode(prompt),
                           max_tokens)
        rid = f"{'chatcmpl' if kind == 'chat' else 'cmpl'}-{uuid.uuid4().hex[:24]}"
        created = int(_time.time())
        base = {"id": rid, "created": created, "model": _mname}

        if not body.get("stream", False):
            nchoice = max(1, min(int(body.get("n", 1)), 16))
            waiters = [runner.submit(ids, sp) for _ in range(nchoice)]

            def wait():
                deadline = float(body.get("timeout", 600))
                for w in waiters:
                    w["done"].wait(deadline)
                return waiters
            await anyio.to_thread.run_sync(wait)
            if not all(w["done"].is_set() for w in waiters):
                for w in waiters:
                    if not w["done"].is_set():
                        runner.cancel(w)  # free KV pages + batch slot
                return JSONResponse({"error": {"message": "timeout",
                                               "type": "timeout"}},
                                    status_code=504)
            err = next((w["error"] for w in waiters if w["error"]), None)
            if err:
                return JSONResponse({"error": {"message": err,
                                               "type": "overloaded"}},
                                    status_code=503)
            choices, out_tokens = [], 0
            for i, w in enumerate(waiters):
                text = runner.tokenizer.decode(w["output"])
                stopped = False
                for s in stops:
                    idx = text.find(s)
                    if idx >= 0:
                        text, stopped = text[:idx], True
                fr = _finish_reason(len(w["output"]), max_tokens, stopped)
                out_tokens += len(w["output"])
                if kind == "chat" and tool_mode:
                    import uuid as _uuid
                    try:
                        call = json.loads(text.strip())
                        args = call.get("arguments", {})
                        choice = {"index": i, "finish_reason": "tool_calls",
                                  "message": {"role": "assistant",
                                              "content": None,
                                              "tool_calls": [{
                                                  "id": "call_" +
                                                  _uuid.uuid4().hex[:24],
                                                  "type": "function",
                                                  "function": {
                                                      "name": call.get("name"),
                                                      "arguments":
                                                      json.dumps(args)}}]}}
                    except Exception:
                        # truncated by max_tokens before the close — the
                        # grammar guarantees a prefix of valid JSON only
                        choice = {"index": i, "finish_reason": "length",
                                  "message": {"role": "assistant",
                                              "content": text}}
                elif kind == "chat":
                    choice = {"index": i, "finish_reason": fr,
                              "message": {"role": "assistant",
                                          "content": text}}
                else:
                    choice = {"index": i, "finish_reason": fr, "text": text}
                if lp_n and w.get("logprobs") is not None:
                    dec = runner.tokenizer.decode
                    if kind == "chat":
                        choice["logprobs"] = {"content": [
                            {"token": dec([tok]), "logprob": e["logprob"],
                             "top_logprobs": [
                                 {"token": dec([int(t)]),
                                  "logprob": float(v)}
                                 for t, v in e["top"]]}
                            for tok, e in zip(w["output"], w["logprobs"])]}
                    else:
                        choice["logprobs"] = {
                            "tokens": [dec([t]) for t in w["output"]],
                            "token_logprobs": [e["logprob"]
                                               for e in w["logprobs"]],
                            "top_logprobs": [
                                {dec([int(t)]): float(v)
                                 for t, v in e["top"]}
                                for e in w["logprobs"]]}
                choices.append(choice)
            usage = {"prompt_tokens": len(ids),
                     "completion_tokens": out_tokens,
                     "total_tokens": len(ids) + out_tokens}
            obj_name = "chat.completion" if kind == "chat" else "text_completion"
            return {**base, "object": obj_name, "choices": choices,
                    "usage": usage}

        sq: queue.Queue = queue.Queue()
        w = runner.submit(ids, sp, stream_q=sq)
        obj = "chat.completion.chunk" if kind == "chat" else "text_completion"

        def chunk(piece: str | None, fr: str | None) -> str:
            if kind == "chat":
                delta = {"content": piece} if piece is not None else {}
                if piece is None and fr is None:
                    delta = {"role": "assistant"}
                c = {"index": 0, "delta": delta, "finish_reason": fr}
            else:
                c = {"index": 0, "text": piece or "", "finish_reason": fr}
            return "data: " + json.dumps({**base, "object": obj,
                                          "choices": [c]}) + "\n\n"

        async def gen():
            n_out = 0
            if kind == "chat" and tool_mode:
                # tool-call streaming: the call JSON is buffered (its
                # name field only exists once generated) and emitted as
                # one OpenAI-shaped tool_calls delta
                import uuid as _uuid
                buf = []
                while True:
                    tok, done = await anyio.to_thread.run_sync(sq.get)
                    if tok is not None:
                        buf.append(tok)
                    if done:
                        break
                text = runner.tokenizer.decode(buf)
                try:
                    call = json.loads(text.strip())
                    delta = {"role": "assistant", "tool_calls": [{
                        "index": 0,
                        "id": "call_" + _uuid.uuid4().hex[:24],
                        "type": "function",
                        "function": {
                            "name": call.get("name"),
                            "arguments": json.dumps(
                                call.get("arguments", {}))}}]}
                    fr = "tool_calls"
                except Exception:  # truncated mid-call
                    delta = {"role": "assistant", "content": text}
                    fr = "length"
                c = {"index": 0, "delta": delta, "finish_reason": None}
                yield "data: " + json.dumps({**base, "object": obj,
                                             "choices": [c]}) + "\n\n"
                yield chunk(None, fr)
                yield "data: [DONE]\n\n"
                return
            if kind == "chat":
                yield chunk(None, None)  # leading role delta
            hold = max((len(s) for s in stops), default=1) - 1
            acc = ""
            stopped = False
            while True:
                tok, done = await anyio.to_thread.run_sync(sq.get)
                if tok is not None:
                    n_out += 1
                    acc += runner.tokenizer.decode([tok])
                cuts = [i for i in (acc.find(s) for s in stops) if i >= 0]
                if cuts:
                    head = acc[:min(cuts)]
                    if head:
                        yield chunk(head, None)
                    runner.cancel(w)
                    stopped, done = True, True
                elif done:
                    if acc:
                        yield chunk(acc, None)
                elif len(acc) > hold:
                    emit, acc = acc[:len(acc) - hold], acc[len(acc) - hold:]
                    if emit:
                        yield chunk(emit, None)
                if done:
                    yield chunk(None, _finish_reason(n_out, max_tokens,
                                                     stopped))
                    if (body.get("stream_options") or {}).get(
                            "include_usage"):
                        yield "data: " + json.dumps({
                            **base, "object": obj, "choices": [],
                            "usage": {"prompt_tokens": len(ids),
                                      "completion_tokens": n_out,
                                      "total_tokens": len(ids) + n_out},
                        }) + "\n\n"
                    yield "data: [DONE]\n\n"
                    return
        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.post("/v1/chat/completions")
    async def chat_completions(req: Request):
        body = await req.json()
        rnr, mname = pick(body)
        prompt = _assemble_chat_prompt(body.get("messages", []), mname, rnr)
        return await _oai_generate(body, prompt, "chat")

    @app.post("/v1/completions")
    async def completions(req: Request):
        body = await req.json()
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = "".join(str(p) for p in prompt)
        return await _oai_generate(body, prompt, "text")

    @app.get("/v1/models")
    async def models():
        import time as _time
        return {"object": "list",
                "data": [{"id": name, "object": "model",
                          "created": int(_time.time()),
                          "owned_by": "agentfield_amd"}
                         for name in runners]}

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", action="append", default=None,
                    help="model to serve (repeatable: the first is the "
                         "default, extras co-reside in HBM and route by "
                         "the request's model field)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8710)
    ap.add_argument("--max-num-seqs", type=int, default=256)
    ap.add_argument("--max-prefill-tokens", type=int, default=None,
                    help="prefill admission budget per step (engine default "
                         "when unset)")
    ap.add_argument("--no-graphs", action="store_true")
    ap.add_argument("--spec-lookup", type=int, default=0,
                    help="prompt-lookup speculative decoding draft length "
                         "(greedy-exact; 0 disables)")
    ap.add_argument("--spec-draft", default=None,
                    help="draft MODEL for speculative decoding (a config "
                         "name, e.g. debug-1b for a llama-3-8b target; "
                         "greedy-exact verify)")
    ap.add_argument("--spec-draft-k", type=int, default=4)
    ap.add_argument("--prefix-cache", action="store_true",
                    help="share paged KV across requests with a common "
                         "prompt prefix (refcounted pages, LRU eviction)")
    args = ap.parse_args()

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device.startswith("cuda:"):
        torch.cuda.set_device(device)
    models = args.model or ["llama-3-8b"]

    def build(name: str) -> EngineRunner:
        kw = {}
        if not device.startswith("cuda"):
            kw = {"num_pages": 512, "max_num_seqs": 8,
                  "dtype": torch.float32}
        tokenizer = load_tokenizer()
        if tokenizer.eos_id is not None:
            kw["eos_id"] = tokenizer.eos_id  # HF vocab EOS != byte EOS
        if args.max_prefill_tokens:
            kw["max_prefill_tokens"] = args.max_prefill_tokens
        eng = LLMEngine(
            CONFIGS[name], device=device,
            max_num_seqs=kw.pop("max_num_seqs", args.max_num_seqs),
            enable_graphs=not args.no_graphs and device.startswith("cuda"),
            spec_lookup=args.spec_lookup,
            spec_draft=args.spec_draft, spec_draft_k=args.spec_draft_k,
            prefix_cache=args.prefix_cache, **kw)
        return EngineRunner(eng, tokenizer)

    runners = {name: build(name) for name in models}
    primary = models[0]
    app = create_engine_app(runners[primary], primary,
                            extra_models={n: r for n, r in runners.items()
                                          if n != primary})
    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
