"""app.ai() — the model seam, served by the in-process MI355X engine.

Where the reference forwards app.ai() to external providers through LiteLLM
(everything below get_litellm_params is replaced), here the
call goes: tokenize -> continuous-batching engine (hipGraph decode loop on
the local GPU) -> detokenize.  The public signature and the hierarchical
config merge (agent defaults < method < runtime) match the reference.
"""
from __future__ import annotations

import json
import os
import queue
import threading
from dataclasses import dataclass, field, replace

from ..engine import LLMEngine, SamplingParams
from ..models import CONFIGS


@dataclass
class AIConfig:
    model: str = "llama-3-8b"
    temperature: float = 0.0
    max_tokens: int = 256
    system_prompt: str | None = None
    stop: tuple = ()
    timeout: float = 600.0
    device: str | None = None           # None -> cuda if available
    fallback_models: tuple = ()
    json_only: bool = False             # grammar-constrained valid JSON
    json_schema: dict | None = None     # schema-constrained (keys/types)
    ignore_eos: bool = False            # benchmark mode: always decode
                                        # max_tokens (random-init weights
                                        # emit EOS at chance rate)
    extra: dict = field(default_factory=dict)

    def merged(self, **overrides) -> "AIConfig":
        clean = {k: v for k, v in overrides.items() if v is not None}
        extra = {**self.extra, **clean.pop("extra", {})}
        known = {k: v for k, v in clean.items() if hasattr(self, k)}
        return replace(self, extra=extra, **known)


class ByteTokenizer:
    """Offline-safe default tokenizer: UTF-8 bytes + specials.  Real
    deployments point AGENTFIELD_TOKENIZER at a HF tokenizer directory."""
    BOS, EOS, PAD = 1, 2, 3
    OFFSET = 4

    def __init__(self, vocab_size: int = 128256):
        self.vocab_size = vocab_size

    @property
    def eos_id(self) -> int:
        return self.EOS

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        ids = [b + self.OFFSET for b in text.encode("utf-8")]
        return ([self.BOS] if add_bos else []) + ids

    def decode(self, ids: list[int]) -> str:
        data = bytes(i - self.OFFSET for i in ids
                     if self.OFFSET <= i < self.OFFSET + 256)
        return data.decode("utf-8", errors="replace")


def load_tokenizer(path: str | None = None):
    path = path or os.environ.get("AGENTFIELD_TOKENIZER")
    if path:
        from transformers import AutoTokenizer

        class _HF:
            def __init__(self, tok):
                self.tok = tok
                self.eos_id = tok.eos_token_id

            def encode(self, text, add_bos=True):
                return self.tok.encode(text)

            def decode(self, ids):
                return self.tok.decode(ids, skip_special_tokens=True)

            def apply_chat_template(self, messages):
                """Model-native chat formatting (e.g. Llama-3 header
                tokens) for the OpenAI chat endpoint."""
                return self.tok.apply_chat_template(
                    messages, tokenize=False, add_generation_prompt=True)
        return _HF(AutoTokenizer.from_pretrained(path))
    return ByteTokenizer()


class EngineRunner:
    """Single driver thread stepping one engine; ai() calls submit prompts
    and block on a per-request event.  This is the in-process counterpart of
    the control plane's async queue feeding the scheduler."""

    def __init__(self, engine: LLMEngine, tokenizer=None):
        self.engine = engine
        self.tokenizer = tokenizer or ByteTokenizer(engine.cfg.vocab_size)
        self._submit: queue.Queue = queue.Queue()
        self._cancel: queue.Queue = queue.Queue()
        self._waiters: dict[int, dict] = {}
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._stop = False
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="af-engine")
        self._thread.start()

    def submit(self, prompt_ids: list[int], sampling: SamplingParams,
               stream_q: queue.Queue | None = None) -> dict:
        waiter = {"done": threading.Event(), "output": None, "error": None,
                  "stream": stream_q}
        self._submit.put((prompt_ids, sampling, waiter))
        self._wake.set()
        return waiter

    def ensure_json_support(self) -> None:
        """Make grammar-constrained JSON decoding available: byte
        tokenizers mask bytes directly; HF/BPE tokenizers get a
        token-level grammar compiled (once) from their vocabulary and
        attached to the engine (engine/token_grammar.py)."""
        if isinstance(self.tokenizer, ByteTokenizer):
            return
        if getattr(self.engine, "token_grammar", None) is not None:
            return
        from ..engine.token_grammar import (TokenJsonGrammar,
                                            vocab_bytes_from_hf)
        tok = getattr(self.tokenizer, "tok", None)
        if tok is None:
            raise ValueError("json_only needs a byte or HF tokenizer")
        vocab = vocab_bytes_from_hf(tok, self.engine.cfg.vocab_size)
        eos = self.tokenizer.eos_id
        eos = self.engine.eos_id if eos is None else eos
        self.engine.set_token_grammar(TokenJsonGrammar(vocab, eos_id=eos))

    def _check_json_mode(self, cfg: AIConfig) -> bool:
        if cfg.json_only or cfg.json_schema is not None:
            self.ensure_json_support()
        return cfg.json_only or cfg.json_schema is not None

    def _compiled_schema(self, cfg: AIConfig):
        if cfg.json_schema is None:
            return None
        if isinstance(cfg.json_schema, dict) and "anyOf" in cfg.json_schema:
            return cfg.json_schema  # engine compiles root anyOf (make_fsm)
        from ..engine.schemafsm import SchemaSpec
        return SchemaSpec(cfg.json_schema)

    def generate_text(self, prompt: str, cfg: AIConfig) -> str:
        ids = self.tokenizer.encode(prompt)
        limit = self.engine.cfg.max_position - 8
        if len(ids) + cfg.max_tokens > limit:
            ids = ids[-(limit - cfg.max_tokens):]  # token-aware trim
        sp = SamplingParams(max_tokens=cfg.max_tokens,
                            temperature=cfg.temperature,
                            ignore_eos=cfg.ignore_eos,
                            json_mode=self._check_json_mode(cfg),
                            json_schema=self._compiled_schema(cfg))
        w = self.submit(ids, sp)
        if not w["done"].wait(cfg.timeout):
            raise TimeoutError("engine generate timed out")
        if w["error"]:
            raise RuntimeError(w["error"])
        text = self.tokenizer.decode(w["output"])
        for s in cfg.stop or ():
            idx = text.find(s)
            if idx >= 0:
                text = text[:idx]
        return text

    def stream_text(self, prompt: str, cfg: AIConfig):
        ids = self.tokenizer.encode(prompt)
        sp = SamplingParams(max_tokens=cfg.max_tokens,
                            temperature=cfg.temperature,
                            json_mode=self._check_json_mode(cfg),
                            json_schema=self._compiled_schema(cfg))
        sq: queue.Queue = queue.Queue()
        w = self.submit(ids, sp, stream_q=sq)
        stops = tuple(s for s in (cfg.stop or ()) if s)
        # holdback window: never emit the last max(len(stop))-1 chars until
        # more text arrives, so a stop string split across token pieces is
        # still caught before any of it reaches the caller
        hold = max((len(s) for s in stops), default=1) - 1
        acc = ""
        try:
            while True:
                tok, done = sq.get(timeout=cfg.timeout)
                if tok is not None:
                    acc += self.tokenizer.decode([tok])
                if stops:
                    cuts = [i for i in (acc.find(s) for s in stops) if i >= 0]
                    if cuts:
                        head = acc[:min(cuts)]
                        if head:
                            yield head
                        self.cancel(w)  # free the engine's remaining decode
                        return
                if done:
                    if acc:
                        yield acc
                    return
                if len(acc) > hold:
                    emit, acc = acc[:len(acc) - hold], acc[len(acc) - hold:]
                    if emit:
                        yield emit
        except GeneratorExit:
            self.cancel(w)  # caller abandoned the stream
            raise

    def cancel(self, waiter: dict) -> None:
        """Cancel a submitted request (engine calls stay on the loop
        thread; the waiter is resolved as cancelled)."""
        self._cancel.put(waiter)
        self._wake.set()

    def _loop(self):
        eng = self.engine
        pending: dict[int, dict] = {}
        while not self._stop:
            moved = False
            while True:
                try:
                    ids, sp, waiter = self._submit.get_nowait()
                except queue.Empty:
                    break
                # Bad requests (prompt+max_tokens over the context window,
                # malformed sampling params) must resolve this waiter, not
                # kill the driver thread and wedge every other caller.
                try:
                    rid = eng.add_request(ids, sp)
                except Exception as e:  # noqa: BLE001 — loop must survive
                    waiter["error"] = f"rejected: {e}"
                    waiter["done"].set()
                    moved = True
                    continue
                if rid is None:
                    waiter["error"] = "engine queue full"
                    waiter["done"].set()
                else:
                    waiter["rid"] = rid
                    pending[rid] = waiter
                moved = True
            while True:
                try:
                    w = self._cancel.get_nowait()
                except queue.Empty:
                    break
                rid = w.get("rid")
                if rid is not None and rid in pending:
                    eng.cancel(rid)
                    pending.pop(rid, None)
                    w["error"] = w["error"] or "cancelled"
                    w["done"].set()
                moved = True
            if eng.has_work():
                try:
                    events = eng.step()
                except Exception as e:  # noqa: BLE001 — resolve, don't hang
                    for w in pending.values():
                        w["error"] = f"engine step failed: {e}"
                        w["done"].set()
                        if w["stream"] is not None:
                            w["stream"].put((None, True))
                    pending.clear()
                    continue
                for (rid, tok, done) in events:
                    w = pending.get(rid)
                    if w and w["stream"] is not None:
                        w["stream"].put((tok, done))
                moved = True
                for rid in list(pending):
                    fin = eng.get_finished(rid)
                    if fin is not None:
                        w = pending.pop(rid)
                        w["output"] = fin.output_ids
                        w["logprobs"] = fin.logprobs
                        w["done"].set()
            if not moved:
                self._wake.wait(0.005)
                self._wake.clear()

    def shutdown(self):
        self._stop = True
        self._wake.set()


_runners: dict[str, EngineRunner] = {}
_runners_lock = threading.Lock()


def get_runner(cfg: AIConfig) -> EngineRunner:
    """Process-wide engine registry: one engine per model name.  When
    AGENTFIELD_ENGINE_URLS is set (comma-separated engine servers), calls
    route to the DP replica fleet instead of an in-process engine."""
    import torch
    key = cfg.model
    with _runners_lock:
        if key in _runners:
            return _runners[key]
        urls = os.environ.get("AGENTFIELD_ENGINE_URLS")
        if urls:
            from ..serving.router import DPRouter, RemoteRunner
            runner = RemoteRunner(DPRouter(urls.split(",")))
            _runners[key] = runner
            return runner
        device = cfg.device or ("cuda" if torch.cuda.is_available() else "cpu")
        model_cfg = CONFIGS[cfg.model]  # KeyError -> fallback chain
        tokenizer = load_tokenizer()
        kw = {}
        if device == "cpu":
            kw = {"num_pages": 512, "max_num_seqs": 8, "enable_graphs": False,
                  "dtype": torch.float32}
        if tokenizer.eos_id is not None:
            # the engine's stop check must use THIS vocabulary's EOS (an
            # HF tokenizer's differs from the byte tokenizer's 2)
            kw["eos_id"] = tokenizer.eos_id
        # agent workloads repeat system prompts on every ai() call: prefix
        # caching turns that repeated prefill into a block-table lookup
        # (opt out with AGENTFIELD_NO_PREFIX_CACHE=1)
        kw["prefix_cache"] = os.environ.get(
            "AGENTFIELD_NO_PREFIX_CACHE") != "1"
        eng = LLMEngine(model_cfg, device=device, **kw)
        runner = EngineRunner(eng, tokenizer)
        _runners[key] = runner
        return runner


def set_runner(model: str, runner: EngineRunner) -> None:
    with _runners_lock:
        _runners[model] = runner


class AgentAI:
    """The ai() callable bound to an Agent (hierarchical config merge)."""

    def __init__(self, default_config: AIConfig | None = None):
        self.config = default_config or AIConfig(
            model=os.environ.get("AGENTFIELD_AI_MODEL", "llama-3-8b"))
        self._agent = None  # set by Agent: enables ai(tools=...)

    def __call__(self, *prompt_parts, system: str | None = None,
                 user: str | None = None, schema=None, stream: bool = False,
                 tools=None, max_tool_rounds: int = 4, **overrides):
        if tools is not None:
            body = (user if user is not None
                    else "\n".join(str(p) for p in prompt_parts))
            return self._tool_loop(body, system, tools, max_tool_rounds,
                                   self.config.merged(**overrides))
        cfg = self.config.merged(**overrides)
        if schema is not None:
            # schema requests get SCHEMA-constrained decoding: the engine
            # GUARANTEES the output parses AND matches the schema subset
            # (keys/types/enums — engine/schemafsm.py), where the
            # reference can only prompt-and-validate
            schema_json = (schema.model_json_schema()
                           if hasattr(schema, "model_json_schema")
                           else schema)
            cfg = cfg.merged(json_only=True, json_schema=schema_json)
        parts = []
        sys_p = system or cfg.system_prompt
        if sys_p:
            parts.append(f"<|system|>\n{sys_p}")
        body = user if user is not None else "\n".join(str(p) for p in prompt_parts)
        if schema is not None:
            schema_json = (schema.model_json_schema()
                           if hasattr(schema, "model_json_schema") else schema)
            parts.append(f"<|system|>\nRespond ONLY with JSON matching this "
                         f"schema:\n{json.dumps(schema_json)}")
        parts.append(f"<|user|>\n{body}\n<|assistant|>\n")
        prompt = "\n".join(parts)
        if stream:
            return get_runner(cfg).stream_text(prompt, cfg)
        text = None
        last_err = None
        for model in (cfg.model,) + tuple(cfg.fallback_models or ()):
            mcfg = cfg if model == cfg.model else cfg.merged(model=model)
            try:
                text = get_runner(mcfg).generate_text(prompt, mcfg)
                break
            except (RuntimeError, TimeoutError, KeyError) as e:
                last_err = e  # engine full / timed out / unknown model
        if text is None:
            raise RuntimeError(
                f"all models failed ({(cfg.model,) + tuple(cfg.fallback_models or ())}): "
                f"{last_err}") from last_err
        if schema is not None:
            try:
                data = json.loads(text)
                if hasattr(schema, "model_validate"):
                    return schema.model_validate(data)
                return data
            except (ValueError, TypeError):
                return text  # schema validation failed; return raw text
        return text

    # ------------------------------------------------- skills-as-tools
    _FINAL = {"type": "object",
              "properties": {"name": {"enum": ["final_answer"]},
                             "arguments": {
                                 "type": "object",
                                 "properties": {"text": {"type": "string"}},
                                 "required": ["text"]}},
              "required": ["name", "arguments"]}

    def _tool_loop(self, body: str, system, tools, max_rounds: int,
                   cfg: "AIConfig") -> str:
        """Agentic tool use over the agent's OWN @skill functions: every
        round the model emits a schema-GUARANTEED tool call (anyOf over
        the skill signatures + a built-in final_answer), the skill runs
        locally, and its result joins the transcript.  The last round is
        constrained to final_answer only, so the loop always terminates
        with an answer.  (The reference routes tool use through external
        providers' function calling; here the constrained decoder makes
        malformed calls impossible.)"""
        import asyncio as _aio

        if self._agent is None:
            raise RuntimeError("ai(tools=...) needs an Agent-bound ai")
        skills = self._agent._skills
        names = list(skills) if tools is True else list(tools)
        missing = [n for n in names if n not in skills]
        if missing:
            raise KeyError(f"unknown skills for tools=: {missing}")

        def call_schema(nm):
            m = skills[nm]
            return {"type": "object",
                    "properties": {"name": {"enum": [nm]},
                                   "arguments": m.input_schema()},
                    "required": ["name", "arguments"]}

        alts = [call_schema(n) for n in names] + [self._FINAL]
        catalog = json.dumps(
            [{"name": n, "parameters": skills[n].input_schema()}
             for n in names] +
            [{"name": "final_answer",
              "parameters": {"text": "the answer"}}])
        transcript: list[str] = []
        for rnd in range(max_rounds):
            last = rnd == max_rounds - 1
            schema = self._FINAL if last else                 (alts[0] if len(alts) == 1 else {"anyOf": alts})
            rcfg = cfg.merged(json_only=True, json_schema=schema)
            parts = []
            if system or cfg.system_prompt:
                parts.append(f"<|system|>\n{system or cfg.system_prompt}")
            parts.append("<|system|>\nTools available:\n" + catalog +
                         "\nRespond with a single JSON tool call; use "
                         "final_answer to finish.")
            parts.append(f"<|user|>\n{body}")
            parts.extend(transcript)
            parts.append("<|assistant|>\n")
            out = get_runner(rcfg).generate_text("\n".join(parts), rcfg)
            try:
                call = json.loads(out)
            except ValueError:  # truncated by max_tokens
                return out
            if call.get("name") == "final_answer":
                return str(call.get("arguments", {}).get("text", ""))
            meta = skills.get(call.get("name"))
            args = call.get("arguments") or {}
            try:
                result = (_aio.run(meta.fn(**args)) if meta.is_async
                          else meta.fn(**args))
            except Exception as e:  # the model sees the failure and adapts
                result = {"error": str(e)}
            transcript.append(
                f"<|assistant|>\n{out}\n<|tool|>\n"
                f"{json.dumps(result, default=str)}")
        raise AssertionError("unreachable: last round forces final_answer")

    def with_multimodal(self, *parts, **kw):
        """ai() over mixed text/image/audio inputs.  Detection and message
        assembly are model-independent (sdk/multimodal.py); a text-only
        model raises UnsupportedModality with the offending part types."""
        from .multimodal import build_content, require_text
        cfg = self.config.merged(**{k: v for k, v in kw.items()
                                    if k in AIConfig.__dataclass_fields__})
        content = build_content(parts)
        text = require_text(content, cfg.model)  # text-only model families
        kw.pop("user", None)
        return self(user=text, **kw)

    def with_vision(self, prompt, *images, **kw):
        return self.with_multimodal(prompt, *images, **kw)

    def with_audio(self, prompt, *clips, **kw):
        return self.with_multimodal(prompt, *clips, **kw)
