"""The `af` CLI (reference parity: C32 — cobra `af` with server/init/run/
list/vc/... subcommands, reimplemented with typer).

  af server            run the control plane
  af engine            run a GPU engine server (one per GPU)
  af init NAME         scaffold an agent project
  af run FILE          run an agent app (imports FILE, serves the Agent)
  af list              list registered nodes
  af executions        recent executions
  af vc verify FILE    offline verifiable-credential verification
  af status            control-plane health
"""
from __future__ import annotations

import importlib.util
import json
import os
import sys
from pathlib import Path

import typer

def _die_with_parent():
    """preexec_fn: the child receives SIGTERM if its parent dies — no
    orphaned control-plane workers / dev children when the launcher is
    killed uncleanly (measured: an orphaned worker squatting on the
    bench's derived ports 404'd every later run)."""
    try:
        import ctypes
        libc = ctypes.CDLL("libc.so.6", use_errno=True)
        libc.prctl(1, 15)  # PR_SET_PDEATHSIG = 1, SIGTERM = 15
    except Exception:
        pass


app = typer.Typer(add_completion=False, no_args_is_help=True)

DEFAULT_URL = os.environ.get("AGENTFIELD_URL", "http://127.0.0.1:8520")


@app.command()
def server(host: str = "0.0.0.0", port: int = 8520,
           db: str = typer.Option("agentfield.db", help="SQLite path"),
           data_dir: str = typer.Option(".agentfield", help="payload/keys dir"),
           config: str = typer.Option(None, help="YAML config file"),
           workers: int = typer.Option(1, help="control-plane worker "
                                       "processes (ports port..port+N-1 "
                                       "over one shared WAL database; "
                                       "status callbacks stick to the "
                                       "dispatching worker)"),
           no_did: bool = False):
    """Run the control plane."""
    if workers > 1:
        # multi-worker plane (reference parity: the Go plane's NumCPU
        # worker pool, execute.go:1373 — here realized as N processes
        # because Python serializes a single process on the GIL).
        # Worker i listens on port+i; worker 0 runs the background
        # services (webhook poller, presence, cleanup); all workers
        # share the SQLite WAL file.  Agents may register at any worker;
        # X-AgentField-Callback pins status callbacks to the worker
        # holding the sync waiter.
        import subprocess
        cb_host = "127.0.0.1" if host == "0.0.0.0" else host
        urls = ",".join(f"http://{cb_host}:{port + i}"
                        for i in range(workers))
        procs = []
        for i in range(workers):
            env = {**os.environ,
                   "AGENTFIELD_PUBLIC_URL": f"http://{cb_host}:{port + i}",
                   "AGENTFIELD_WORKER_URLS": urls,
                   "AGENTFIELD_ADMIN_GRPC_PORT": str(port + workers + 100 + i)}
            if i > 0:
                env["AGENTFIELD_BACKGROUND_SERVICES"] = "0"
            cmd = [sys.executable, "-m", "agentfield_amd", "server",
                   "--host", host, "--port", str(port + i), "--db", db,
                   "--data-dir", data_dir, "--workers", "1"]
            if config:
                cmd += ["--config", config]
            if no_did:
                cmd += ["--no-did"]
            procs.append(subprocess.Popen(cmd, env=env,
                                          preexec_fn=_die_with_parent))
        typer.echo(f"agentfield-amd control plane x{workers} on "
                   f"{host}:{port}-{port + workers - 1} (db={db})")
        import signal as _signal

        def _stop(*_a):
            raise SystemExit(0)

        _signal.signal(_signal.SIGTERM, _stop)
        try:
            for p in procs:
                p.wait()
        finally:
            # the parent dying must take the worker fleet with it
            # (bench/stress terminate() the parent)
            for p in procs:
                if p.poll() is None:
                    p.terminate()
            for p in procs:
                try:
                    p.wait(timeout=5)
                except Exception:
                    p.kill()
        return
    import uvicorn
    from .controlplane import ControlPlane, create_app
    from .controlplane.server import Config
    kw = {}
    if config:
        import yaml
        kw = yaml.safe_load(Path(config).read_text()) or {}
    data = Path(kw.pop("data_dir", data_dir))
    data.mkdir(parents=True, exist_ok=True)
    cfg = Config(db_path=kw.pop("db_path", db),
                 payload_dir=str(data / "payloads"),
                 keystore_path=str(data / "keystore.key"),
                 did_enabled=not no_did,
                 admin_grpc_port=int(os.environ.get(
                     "AGENTFIELD_ADMIN_GRPC_PORT", port + 100)), **kw)
    cp = ControlPlane(cfg)
    typer.echo(f"agentfield-amd control plane on {host}:{port} (db={cfg.db_path})")
    uvicorn.run(create_app(cp), host=host, port=port, log_level="warning",
                access_log=False)


@app.command()
def engine(model: list[str] = ["llama-3-8b"], device: str = None,
           host: str = "127.0.0.1", port: int = 8710,
           max_num_seqs: int = 256, max_prefill_tokens: int = 0,
           spec_lookup: int = 0, no_graphs: bool = False,
           prefix_cache: bool = False):
    """Run one GPU engine server (start one per GPU for DP; repeat
    --model to co-serve several models from one replica)."""
    from .serving.engine_server import main as engine_main
    argv = ["--host", host, "--port", str(port),
            "--max-num-seqs", str(max_num_seqs),
            "--spec-lookup", str(spec_lookup)]
    if max_prefill_tokens:
        argv += ["--max-prefill-tokens", str(max_prefill_tokens)]
    if no_graphs:
        argv.append("--no-graphs")
    for m in model:
        argv += ["--model", m]
    if prefix_cache:
        argv.append("--prefix-cache")
    if device:
        argv += ["--device", device]
    sys.argv = ["engine_server"] + argv
    engine_main()


AGENT_TEMPLATE = '''"""Agent scaffolded by `af init`."""
from agentfield_amd.sdk import Agent

app = Agent("{name}")


@app.reasoner()
def greet(name: str):
    """A minimal reasoner; swap the body for app.ai(...) calls."""
    return {{"greeting": f"hello {{name}}"}}


@app.reasoner()
def think(question: str):
    answer = app.ai(question, system="Answer concisely.", max_tokens=128)
    return {{"answer": answer}}


if __name__ == "__main__":
    app.serve(port=8600)
'''


FULL_TEMPLATE = '''"""Agent scaffolded by `af init --template full`.

Shows the main SDK surfaces: reasoners, deterministic skills with result
caching, app.ai() structured output, cross-agent calls, shared memory
and lifecycle actions.  Start the control plane (`af server`), then
`af dev {name}/` for the watch-reload loop.
"""
from agentfield_amd.sdk import Agent

app = Agent("{name}", vc_enabled=False)


@app.reasoner(tags=["demo"])
def answer(question: str):
    """LLM-backed reasoner served by the in-process MI355X engine."""
    text = app.ai(question, system="Answer concisely.", max_tokens=128)
    app.memory.workflow.set("last_question", question)
    return {{"answer": text}}


@app.reasoner()
def extract(text: str):
    """Structured output: the engine grammar-guarantees valid JSON."""
    data = app.ai(f"Extract entities from: {{text}}",
                  schema={{"type": "object",
                           "properties": {{"entities": {{"type": "array"}}}}}})
    return {{"data": data}}


@app.skill(cache_results=True)
def word_count(text: str):
    """Deterministic skill; repeated inputs hit the result cache."""
    return {{"words": len(text.split())}}


@app.reasoner()
def delegate(task: str):
    """Nested cross-agent call — traced into the workflow DAG."""
    return {{"handled_by": "{name}",
             "note": "use app.call('other.reasoner', ...) to fan out"}}


@app.on_action("reload-config")
def reload_config(payload):
    """Handle control-plane lifecycle actions (claim/ack lease queue)."""
    return {{"reloaded": True}}


if __name__ == "__main__":
    app.serve(port=8600)
'''

FULL_README = """# {name}

Scaffolded by `af init {name} --template full`.

    af server                 # control plane
    af dev {name}/            # watch-reload dev loop
    af run {name}/agent.py    # plain run

    curl -X POST localhost:8520/api/v1/execute/{name}.answer \\
         -d '{{"input": {{"question": "hello"}}}}'

Add MCP servers in `mcp.json`; discover them with `af mcp discover`.
"""


@app.command()
def init(name: str, directory: str = ".",
         template: str = typer.Option(
             "minimal", help="minimal | full (reference init.go templates)")):
    """Scaffold a new agent project."""
    root = Path(directory) / name
    root.mkdir(parents=True, exist_ok=True)
    if template == "full":
        (root / "agent.py").write_text(FULL_TEMPLATE.format(name=name))
        (root / "README.md").write_text(FULL_README.format(name=name))
        (root / "mcp.json").write_text(json.dumps(
            {"mcpServers": {}}, indent=2))
    else:
        (root / "agent.py").write_text(AGENT_TEMPLATE.format(name=name))
    (root / "agentfield.yaml").write_text(
        f"name: {name}\nentrypoint: agent.py\n")
    typer.echo(f"scaffolded {root}/agent.py — run with: af run {root}/agent.py")


@app.command()
def run(path: str, port: int = 8600, host: str = "127.0.0.1",
        agentfield_url: str = DEFAULT_URL):
    """Run an agent app file (finds the Agent instance and serves it)."""
    p = Path(path)
    if p.is_dir():
        p = p / "agent.py"
    spec = importlib.util.spec_from_file_location("af_user_agent", p)
    mod = importlib.util.module_from_spec(spec)
    os.environ.setdefault("AGENTFIELD_URL", agentfield_url)
    spec.loader.exec_module(mod)
    from .sdk import Agent
    agents = [v for v in vars(mod).values() if isinstance(v, Agent)]
    if not agents:
        typer.echo("no Agent instance found in module", err=True)
        raise typer.Exit(1)
    agents[0].serve(host=host, port=port)


@app.command()
def dev(path: str, port: int = 8600, host: str = "127.0.0.1",
        agentfield_url: str = DEFAULT_URL,
        poll: float = typer.Option(0.5, help="file-watch poll interval")):
    """Watch-reload dev loop (reference C32: `af dev`, dev.go:37): run
    the agent and restart it whenever a .py/.yaml/.json file under its
    directory changes."""
    import subprocess
    import time as _t
    p = Path(path)
    watch_root = p if p.is_dir() else p.parent

    def snapshot():
        out = {}
        for pat in ("**/*.py", "**/*.yaml", "**/*.json"):
            for f in watch_root.glob(pat):
                if ".agentfield" in f.parts or "__pycache__" in f.parts:
                    continue
                try:
                    out[str(f)] = f.stat().st_mtime
                except OSError:
                    pass
        return out

    def spawn():
        return subprocess.Popen(
            [sys.executable, "-m", "agentfield_amd", "run", str(p),
             "--port", str(port), "--host", host,
             "--agentfield-url", agentfield_url],
            env={**os.environ}, preexec_fn=_die_with_parent)

    typer.echo(f"dev: watching {watch_root} (restart on change, ctrl-c to"
               " stop)")
    state = snapshot()
    proc = spawn()
    try:
        while True:
            _t.sleep(poll)
            if proc.poll() is not None:
                typer.echo(f"dev: agent exited rc={proc.returncode}; "
                           "waiting for a change to restart")
            cur = snapshot()
            if cur != state:
                changed = [k for k in cur
                           if state.get(k) != cur[k]] or \
                    [k for k in state if k not in cur]
                typer.echo(f"dev: change in {Path(changed[0]).name} — "
                           "restarting")
                state = cur
                if proc.poll() is None:
                    proc.terminate()
                    try:
                        proc.wait(timeout=5)
                    except subprocess.TimeoutExpired:
                        proc.kill()
                proc = spawn()
    except KeyboardInterrupt:
        pass
    finally:
        if proc.poll() is None:
            proc.terminate()


def _client():
    import httpx
    return httpx.Client(base_url=DEFAULT_URL, timeout=10.0)


@app.command("list")
def list_nodes():
    """List registered agent nodes."""
    r = _client().get("/api/v1/nodes").json()
    for n in r.get("nodes", []):
        typer.echo(f"{n['id']:24s} {n['status']:10s} {n.get('base_url', '')}"
                   f"  reasoners={len(n.get('reasoners', []))}")


@app.command()
def executions(limit: int = 20, status: str = None):
    """Recent executions."""
    params = {"limit": limit}
    if status:
        params["status"] = status
    r = _client().get("/api/ui/v1/executions", params=params).json()
    for e in r.get("executions", []):
        typer.echo(f"{e['id']} {e['status']:10s} "
                   f"{e.get('node_id')}.{e.get('reasoner_id')} "
                   f"{e.get('duration_ms') or ''}")


@app.command()
def status():
    """Control-plane health."""
    try:
        r = _client().get("/api/v1/health").json()
        typer.echo(json.dumps(r, indent=2))
    except Exception as e:
        typer.echo(f"control plane unreachable: {e}", err=True)
        raise typer.Exit(1)


def _registry():
    from .controlplane.packages import PackageRegistry
    return PackageRegistry(os.environ.get("AGENTFIELD_DATA", ".agentfield"))


def _procs():
    from .controlplane.packages import ProcessManager
    return ProcessManager(os.environ.get("AGENTFIELD_DATA", ".agentfield"))


@app.command()
def install(source: str, name: str = None):
    """Install an agent package from a directory, archive or git repo."""
    ent = _registry().install(source, name)
    typer.echo(f"installed {ent['name']} -> {ent['path']}")


@app.command()
def add(source: str, name: str = None):
    """Alias of install."""
    install(source, name)


@app.command()
def uninstall(name: str):
    if _registry().uninstall(name):
        typer.echo(f"uninstalled {name}")
    else:
        typer.echo("not installed", err=True)
        raise typer.Exit(1)


@app.command("packages")
def list_packages():
    for p in _registry().list():
        typer.echo(f"{p['name']:24s} {p['entrypoint']:12s} {p['path']}")


@app.command()
def start(name: str, agentfield_url: str = DEFAULT_URL, port: int = None):
    """Start an installed agent package as a managed process."""
    pkg = _registry().get(name)
    if pkg is None:
        typer.echo(f"package '{name}' not installed", err=True)
        raise typer.Exit(1)
    pm = _procs()
    info = pm.start(pkg, agentfield_url, port)
    ok = pm.wait_ready(name)
    typer.echo(f"{name}: pid={info['pid']} {info['base_url']} "
               f"{'ready' if ok else 'NOT READY (see logs)'}")


@app.command()
def doctor():
    """Environment diagnostics: ROCm toolchain, GPU visibility, native
    extensions, engine smoke.  Run this first when anything misbehaves."""
    import shutil
    import subprocess as sp

    def row(label, ok, detail=""):
        mark = "ok " if ok else "MISSING"
        typer.echo(f"  [{mark:7s}] {label}{': ' + detail if detail else ''}")

    typer.echo("toolchain:")
    hipcc = shutil.which("hipcc")
    row("hipcc", bool(hipcc), hipcc or "")
    for tool in ("rocm-smi", "rocprofv3", "g++", "cmake"):
        row(tool, bool(shutil.which(tool)))
    typer.echo("python:")
    import torch
    row("torch", True, torch.__version__)
    cuda = torch.cuda.is_available()
    row("GPU visible", cuda,
        torch.cuda.get_device_name(0) if cuda else "CPU-only mode")
    if cuda:
        arch = torch.cuda.get_device_properties(0).gcnArchName
        row("gfx950", "gfx950" in arch, arch)
    typer.echo("extensions:")
    from pathlib import Path as _P
    pkg = _P(__file__).resolve().parent
    so = pkg / "libafops.so"
    row("libafops.so (HIP kernels)", so.exists(),
        f"{so.stat().st_size // 1024} KiB" if so.exists() else
        "run python agentfield_amd/build.py")
    try:
        import agentfield_amd._native  # noqa: F401
        row("_native (C++ scheduler/crypto)", True)
    except ImportError as e:
        row("_native (C++ scheduler/crypto)", False, str(e)[:60])
    typer.echo("engine:")
    try:
        import torch as _t
        from .engine import LLMEngine, SamplingParams
        from .models import CONFIGS
        dev = "cuda" if cuda else "cpu"
        kw = {} if cuda else {"dtype": _t.float32, "num_pages": 64,
                              "page_size": 4, "enable_graphs": False}
        eng = LLMEngine(CONFIGS["tiny"], device=dev, max_num_seqs=2,
                        seed=0, **kw)
        out = eng.generate([[1, 2, 3]],
                           SamplingParams(max_tokens=4, ignore_eos=True))[0]
        row("tiny-model decode", len(out) == 4, f"{dev}: {out}")
    except Exception as e:
        row("tiny-model decode", False, str(e)[:80])


@app.command()
def stop(name: str):
    """Stop a managed agent process."""
    if _procs().stop(name):
        typer.echo(f"stopped {name}")
    else:
        typer.echo(f"{name} was not running")


@app.command()
def logs(name: str, lines: int = 50):
    """Tail a managed agent's log."""
    typer.echo(_procs().logs(name, lines))


@app.command()
def ps():
    """Status of managed agent processes."""
    pm = _procs()
    for pkg in _registry().list():
        st = pm.status(pkg["name"])
        state = f"pid={st['pid']} {st.get('base_url','')}" if st.get("running") else "stopped"
        typer.echo(f"{pkg['name']:24s} {state}")


@app.command("config")
def show_config(config: str = typer.Option(None, help="YAML config file")):
    """Show the effective control-plane configuration (env > YAML > defaults)."""
    from .controlplane.server import Config
    kw = {}
    if config:
        import yaml
        kw = yaml.safe_load(Path(config).read_text()) or {}
    cfg = Config(**kw)
    typer.echo(json.dumps({k: v for k, v in vars(cfg).items()
                           if not k.startswith("_")}, indent=2, default=str))


mcp_app = typer.Typer()
app.add_typer(mcp_app, name="mcp",
              help="MCP server tools (reference C32 `af mcp` verbs)")


def _mcp_config(project: str) -> dict:
    from .mcp.manager import discover_config
    return discover_config(project)


@mcp_app.command("discover")
def mcp_discover(project: str = "."):
    """Run the discovery fallback chain (stdio -> HTTP -> static
    analysis) for every configured server and cache capabilities."""
    from .mcp.discovery import CapabilityCache, discover_server
    cache = CapabilityCache(project)
    cfg = _mcp_config(project)
    if not cfg:
        typer.echo("no MCP config (mcp.json) found", err=True)
        raise typer.Exit(1)
    for alias, spec in cfg.items():
        entry = discover_server(alias, spec, project, cache)
        typer.echo(f"{alias:20s} source={entry['source']:6s} "
                   f"tools={[t['name'] for t in entry['tools']]}")


@mcp_app.command("status")
def mcp_status(project: str = "."):
    """Configured servers, their cached capabilities, and liveness of
    CLI-managed processes."""
    from .mcp.discovery import CapabilityCache
    cache = CapabilityCache(project)
    cfg = _mcp_config(project)
    aliases = sorted(set(cfg) | set(cache.aliases()))
    if not aliases:
        typer.echo("no MCP servers configured or cached")
        return
    for alias in aliases:
        entry = cache.get(alias) or {}
        pid = _mcp_pid(project, alias)
        state = "running" if pid else "stopped"
        tools = [t["name"] for t in entry.get("tools", [])]
        typer.echo(f"{alias:20s} {state:8s} "
                   f"source={entry.get('source', '-'):6s} tools={tools}")


def _mcp_dir(project: str, alias: str) -> Path:
    return Path(project) / ".agentfield" / "mcp" / alias


def _mcp_pid(project: str, alias: str) -> int | None:
    p = _mcp_dir(project, alias) / "pid"
    if not p.exists():
        return None
    try:
        pid = int(p.read_text().strip())
        os.kill(pid, 0)
        return pid
    except (ValueError, ProcessLookupError, PermissionError):
        return None


@mcp_app.command("start")
def mcp_start(name: str, project: str = "."):
    """Start a configured MCP server as a detached managed process
    (pidfile + log under .agentfield/mcp/<name>/)."""
    import subprocess
    spec = _mcp_config(project).get(name)
    if spec is None or not spec.get("command"):
        typer.echo(f"no startable MCP server '{name}' configured", err=True)
        raise typer.Exit(1)
    if _mcp_pid(project, name):
        typer.echo(f"{name} already running")
        return
    d = _mcp_dir(project, name)
    d.mkdir(parents=True, exist_ok=True)
    log = open(d / "server.log", "ab")
    proc = subprocess.Popen(
        [spec["command"], *spec.get("args", [])],
        stdin=subprocess.PIPE, stdout=log, stderr=log,
        cwd=spec.get("cwd"), env={**os.environ, **spec.get("env", {})},
        start_new_session=True)
    (d / "pid").write_text(str(proc.pid))
    typer.echo(f"{name}: pid={proc.pid} log={d / 'server.log'}")


@mcp_app.command("stop")
def mcp_stop(name: str, project: str = "."):
    pid = _mcp_pid(project, name)
    if pid is None:
        typer.echo(f"{name} not running")
        return
    import signal as _signal
    os.kill(pid, _signal.SIGTERM)
    (_mcp_dir(project, name) / "pid").unlink(missing_ok=True)
    typer.echo(f"stopped {name} (pid {pid})")


@mcp_app.command("restart")
def mcp_restart(name: str, project: str = "."):
    mcp_stop(name, project)
    mcp_start(name, project)


@mcp_app.command("logs")
def mcp_logs(name: str, lines: int = 50, project: str = "."):
    p = _mcp_dir(project, name) / "server.log"
    if not p.exists():
        typer.echo("no log yet")
        return
    content = p.read_text(errors="replace").splitlines()
    typer.echo("\n".join(content[-lines:]))


@mcp_app.command("skills")
def mcp_skills(project: str = ".", out: str = "mcp_skills"):
    """Generate importable skill modules from cached capabilities
    (reference skill_generator.go)."""
    from .mcp.discovery import CapabilityCache, generate_skill_file
    cache = CapabilityCache(project)
    wrote = []
    for alias in cache.aliases():
        entry = cache.get(alias)
        if entry and entry.get("tools"):
            wrote.append(str(generate_skill_file(alias, entry["tools"],
                                                 Path(project) / out)))
    if not wrote:
        typer.echo("no cached capabilities — run `af mcp discover` first",
                   err=True)
        raise typer.Exit(1)
    for w in wrote:
        typer.echo(w)


@mcp_app.command("call")
def mcp_call(tool: str, args_json: str = "{}", project_dir: str = "."):
    """Start the project's MCP servers and invoke one tool."""
    from .mcp import MCPManager
    mgr = MCPManager()
    try:
        mgr.start_all(project_dir)
        out = mgr.call(tool, json.loads(args_json))
        typer.echo(json.dumps(out, indent=2))
    finally:
        mgr.stop_all()


vc_app = typer.Typer()
app.add_typer(vc_app, name="vc", help="Verifiable-credential tools")


@vc_app.command("verify")
def vc_verify(file: str, report: bool = typer.Option(
        False, "--report", help="comprehensive scored report (chain-aware)")):
    """Offline verification of a VC JSON document or an exported chain
    (reference: `af vc verify` + vc_verification_enhanced.go)."""
    from .controlplane.did import VCService
    doc = json.loads(Path(file).read_text())
    if "vc" in doc:
        doc = doc["vc"]
    if report or "credentials" in doc or isinstance(doc, list):
        res = offline_chain_report(doc)
        typer.echo(json.dumps(res, indent=2))
        raise typer.Exit(0 if res["valid"] else 1)
    res = VCService.verify_document(doc)
    typer.echo(json.dumps(res, indent=2))
    raise typer.Exit(0 if res["valid"] else 1)


def offline_chain_report(doc) -> dict:
    """Comprehensive offline report over one VC or an exported chain
    ({"credentials": [...]} from /api/v1/did/export/vcs): per-credential
    signature + structure + compliance scoring without any control-plane
    state (stored-record cross-checks require the online endpoint
    POST /api/ui/v1/executions/:id/verify-vc)."""
    import time as _t
    from .controlplane.did import VCService

    creds = doc.get("credentials") if isinstance(doc, dict) else doc
    if creds is None:
        creds = [doc]
    svc = VCService.__new__(VCService)  # offline: no storage/dids needed
    components = {}
    for c in creds:
        subj = c.get("credentialSubject", {})
        eid = subj.get("execution_id") or c.get("id", "?")
        rec = {"issuer_did": c.get("issuer"),
               "execution_id": subj.get("execution_id"),
               "run_id": subj.get("workflow_id")}
        result = {"verification_timestamp":
                  _t.strftime("%Y-%m-%dT%H:%M:%SZ", _t.gmtime()),
                  "critical_issues": [], "warnings": []}
        result["integrity_checks"] = svc._integrity_checks(rec, c, None)
        result["security_analysis"] = svc._security_analysis(rec, c)
        result["compliance_checks"] = svc._compliance_checks(c)
        for sec in ("integrity_checks", "security_analysis",
                    "compliance_checks"):
            for issue in result[sec]["issues"]:
                if issue["severity"] == "critical":
                    result["critical_issues"].append(issue)
                elif issue["severity"] == "warning":
                    result["warnings"].append(issue)
        result["valid"] = not result["critical_issues"]
        result["overall_score"] = svc._score(result)
        components[eid] = result
    scores = [c["overall_score"] for c in components.values()]
    return {
        "components": components,
        "count": len(components),
        "valid": bool(components) and all(c["valid"]
                                          for c in components.values()),
        "overall_score": round(sum(scores) / len(scores), 2)
        if scores else 0.0,
    }


@app.command()
def bench(steps: int = 4, warmup: int = 1, model: str = "llama-3-8b",
          calls: int = 64):
    """Run the serving benchmark (wraps bench.py)."""
    import subprocess
    repo = Path(__file__).resolve().parent.parent
    subprocess.run([sys.executable, str(repo / "bench.py"),
                    "--steps", str(steps), "--warmup", str(warmup),
                    "--model", model, "--calls", str(calls)], check=True)


def main():
    app()


if __name__ == "__main__":
    main()
