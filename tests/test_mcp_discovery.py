"""MCP discovery parity (reference C31, capability_discovery.go:442-1360):
HTTP JSON-RPC transport, static-analysis fallback, capability caching,
skill-file generation, `af mcp` verbs, and MCP health in heartbeats."""
import json
import subprocess
import sys
import time
from pathlib import Path

import httpx
import pytest

from agentfield_amd.mcp.discovery import (CapabilityCache, MCPHttpClient,
                                          discover_server,
                                          generate_skill_file,
                                          static_analysis)

ROOT = Path(__file__).resolve().parent.parent
FIXTURES = Path(__file__).parent / "fixtures"


NODE_SRC = """
const server = new McpServer({name: "files"});
server.tool("read_file", "Read a file", {}, async (a) => {});
server.registerTool("write_file", {description: "Write"}, fn);
server.registerResource("workspace", uri, meta, fn);
"""

PY_SRC = """
from mcp.server.fastmcp import FastMCP
mcp = FastMCP("calc")

@mcp.tool()
def add(a: int, b: int) -> int:
    return a + b

@mcp.tool(name="multiply")
def mul(a: int, b: int) -> int:
    return a * b

@mcp.resource("calc://history")
def history():
    return []
"""


def test_static_analysis_node_and_python(tmp_path):
    (tmp_path / "server.js").write_text(NODE_SRC)
    (tmp_path / "calc.py").write_text(PY_SRC)
    (tmp_path / "package.json").write_text(json.dumps(
        {"name": "x", "mcp": {"tools": ["from_manifest"]}}))
    tools, resources = static_analysis(tmp_path)
    names = {t["name"] for t in tools}
    assert {"read_file", "write_file", "add", "multiply",
            "from_manifest"} <= names
    assert {r["name"] for r in resources} >= {"workspace", "calc://history"}


def test_discovery_chain_stdio_then_cache(tmp_path):
    spec = {"command": sys.executable,
            "args": [str(FIXTURES / "dummy_mcp_server.py")]}
    entry = discover_server("dummy", spec, tmp_path)
    assert entry["source"] == "stdio"
    assert [t["name"] for t in entry["tools"]] == ["adder"]
    cache = CapabilityCache(tmp_path)
    assert cache.get("dummy")["tools"][0]["name"] == "adder"
    assert cache.aliases() == ["dummy"]


def test_discovery_falls_back_to_static(tmp_path):
    sdir = tmp_path / "srv"
    sdir.mkdir()
    (sdir / "main.py").write_text(PY_SRC)
    spec = {"command": "/nonexistent/bin", "cwd": str(sdir)}
    entry = discover_server("broken", spec, tmp_path)
    assert entry["source"] == "static"
    assert {t["name"] for t in entry["tools"]} >= {"add", "multiply"}


def test_http_jsonrpc_discovery(tmp_path):
    """discoverFromURL parity: JSON-RPC 2.0 over HTTP POST."""
    from fastapi import FastAPI, Request

    sys.path.insert(0, str(ROOT / "tests"))
    from helpers import AppServer

    app = FastAPI()

    @app.post("/mcp")
    async def rpc(req: Request):
        msg = await req.json()
        m = msg["method"]
        result = {}
        if m == "initialize":
            result = {"serverInfo": {"name": "httpd"}, "capabilities": {}}
        elif m == "tools/list":
            result = {"tools": [{"name": "ping",
                                 "description": "pong"}]}
        elif m == "tools/call":
            result = {"content": [{"type": "text", "text": "pong"}]}
        return {"jsonrpc": "2.0", "id": msg["id"], "result": result}

    srv = AppServer(app).start()
    try:
        c = MCPHttpClient(srv.base_url + "/mcp")
        c.initialize()
        assert c.server_info["name"] == "httpd"
        assert [t["name"] for t in c.list_tools()] == ["ping"]
        out = c.call_tool("ping", {})
        assert out["content"][0]["text"] == "pong"
        # the discovery chain prefers the URL transport
        entry = discover_server("httpd", {"url": srv.base_url + "/mcp"},
                                tmp_path)
        assert entry["source"] == "http"
        assert entry["tools"][0]["name"] == "ping"
    finally:
        srv.stop()


def test_skill_file_generation_and_registration(tmp_path):
    from agentfield_amd.mcp import MCPManager
    from agentfield_amd.sdk import Agent
    p = generate_skill_file("dummy", [{"name": "adder"}], tmp_path)
    assert p.exists()
    import importlib.util
    spec = importlib.util.spec_from_file_location("genmod", p)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    app = Agent("host", auto_register=False)
    mgr = MCPManager()
    mgr.start_server("dummy", {"command": sys.executable,
                               "args": [str(FIXTURES /
                                            "dummy_mcp_server.py")]})
    try:
        names = mod.register(app, mgr)
        assert names == ["mcp_adder"]
        out = app._skills["mcp_adder"].fn(a=2, b=5)
        assert "7" in json.dumps(out)
        # manager status feeds the enhanced heartbeat
        st = mgr.status()
        assert st[0]["name"] == "dummy" and st[0]["alive"]
        assert st[0]["tools"] == ["adder"]
    finally:
        mgr.stop_all()


def test_af_mcp_cli_verbs(tmp_path):
    (tmp_path / "mcp.json").write_text(json.dumps({"mcpServers": {
        "dummy": {"command": sys.executable,
                  "args": [str(FIXTURES / "dummy_mcp_server.py")]}}}))

    def af(*args):
        return subprocess.run(
            [sys.executable, "-m", "agentfield_amd", "mcp", *args,
             "--project", str(tmp_path)],
            capture_output=True, text=True, cwd=ROOT)

    r = af("discover")
    assert r.returncode == 0 and "stdio" in r.stdout and "adder" in r.stdout
    r = af("status")
    assert "dummy" in r.stdout and "stopped" in r.stdout
    r = af("skills")
    assert r.returncode == 0
    gen = tmp_path / "mcp_skills" / "mcp_dummy_skills.py"
    assert gen.exists() and "TOOLS = ['adder']" in gen.read_text()
    # start -> running in status -> logs -> stop
    r = af("start", "dummy")
    assert "pid=" in r.stdout
    try:
        time.sleep(0.3)
        assert "running" in af("status").stdout
    finally:
        r = af("stop", "dummy")
        assert "stopped" in r.stdout


def test_heartbeat_carries_mcp_health():
    sys.path.insert(0, str(ROOT / "tests"))
    from helpers import AppServer
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config
    from agentfield_amd.sdk import Agent

    cp = ControlPlane(Config(background_services=False, did_enabled=False))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    try:
        agent = Agent("mcphost", agentfield_url=srv.base_url,
                      auto_register=False, heartbeat_interval=0.1)
        agent.base_url = "http://127.0.0.1:1"  # never probed here
        assert agent.register()
        from agentfield_amd.mcp import MCPManager
        agent.mcp = MCPManager()
        agent.mcp.start_server("dummy", {
            "command": sys.executable,
            "args": [str(FIXTURES / "dummy_mcp_server.py")]})
        try:
            agent.start_background()
            deadline = time.time() + 5
            seen = None
            while time.time() < deadline:
                r = httpx.get(srv.base_url + "/api/ui/v1/mcp",
                              timeout=2.0).json()
                if r.get("servers"):
                    seen = r["servers"]
                    break
                time.sleep(0.1)
            assert seen and seen[0]["name"] == "dummy" and seen[0]["alive"]
        finally:
            agent._hb_stop.set()
            agent.mcp.stop_all()
    finally:
        srv.stop()
