"""MCP capability discovery with fallbacks (reference parity: C31,
capability_discovery.go:442-1360).

Discovery order per server, mirroring the reference:
  1. live stdio JSON-RPC (tryStdioDiscovery) — spawn, initialize,
     tools/list + resources/list
  2. live HTTP JSON-RPC (tryHTTPDiscovery / discoverFromURL) for specs
     with a `url`
  3. static source analysis (discoverFromStaticAnalysis) — regex scan of
     Node/Python server sources for tool/resource registrations
  4. manifest metadata (parseManifestFile / package.json "mcp" block)

Results cache to <project>/.agentfield/mcp/<alias>/capabilities.json
(CacheCapabilities) and refresh on demand.
"""
from __future__ import annotations

import json
import re
import time
from pathlib import Path

from .client import MCPError, MCPStdioClient


class MCPHttpClient:
    """JSON-RPC 2.0 over HTTP POST (MCP streamable-http transport)."""

    def __init__(self, url: str, timeout: float = 10.0):
        self.url = url
        self.timeout = timeout
        self._id = 0
        self.server_info: dict = {}

    def _rpc(self, method: str, params: dict | None = None) -> dict:
        import httpx
        self._id += 1
        req = {"jsonrpc": "2.0", "id": self._id, "method": method}
        if params is not None:
            req["params"] = params
        r = httpx.post(self.url, json=req, timeout=self.timeout,
                       headers={"Accept": "application/json"})
        r.raise_for_status()
        msg = r.json()
        if "error" in msg:
            raise MCPError(str(msg["error"]))
        return msg.get("result", {})

    def initialize(self) -> dict:
        res = self._rpc("initialize", {
            "protocolVersion": "2024-11-05", "capabilities": {},
            "clientInfo": {"name": "agentfield-amd", "version": "0.1.0"}})
        self.server_info = res.get("serverInfo", {})
        return res

    def list_tools(self) -> list[dict]:
        return self._rpc("tools/list").get("tools", [])

    def list_resources(self) -> list[dict]:
        try:
            return self._rpc("resources/list").get("resources", [])
        except Exception:
            return []

    def call_tool(self, name: str, arguments: dict) -> dict:
        return self._rpc("tools/call", {"name": name, "arguments": arguments})

    @property
    def alive(self) -> bool:
        try:
            self.list_tools()
            return True
        except Exception:
            return False

    def close(self) -> None:
        pass


# ---------------------------------------------------------------- static
# Registration patterns the reference greps for (extractToolNameFrom*)
_JS_TOOL = [
    re.compile(r"""name:\s*["']([\w.-]+)["']"""),
    re.compile(r"""server\.tool\(\s*["']([\w.-]+)["']"""),
    re.compile(r"""registerTool\(\s*["']([\w.-]+)["']"""),
]
_PY_TOOL = [
    re.compile(r"""@(?:\w+\.)?tool\(\s*(?:name\s*=\s*)?["']([\w.-]+)["']"""),
    re.compile(r"""Tool\(\s*name\s*=\s*["']([\w.-]+)["']"""),
    re.compile(r"""add_tool\(\s*["']([\w.-]+)["']"""),
]
_PY_TOOL_DECOR = re.compile(
    r"""@(?:\w+\.)?tool\(\s*\)\s*\n\s*(?:async\s+)?def\s+(\w+)""")
_PY_RES = re.compile(r"""@(?:\w+\.)?resource\(\s*["']([^"']+)["']""")
_JS_RES = re.compile(r"""registerResource\(\s*["']([\w.-]+)["']""")


def _scan_text(text: str, patterns) -> list[str]:
    names: list[str] = []
    for pat in patterns:
        for m in pat.finditer(text):
            if m.group(1) not in names:
                names.append(m.group(1))
    return names


def static_analysis(server_dir: str | Path) -> tuple[list[dict], list[dict]]:
    """Best-effort tool/resource extraction from sources when the server
    cannot be started (discoverFromStaticAnalysis)."""
    root = Path(server_dir)
    tools: list[dict] = []
    resources: list[dict] = []

    def add_tools(names, src):
        for n in names:
            if not any(t["name"] == n for t in tools):
                tools.append({"name": n, "description": f"from {src}",
                              "discovered": "static"})

    # package.json "mcp" metadata block (parseNodeJSPackage/parseMCPMetadata)
    pkg = root / "package.json"
    if pkg.exists():
        try:
            data = json.loads(pkg.read_text())
            mcp = data.get("mcp", {})
            for t in mcp.get("tools", []):
                name = t if isinstance(t, str) else t.get("name")
                if name:
                    add_tools([name], "package.json")
            for r in mcp.get("resources", []):
                name = r if isinstance(r, str) else r.get("name")
                if name:
                    resources.append({"name": name,
                                      "discovered": "static"})
        except ValueError:
            pass
    # mcp.manifest.json (parseManifestFile)
    man = root / "mcp.manifest.json"
    if man.exists():
        try:
            data = json.loads(man.read_text())
            add_tools([t.get("name") for t in data.get("tools", [])
                       if t.get("name")], "manifest")
        except ValueError:
            pass
    for js in list(root.glob("**/*.js"))[:50] + list(root.glob("**/*.ts"))[:50]:
        try:
            text = js.read_text(errors="replace")
        except OSError:
            continue
        if "tool" not in text and "Tool" not in text:
            continue
        add_tools(_scan_text(text, _JS_TOOL), js.name)
        for n in _JS_RES.findall(text):
            resources.append({"name": n, "discovered": "static"})
    for py in list(root.glob("**/*.py"))[:50]:
        try:
            text = py.read_text(errors="replace")
        except OSError:
            continue
        add_tools(_scan_text(text, _PY_TOOL), py.name)
        add_tools(_PY_TOOL_DECOR.findall(text), py.name)
        for n in _PY_RES.findall(text):
            resources.append({"name": n, "discovered": "static"})
    return tools, resources


# ---------------------------------------------------------------- cache
class CapabilityCache:
    """<project>/.agentfield/mcp/<alias>/capabilities.json
    (CacheCapabilities / GetServerCapability)."""

    def __init__(self, project_dir: str | Path):
        self.root = Path(project_dir) / ".agentfield" / "mcp"

    def path(self, alias: str) -> Path:
        return self.root / alias / "capabilities.json"

    def put(self, alias: str, tools, resources, source: str) -> dict:
        entry = {"server": alias, "discovered_at": time.time(),
                 "source": source, "tools": tools, "resources": resources}
        p = self.path(alias)
        p.parent.mkdir(parents=True, exist_ok=True)
        p.write_text(json.dumps(entry, indent=2))
        return entry

    def get(self, alias: str) -> dict | None:
        p = self.path(alias)
        if p.exists():
            try:
                return json.loads(p.read_text())
            except ValueError:
                return None
        return None

    def aliases(self) -> list[str]:
        if not self.root.exists():
            return []
        return sorted(d.name for d in self.root.iterdir() if d.is_dir())


def discover_server(alias: str, spec: dict, project_dir: str | Path = ".",
                    cache: CapabilityCache | None = None) -> dict:
    """Full fallback chain for one server spec; caches the result."""
    cache = cache or CapabilityCache(project_dir)
    tools: list[dict] = []
    resources: list[dict] = []
    source = "none"
    if spec.get("url"):
        try:
            c = MCPHttpClient(spec["url"])
            c.initialize()
            tools, resources = c.list_tools(), c.list_resources()
            source = "http"
        except Exception:
            pass
    if source == "none" and spec.get("command"):
        try:
            c = MCPStdioClient([spec["command"], *spec.get("args", [])],
                               env=spec.get("env"), cwd=spec.get("cwd"))
            try:
                c.initialize()
                tools, resources = c.list_tools(), c.list_resources()
                source = "stdio"
            finally:
                c.close()
        except Exception:
            pass
    if source == "none":
        sdir = spec.get("cwd") or spec.get("dir") or project_dir
        tools, resources = static_analysis(sdir)
        source = "static" if tools or resources else "none"
    return cache.put(alias, tools, resources, source)


# -------------------------------------------------------- skill files
def generate_skill_file(alias: str, tools: list[dict],
                        out_dir: str | Path) -> Path:
    """Write an importable skills module for a server's tools
    (skill_generator.go parity: the generated file registers one
    @app.skill per tool on any Agent passed to register())."""
    out = Path(out_dir) / f"mcp_{alias}_skills.py"
    lines = [
        '"""Auto-generated MCP skill bindings for server '
        f"'{alias}' — regenerate with `af mcp skills`.\"\"\"",
        "from agentfield_amd.mcp import MCPManager",
        "",
        f"SERVER = {alias!r}",
        f"TOOLS = {[t['name'] for t in tools]!r}",
        "",
        "",
        "def register(app, manager: MCPManager):",
        '    """Attach one skill per discovered tool."""',
        "    names = []",
        "    for tool in TOOLS:",
        "        def make(tn):",
        "            def skill(**kwargs):",
        "                return manager.call(tn, kwargs)",
        "            skill.__name__ = f'mcp_{tn}'",
        "            return skill",
        "        app.skill(name=f'mcp_{tool}', tags=['mcp', SERVER])("
        "make(tool))",
        "        names.append(f'mcp_{tool}')",
        "    return names",
        "",
    ]
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text("\n".join(lines))
    return out
