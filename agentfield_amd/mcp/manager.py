"""MCP server manager: config discovery, lifecycle, tool->skill generation
(reference parity: C31 manager/capability_discovery/skill_generator)."""
from __future__ import annotations

import json
from pathlib import Path

from .client import MCPStdioClient

CONFIG_NAMES = ("mcp.json", ".mcp.json", "mcp_servers.json")


def discover_config(project_dir: str) -> dict:
    """Find {"mcpServers": {name: {command, args, env, cwd}}}."""
    root = Path(project_dir)
    for name in CONFIG_NAMES:
        p = root / name
        if p.exists():
            cfg = json.loads(p.read_text())
            return cfg.get("mcpServers", cfg)
    return {}


class MCPManager:
    def __init__(self):
        self.servers: dict[str, object] = {}
        self.specs: dict[str, dict] = {}
        self.tools: dict[str, tuple[str, dict]] = {}  # tool -> (server, schema)

    def start_server(self, name: str, spec: dict) -> list[dict]:
        if spec.get("url"):
            # HTTP JSON-RPC transport (reference tryHTTPDiscovery)
            from .discovery import MCPHttpClient
            client = MCPHttpClient(spec["url"])
        else:
            cmd = [spec["command"], *spec.get("args", [])]
            client = MCPStdioClient(cmd, env=spec.get("env"),
                                    cwd=spec.get("cwd"))
        client.initialize()
        self.servers[name] = client
        self.specs[name] = spec
        tools = client.list_tools()
        for t in tools:
            self.tools[t["name"]] = (name, t)
        return tools

    def stop_server(self, name: str) -> bool:
        c = self.servers.pop(name, None)
        if c is None:
            return False
        c.close()
        self.tools = {t: v for t, v in self.tools.items() if v[0] != name}
        return True

    def restart_server(self, name: str) -> list[dict]:
        spec = self.specs[name]
        self.stop_server(name)
        return self.start_server(name, spec)

    def status(self) -> list[dict]:
        """Per-server health for the enhanced heartbeat / af mcp status."""
        return [{"name": n, "alive": bool(getattr(c, "alive", False)),
                 "transport": "http" if self.specs.get(n, {}).get("url")
                 else "stdio",
                 "tools": sorted(t for t, (srv, _s) in self.tools.items()
                                 if srv == n)}
                for n, c in self.servers.items()]

    def start_all(self, project_dir: str) -> dict[str, list[dict]]:
        out = {}
        for name, spec in discover_config(project_dir).items():
            try:
                out[name] = self.start_server(name, spec)
            except Exception as e:
                out[name] = [{"error": str(e)}]
        return out

    def call(self, tool: str, arguments: dict) -> dict:
        server, _schema = self.tools[tool]
        return self.servers[server].call_tool(tool, arguments)

    def health(self) -> dict:
        return {name: c.alive for name, c in self.servers.items()}

    def stop_all(self) -> None:
        for c in self.servers.values():
            c.close()
        self.servers.clear()
        self.tools.clear()

    def register_as_skills(self, agent) -> list[str]:
        """Auto-generate one @app.skill per discovered MCP tool (the
        reference's dynamic_skills / skill_generator behavior)."""
        registered = []
        for tool_name, (server, schema) in self.tools.items():
            skill_name = f"mcp_{tool_name}"

            def make(tn):
                def mcp_skill(**kwargs):
                    return self.call(tn, kwargs)
                mcp_skill.__name__ = f"mcp_{tn}"
                mcp_skill.__doc__ = schema.get("description", "")
                return mcp_skill

            agent.skill(name=skill_name, tags=["mcp", server])(make(tool_name))
            registered.append(skill_name)
        return registered
