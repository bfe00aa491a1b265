"""Model-Context-Protocol stdio client (reference parity: C31/P17 —
stdio JSON-RPC discovery of tools + invocation; mcp_stdio_bridge.py).

Speaks JSON-RPC 2.0 over a child process's stdin/stdout per the MCP spec:
initialize -> tools/list -> tools/call.
"""
from __future__ import annotations

import json
import subprocess
import threading


class MCPError(RuntimeError):
    pass


class MCPStdioClient:
    def __init__(self, command: list[str], env: dict | None = None,
                 cwd: str | None = None, timeout: float = 30.0):
        import os
        self.timeout = timeout
        self.proc = subprocess.Popen(
            command, stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            stderr=subprocess.DEVNULL, cwd=cwd,
            env={**os.environ, **(env or {})}, text=True, bufsize=1)
        self._id = 0
        self._lock = threading.Lock()
        self.server_info: dict = {}

    def _rpc(self, method: str, params: dict | None = None) -> dict:
        with self._lock:
            self._id += 1
            req = {"jsonrpc": "2.0", "id": self._id, "method": method}
            if params is not None:
                req["params"] = params
            self.proc.stdin.write(json.dumps(req) + "\n")
            self.proc.stdin.flush()
            # read until we get the response for our id (skip notifications)
            while True:
                line = self.proc.stdout.readline()
                if not line:
                    raise MCPError(f"MCP server closed stdout (method={method})")
                try:
                    msg = json.loads(line)
                except ValueError:
                    continue
                if msg.get("id") == self._id:
                    if "error" in msg:
                        raise MCPError(str(msg["error"]))
                    return msg.get("result", {})

    def notify(self, method: str, params: dict | None = None) -> None:
        with self._lock:
            req = {"jsonrpc": "2.0", "method": method}
            if params is not None:
                req["params"] = params
            self.proc.stdin.write(json.dumps(req) + "\n")
            self.proc.stdin.flush()

    def initialize(self) -> dict:
        res = self._rpc("initialize", {
            "protocolVersion": "2024-11-05",
            "capabilities": {},
            "clientInfo": {"name": "agentfield-amd", "version": "0.1.0"},
        })
        self.server_info = res.get("serverInfo", {})
        self.notify("notifications/initialized")
        return res

    def list_tools(self) -> list[dict]:
        return self._rpc("tools/list").get("tools", [])

    def call_tool(self, name: str, arguments: dict) -> dict:
        return self._rpc("tools/call", {"name": name, "arguments": arguments})

    def list_resources(self) -> list[dict]:
        try:
            return self._rpc("resources/list").get("resources", [])
        except MCPError:
            return []

    @property
    def alive(self) -> bool:
        return self.proc.poll() is None

    def close(self) -> None:
        try:
            self.proc.terminate()
            self.proc.wait(timeout=3)
        except Exception:
            try:
                self.proc.kill()
            except Exception:
                pass
