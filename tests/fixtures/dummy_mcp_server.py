"""Minimal MCP stdio server used by tests: one tool 'adder'."""
import json
import sys

TOOLS = [{
    "name": "adder",
    "description": "add two numbers",
    "inputSchema": {"type": "object",
                    "properties": {"a": {"type": "number"},
                                   "b": {"type": "number"}},
                    "required": ["a", "b"]},
}]

for line in sys.stdin:
    try:
        msg = json.loads(line)
    except ValueError:
        continue
    method = msg.get("method")
    mid = msg.get("id")
    if mid is None:
        continue  # notification
    if method == "initialize":
        result = {"protocolVersion": "2024-11-05",
                  "serverInfo": {"name": "dummy", "version": "1.0"},
                  "capabilities": {"tools": {}}}
    elif method == "tools/list":
        result = {"tools": TOOLS}
    elif method == "tools/call":
        p = msg["params"]
        if p["name"] == "adder":
            s = p["arguments"]["a"] + p["arguments"]["b"]
            result = {"content": [{"type": "text", "text": str(s)}]}
        else:
            print(json.dumps({"jsonrpc": "2.0", "id": mid,
                              "error": {"code": -32601,
                                        "message": "no such tool"}}),
                  flush=True)
            continue
    else:
        result = {}
    print(json.dumps({"jsonrpc": "2.0", "id": mid, "result": result}),
          flush=True)
