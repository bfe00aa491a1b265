"""End-to-end demo: control plane + two agents + the in-process engine.

Runs entirely on CPU with the tiny model (on an MI355X box it picks
llama-3-8b on cuda automatically).  Shows the core loop a reference user
knows — decorated reasoners, cross-agent calls building a workflow DAG,
app.ai() — plus what's new here: guaranteed-JSON output and memory scopes.

    PYTHONPATH=. python examples/multi_agent_demo.py
"""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))
sys.path.insert(0, str(Path(__file__).parent.parent / "tests"))

import torch

from agentfield_amd.controlplane import ControlPlane, create_app
from agentfield_amd.controlplane.server import Config
from agentfield_amd.engine import LLMEngine
from agentfield_amd.models import CONFIGS
from agentfield_amd.sdk import Agent
from agentfield_amd.sdk.ai import AIConfig, EngineRunner, set_runner
from helpers import AppServer


def main():
    # 1) model: tiny on CPU, llama-3-8b on GPU
    on_gpu = torch.cuda.is_available()
    model = "llama-3-8b" if on_gpu else "tiny"
    kw = {} if on_gpu else {"device": "cpu", "dtype": torch.float32,
                            "page_size": 4, "num_pages": 256,
                            "max_num_seqs": 8, "enable_graphs": False}
    set_runner(model, EngineRunner(LLMEngine(CONFIGS[model], **kw)))

    # 2) control plane
    cp = ControlPlane(Config(background_services=False, sync_timeout=30.0))
    cp_srv = AppServer(create_app(cp)).start().wait_healthy()
    print(f"control plane: {cp_srv.base_url}")

    # 3) two agents
    writer = Agent("writer", agentfield_url=cp_srv.base_url,
                   auto_register=False)

    @writer.reasoner()
    def draft(topic: str):
        text = writer.ai(f"write about {topic}",
                         model=model, max_tokens=24)
        # default scope follows the execution context (workflow);
        # use the explicit global scope to outlive this run
        writer.memory.globals.set("last_topic", topic)
        return {"draft": text}

    @writer.reasoner()
    def structured(topic: str):
        # grammar-constrained: ALWAYS valid JSON, even from a tiny model
        out = writer.ai(f"facts about {topic}", model=model, max_tokens=24,
                        json_only=True)
        return {"json": json.loads(out.strip())}

    editor = Agent("editor", agentfield_url=cp_srv.base_url,
                   auto_register=False)

    @editor.reasoner()
    def publish(topic: str):
        # cross-agent call through the control plane -> DAG edge
        piece = editor.call("writer.draft", topic=topic)
        return {"published": piece, "by": "editor"}

    servers = []
    for app in (writer, editor):
        s = AppServer(app).start()
        app.base_url = s.base_url
        assert app.register()
        servers.append(s)

    # 4) drive it like a reference user would: REST execute
    import httpx
    r = httpx.post(cp_srv.base_url + "/api/v1/execute/editor.publish",
                   json={"input": {"topic": "xGMI"}}, timeout=120.0).json()
    print("publish ->", json.dumps(r["result"])[:120], "...")
    run_id = r["run_id"]

    r2 = httpx.post(cp_srv.base_url + "/api/v1/execute/writer.structured",
                    json={"input": {"topic": "HBM"}}, timeout=120.0).json()
    print("structured ->", json.dumps(r2["result"])[:120])

    dag = httpx.get(cp_srv.base_url +
                    f"/api/ui/v1/workflows/{run_id}/dag").json()
    print(f"DAG: {len(dag['nodes'])} nodes "
          f"(editor.publish -> writer.draft), status={dag['status']}")
    print("memory last_topic =", writer.memory.globals.get("last_topic"))

    for s in servers:
        s.stop()
    cp_srv.stop()
    print("demo complete")


if __name__ == "__main__":
    main()
