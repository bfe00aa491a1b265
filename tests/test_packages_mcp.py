"""Package manager, process lifecycle, MCP subsystem, checkpointing,
resilience, admin gRPC."""
import json
import sys
import time
from pathlib import Path

import httpx
import pytest
import torch

from agentfield_amd.controlplane.packages import (PackageRegistry,
                                                  ProcessManager)
from agentfield_amd.mcp import MCPManager, MCPStdioClient
from agentfield_amd.sdk.resilience import (CircuitOpenError, ResultCache,
                                           StatelessRateLimiter)

FIXTURES = Path(__file__).parent / "fixtures"


# ------------------------------------------------------------ packages
def _scaffold(tmp_path, name="pkgagent"):
    from typer.testing import CliRunner
    from agentfield_amd.cli import app as cli_app
    CliRunner().invoke(cli_app, ["init", name, "--directory", str(tmp_path)])
    return tmp_path / name


def test_install_list_uninstall(tmp_path):
    src = _scaffold(tmp_path)
    reg = PackageRegistry(str(tmp_path / "data"))
    ent = reg.install(str(src))
    assert ent["name"] == "pkgagent"
    assert (Path(ent["path"]) / "agent.py").exists()
    assert [p["name"] for p in reg.list()] == ["pkgagent"]
    assert reg.get("pkgagent")["entrypoint"] == "agent.py"
    assert reg.uninstall("pkgagent")
    assert reg.list() == []


def test_install_from_git_repo(tmp_path):
    import subprocess
    src = _scaffold(tmp_path, "gitagent")
    subprocess.run(["git", "init", "-q"], cwd=src, check=True)
    subprocess.run(["git", "add", "-A"], cwd=src, check=True)
    subprocess.run(["git", "-c", "user.email=t@t", "-c", "user.name=t",
                    "commit", "-qm", "x"], cwd=src, check=True)
    reg = PackageRegistry(str(tmp_path / "data"))
    ent = reg.install(str(src))  # .git dir present -> clone path
    assert (Path(ent["path"]) / "agent.py").exists()


def test_process_manager_lifecycle(tmp_path):
    src = _scaffold(tmp_path, "procagent")
    reg = PackageRegistry(str(tmp_path / "data"))
    ent = reg.install(str(src))
    pm = ProcessManager(str(tmp_path / "data"))
    info = pm.start(ent, "http://127.0.0.1:1")  # dead control plane is fine
    try:
        assert pm.wait_ready("procagent", timeout=30.0), pm.logs("procagent")
        r = httpx.get(info["base_url"] + "/health", timeout=5.0)
        assert r.json()["node_id"] == "procagent"
        st = pm.status("procagent")
        assert st["running"] and st["pid"] == info["pid"]
    finally:
        assert pm.stop("procagent")
    assert not pm.status("procagent")["running"]


# ------------------------------------------------------------ MCP
def _dummy_spec():
    return {"command": sys.executable,
            "args": [str(FIXTURES / "dummy_mcp_server.py")]}


def test_mcp_stdio_client_tools():
    c = MCPStdioClient([sys.executable, str(FIXTURES / "dummy_mcp_server.py")])
    try:
        init = c.initialize()
        assert c.server_info["name"] == "dummy"
        tools = c.list_tools()
        assert tools[0]["name"] == "adder"
        res = c.call_tool("adder", {"a": 2, "b": 40})
        assert res["content"][0]["text"] == "42"
    finally:
        c.close()


def test_mcp_manager_skill_generation():
    from agentfield_amd.sdk import Agent
    app = Agent("mcphost", auto_register=False)
    mgr = MCPManager()
    try:
        mgr.start_server("dummy", _dummy_spec())
        skills = mgr.register_as_skills(app)
        assert skills == ["mcp_adder"]
        assert "mcp_adder" in app._skills
        assert mgr.health() == {"dummy": True}
        out = mgr.call("adder", {"a": 1, "b": 2})
        assert out["content"][0]["text"] == "3"
    finally:
        mgr.stop_all()


def test_mcp_config_discovery(tmp_path):
    from agentfield_amd.mcp.manager import discover_config
    (tmp_path / "mcp.json").write_text(json.dumps(
        {"mcpServers": {"d": _dummy_spec()}}))
    cfg = discover_config(str(tmp_path))
    assert "d" in cfg and cfg["d"]["command"] == sys.executable


# ------------------------------------------------------------ checkpoint
def test_checkpoint_roundtrip(tmp_path):
    from agentfield_amd.models import CONFIGS, LlamaForCausalLM
    from agentfield_amd.models.checkpoint import (config_from_dir,
                                                  load_checkpoint,
                                                  save_checkpoint)
    cfg = CONFIGS["tiny"]
    m1 = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32).init_random(1)
    save_checkpoint(m1, str(tmp_path / "ckpt"))
    cfg2 = config_from_dir(str(tmp_path / "ckpt"))
    assert cfg2.hidden_size == cfg.hidden_size
    assert cfg2.num_kv_heads == cfg.num_kv_heads
    m2 = LlamaForCausalLM(cfg2, device="cpu", dtype=torch.float32)
    load_checkpoint(m2, str(tmp_path / "ckpt"))
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(),
                                  m2.named_parameters()):
        assert torch.equal(p1, p2), n1


def test_checkpoint_tp_slices(tmp_path):
    from agentfield_amd.models.llama import LlamaConfig
    from agentfield_amd.models import LlamaForCausalLM
    from agentfield_amd.models.checkpoint import (load_checkpoint,
                                                  save_checkpoint)
    cfg = LlamaConfig(name="t", hidden_size=512, intermediate_size=1024,
                      num_layers=1, num_heads=4, num_kv_heads=2,
                      vocab_size=256, max_position=128)
    full = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32).init_random(3)
    save_checkpoint(full, str(tmp_path / "c"))
    shard = LlamaForCausalLM(cfg.shard(2), device="cpu", dtype=torch.float32)
    load_checkpoint(shard, str(tmp_path / "c"), tp=2, rank=1)
    D = cfg.head_dim
    # rank 1's q rows are full rows [2*D : 4*D)
    want_q = full.layers[0].attn.qkv[2 * D:4 * D]
    assert torch.equal(shard.layers[0].attn.qkv[:2 * D], want_q)
    # vocab-parallel lm_head: rank 1 holds global vocab rows [128:256)
    assert shard.lm_head.shape[0] == cfg.vocab_size // 2
    assert torch.equal(shard.lm_head, full.lm_head[128:256])
    # sharded models refuse full-model save
    with pytest.raises(ValueError, match="unsharded"):
        save_checkpoint(shard, str(tmp_path / "c2"))


# ------------------------------------------------------------ resilience
def test_rate_limiter_backoff_and_breaker():
    rl = StatelessRateLimiter(base_delay=0.001, breaker_threshold=3,
                              breaker_reset=0.2, seed=1)
    calls = []

    def flaky():
        calls.append(1)
        raise ValueError("nope")

    with pytest.raises(ValueError):
        rl.call(flaky, retries=5)
    assert rl.is_open
    with pytest.raises(CircuitOpenError):
        rl.call(flaky, retries=0)
    time.sleep(0.25)
    assert not rl.is_open  # half-open allows a probe
    assert rl.call(lambda: 42, retries=0) == 42
    assert not rl.is_open


def test_result_cache_ttl_lru():
    c = ResultCache(max_entries=2, ttl=0.1)
    c.put("a", 1)
    c.put("b", 2)
    assert c.get("a") == 1
    c.put("c", 3)  # evicts LRU ("b")
    assert c.get("b") is None
    time.sleep(0.15)
    assert c.get("a") is None  # expired
    assert c.purge_expired() >= 0


def test_skill_result_cache():
    from agentfield_amd.sdk import Agent
    app = Agent("cachehost", auto_register=False)
    calls = []

    @app.skill(cache_results=True)
    def slow_add(a: int, b: int):
        calls.append(1)
        return a + b

    assert slow_add(1, 2) == 3
    assert slow_add(1, 2) == 3
    assert len(calls) == 1  # second call served from cache
    assert slow_add(2, 2) == 4
    assert len(calls) == 2


# ------------------------------------------------------------ admin gRPC
def test_admin_grpc():
    import asyncio
    from agentfield_amd.controlplane import ControlPlane
    from agentfield_amd.controlplane.server import Config
    from agentfield_amd.controlplane.admin_grpc import (AdminClient,
                                                        start_admin_grpc)
    cp = ControlPlane(Config(background_services=False))
    cp.storage.upsert_node({"id": "n1", "base_url": "http://x",
                            "reasoners": [{"id": "r1"}]})
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    server = start_admin_grpc(cp, port=port)
    try:
        cl = AdminClient(f"127.0.0.1:{port}")
        rs = cl.list_reasoners()
        assert rs["reasoners"][0]["id"] == "r1"
        ns = cl.list_nodes()
        assert ns["nodes"][0]["id"] == "n1"
    finally:
        server.stop(grace=0)


def test_keystore_aes_gcm_encryption(tmp_path):
    import base64
    import secrets as _s
    from agentfield_amd.controlplane.did import Keystore
    kek = _s.token_bytes(32)
    ks = Keystore(str(tmp_path / "k.key"), kek=kek)
    seed1 = ks.seed
    raw = (tmp_path / "k.key").read_text()
    assert raw.startswith("enc:")
    assert base64.urlsafe_b64encode(seed1).decode().rstrip("=") not in raw
    ks2 = Keystore(str(tmp_path / "k.key"), kek=kek)
    assert ks2.seed == seed1
    ks_bad = Keystore(str(tmp_path / "k.key"), kek=_s.token_bytes(32))
    import pytest as _pt
    with _pt.raises(Exception):
        _ = ks_bad.seed


def test_wait_for_result_sse_nudge():
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config
    from agentfield_amd.sdk.client import AgentFieldClient
    from helpers import AppServer
    import threading
    cp = ControlPlane(Config(background_services=False))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    try:
        cp.storage.create_execution({"id": "exec_sse", "run_id": "run_sse",
                                     "input": {}})
        cl = AgentFieldClient(srv.base_url)

        def finish_later():
            time.sleep(0.5)
            # completing publishes the SSE event that nudges the waiter
            import asyncio
            cp.complete_execution("exec_sse", "completed", {"ok": 1})

        threading.Thread(target=finish_later, daemon=True).start()
        t0 = time.time()
        rec = cl.wait_for_result("exec_sse", timeout=20.0, poll_initial=5.0,
                                 poll_max=5.0)
        waited = time.time() - t0
        assert rec["status"] == "completed"
        # with 5s polls, finishing in ~<2s proves the SSE nudge woke us
        assert waited < 4.0, f"SSE nudge did not wake poll loop ({waited:.1f}s)"
    finally:
        srv.stop()
