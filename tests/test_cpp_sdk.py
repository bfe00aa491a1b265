"""C++ SDK integration: build the example agent with make, run it against a
real control plane, execute reasoners/skills through the REST API (the same
harness pattern the reference uses for its Go SDK)."""
import subprocess
import time
from pathlib import Path

import httpx
import pytest

from agentfield_amd.controlplane import ControlPlane, create_app
from agentfield_amd.controlplane.server import Config

from helpers import AppServer, wait_until

CPP_DIR = Path(__file__).parent.parent / "sdk" / "cpp"


@pytest.fixture(scope="module")
def cpp_binary():
    subprocess.run(["make", "-s", "example_agent"], cwd=CPP_DIR, check=True)
    return CPP_DIR / "example_agent"


@pytest.fixture(scope="module")
def stack(cpp_binary):
    cp = ControlPlane(Config(background_services=False, sync_timeout=15.0))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    proc = subprocess.Popen([str(cpp_binary), "cppagent", srv.base_url],
                            stdout=subprocess.PIPE, text=True)
    try:
        wait_until(lambda: cp.storage.get_node("cppagent"), timeout=15.0)
        yield srv, cp
    finally:
        proc.terminate()
        proc.wait(timeout=5)
        srv.stop()


def test_cpp_agent_registers(stack):
    srv, cp = stack
    node = cp.storage.get_node("cppagent")
    assert node["metadata"]["sdk"] == "agentfield_amd_cpp"
    ids = [r["id"] for r in node["reasoners"]]
    assert "shout" in ids and "relay" in ids
    assert [s["id"] for s in node["skills"]] == ["mul"]


def test_cpp_reasoner_via_202_callback(stack):
    srv, _ = stack
    r = httpx.post(srv.base_url + "/api/v1/execute/cppagent.shout",
                   json={"input": {"text": "quiet"}}, timeout=20.0)
    body = r.json()
    assert body["status"] == "completed", body
    assert body["result"] == {"shouted": "QUIET"}


def test_cpp_skill(stack):
    srv, _ = stack
    r = httpx.post(srv.base_url + "/api/v1/execute/cppagent.mul",
                   json={"input": {"a": 6, "b": 7}}, timeout=20.0)
    assert r.json()["result"] == {"product": 42}


def test_cpp_nested_call_dag(stack):
    srv, cp = stack
    r = httpx.post(srv.base_url + "/api/v1/execute/cppagent.relay",
                   json={"input": {"x": 4}}, timeout=30.0)
    body = r.json()
    assert body["status"] == "completed", body
    assert body["result"]["relayed"] == {"product": 40}


def test_cpp_agent_health_direct(stack):
    srv, cp = stack
    node = cp.storage.get_node("cppagent")
    h = httpx.get(node["base_url"] + "/health", timeout=5.0).json()
    assert h["node_id"] == "cppagent"


def test_cpp_action_claim_ack(stack):
    """The C++ agent drains the control plane's action lease queue on its
    heartbeat (5 s in the example binary) and acks with the outcome."""
    _srv, cp = stack
    aid = cp.storage.enqueue_action("cppagent", "ping", {"n": 1})
    bad = cp.storage.enqueue_action("cppagent", "no-handler")

    def acked(i, want=None):
        rows = cp.storage._q("SELECT * FROM node_actions WHERE id=?", (i,))
        return rows and rows[0]["status"] == "acked" and \
            (want is None or rows[0]["ack_status"] == want)

    assert wait_until(lambda: acked(aid, "done"), timeout=15.0)
    assert wait_until(lambda: acked(bad, "ignored"), timeout=15.0)
