"""End-to-end control-plane tests: real localhost servers, real agent apps
(the reference's integration-tier pattern)."""
import hashlib
import hmac
import json
import threading
import time

import httpx
import pytest
from fastapi import FastAPI, Request

from agentfield_amd.controlplane import ControlPlane, create_app
from agentfield_amd.controlplane.server import Config
from agentfield_amd.sdk import Agent

from helpers import AppServer, wait_until


@pytest.fixture(scope="module")
def cp_server():
    cp = ControlPlane(Config(background_services=False, sync_timeout=15.0))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    yield srv, cp
    srv.stop()


@pytest.fixture(scope="module")
def greeting_agent(cp_server):
    srv, _cp = cp_server
    app = Agent("greeter", agentfield_url=srv.base_url, auto_register=False)

    @app.reasoner()
    def greet(name: str):
        return {"greeting": f"hello {name}"}

    @app.reasoner()
    def fail_always(x: int):
        raise ValueError("intentional boom")

    @app.skill()
    def add(a: int, b: int):
        return a + b

    @app.reasoner()
    def relay(name: str):
        # nested cross-agent call -> DAG edge
        inner = app.call("greeter.greet", name=name.upper())
        return {"relayed": inner}

    agent_srv = AppServer(app).start()
    app.base_url = agent_srv.base_url
    assert app.register()
    yield agent_srv, app
    agent_srv.stop()


def test_health_and_metrics(cp_server):
    srv, _ = cp_server
    r = httpx.get(srv.base_url + "/api/v1/health")
    assert r.status_code == 200 and r.json()["status"] == "healthy"
    m = httpx.get(srv.base_url + "/metrics").text
    for name in ("agentfield_gateway_queue_depth", "agentfield_worker_inflight",
                 "agentfield_gateway_backpressure_total",
                 "agentfield_step_duration_seconds"):
        assert name in m


def test_register_lists_node_and_identity(cp_server, greeting_agent):
    srv, _ = cp_server
    _, app = greeting_agent
    r = httpx.get(srv.base_url + "/api/v1/nodes").json()
    ids = [n["id"] for n in r["nodes"]]
    assert "greeter" in ids
    assert app.identity.get("agent_did", "").startswith("did:key:z")
    assert "greet" in app.identity.get("reasoner_dids", {})


def test_sync_execute_202_callback_path(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "world"}}, timeout=20.0)
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["status"] == "completed"
    assert body["result"] == {"greeting": "hello world"}
    assert body["execution_id"].startswith("exec_")
    assert body["run_id"].startswith("run_")
    assert r.headers["X-Execution-ID"] == body["execution_id"]
    assert body["duration_ms"] is not None


def test_sync_execute_failure(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.fail_always",
                   json={"input": {"x": 1}}, timeout=20.0)
    body = r.json()
    assert body["status"] == "failed"
    assert "intentional boom" in (body["error_message"] or "")


def test_execute_unknown_node_404(cp_server):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/nosuch.thing",
                   json={"input": {}})
    assert r.status_code == 404


def test_skill_execution_and_target_type(cp_server, greeting_agent):
    srv, cp = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.add",
                   json={"input": {"a": 2, "b": 40}}, timeout=20.0)
    body = r.json()
    assert body["status"] == "completed"
    assert body["result"] == {"result": 42} or body["result"] == 42
    rec = cp.storage.get_execution(body["execution_id"])
    assert rec["target_type"] == "skill"


def test_async_execute_with_hmac_webhook(cp_server, greeting_agent):
    srv, _ = cp_server
    hits = []
    hook = FastAPI()

    @hook.post("/hook")
    async def recv(req: Request):
        body = await req.body()
        hits.append((dict(req.headers), body))
        return {"ok": True}

    hook_srv = AppServer(hook).start()
    try:
        r = httpx.post(
            srv.base_url + "/api/v1/execute/async/greeter.greet",
            json={"input": {"name": "async"},
                  "webhook": {"url": hook_srv.base_url + "/hook",
                              "secret": "s3cret",
                              "headers": {"X-Custom": "yes"}}})
        assert r.status_code == 202
        body = r.json()
        assert body["status"] == "queued" and body["type"] == "reasoner"
        assert body["workflow_id"] == body["run_id"]
        wait_until(lambda: hits, timeout=15.0)
        headers, raw = hits[0]
        assert headers["x-custom"] == "yes"
        sig = headers["x-agentfield-signature"]
        want = "sha256=" + hmac.new(b"s3cret", raw, hashlib.sha256).hexdigest()
        assert sig == want
        payload = json.loads(raw)
        assert payload["event"] == "execution.completed"
        assert payload["result"] == {"greeting": "hello async"}
        assert payload["execution_id"] == body["execution_id"]
        # poll API agrees
        rec = httpx.get(srv.base_url +
                        f"/api/v1/executions/{body['execution_id']}").json()
        assert rec["status"] == "completed"
        assert rec["webhook_registered"] is True
    finally:
        hook_srv.stop()


def test_webhook_retry_on_failure(cp_server, greeting_agent):
    srv, cp = cp_server
    calls = []
    hook = FastAPI()

    @hook.post("/flaky")
    async def recv(req: Request):
        calls.append(time.time())
        if len(calls) < 2:
            from fastapi.responses import JSONResponse
            return JSONResponse({"err": "nope"}, status_code=500)
        return {"ok": True}

    hook_srv = AppServer(hook).start()
    cp.webhooks.poll_interval = 0.2
    cp.webhooks.backoff_base = 0.1
    try:
        r = httpx.post(srv.base_url + "/api/v1/execute/async/greeter.greet",
                       json={"input": {"name": "retry"},
                             "webhook": {"url": hook_srv.base_url + "/flaky"}})
        eid = r.json()["execution_id"]
        wait_until(lambda: len(calls) >= 2, timeout=20.0)
        wait_until(lambda: cp.storage.get_webhook(eid)["status"] == "delivered",
                   timeout=10.0)
        assert len(cp.storage.webhook_history(eid)) >= 2
    finally:
        hook_srv.stop()


def test_batch_status(cp_server, greeting_agent):
    srv, _ = cp_server
    ids = []
    for i in range(3):
        r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                       json={"input": {"name": str(i)}}, timeout=20.0)
        ids.append(r.json()["execution_id"])
    r = httpx.post(srv.base_url + "/api/v1/executions/batch-status",
                   json={"execution_ids": ids + ["exec_bogus"]})
    body = r.json()
    assert set(body["executions"]) == set(ids)
    assert body["missing"] == ["exec_bogus"]


def test_nested_call_builds_dag(cp_server, greeting_agent):
    srv, cp = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.relay",
                   json={"input": {"name": "dag"}}, timeout=30.0)
    body = r.json()
    assert body["status"] == "completed"
    assert body["result"]["relayed"] == {"greeting": "hello DAG"}
    run_id = body["run_id"]
    dag = wait_until(lambda: (lambda d: d if d.status_code == 200 and
                              len(d.json()["nodes"]) >= 2 else None)(
        httpx.get(srv.base_url + f"/api/ui/v1/workflows/{run_id}/dag")),
        timeout=10.0).json()
    assert dag["total"] >= 2
    child = [n for n in dag["nodes"] if n["parent_execution_id"]]
    assert child, "nested call must appear as a DAG edge"
    assert dag["status"] == "completed"
    # v2 workflow runs API
    runs = httpx.get(srv.base_url + "/api/ui/v2/workflow-runs").json()["runs"]
    assert any(x["run_id"] == run_id for x in runs)


def test_memory_scopes_and_vector(cp_server):
    srv, _ = cp_server
    base = srv.base_url + "/api/v1/memory"
    # explicit global scope
    httpx.post(base + "/set", json={"key": "k1", "value": {"v": 1},
                                    "scope": "global"})
    got = httpx.post(base + "/get", json={"key": "k1", "scope": "global"}).json()
    assert got["found"] and got["value"] == {"v": 1}
    # workflow scope from header (A.5 priority)
    hdr = {"X-Run-ID": "run_test123", "X-Session-ID": "sess1"}
    httpx.post(base + "/set", json={"key": "k2", "value": 7}, headers=hdr)
    got = httpx.post(base + "/get", json={"key": "k2"}, headers=hdr).json()
    assert got["scope"] == "workflow" and got["scope_id"] == "run_test123"
    assert got["value"] == 7
    miss = httpx.post(base + "/get", json={"key": "k2"},
                      headers={"X-Session-ID": "sess1"}).json()
    assert not miss["found"]  # different scope
    keys = httpx.get(base + "/list", params={"scope": "global",
                                             "scope_id": "global"}).json()
    assert "k1" in keys["keys"]
    # vector memory
    httpx.post(base + "/vector/set", json={"key": "a", "embedding": [1, 0, 0],
                                           "scope": "global",
                                           "metadata": {"kind": "x"}})
    httpx.post(base + "/vector/set", json={"key": "b", "embedding": [0, 1, 0],
                                           "scope": "global"})
    res = httpx.post(base + "/vector/search",
                     json={"embedding": [0.9, 0.1, 0], "top_k": 2,
                           "scope": "global"}).json()["results"]
    assert res[0]["key"] == "a"
    res = httpx.post(base + "/vector/search",
                     json={"embedding": [0.9, 0.1, 0], "scope": "global",
                           "filters": {"kind": "x"}}).json()["results"]
    assert [r["key"] for r in res] == ["a"]


def test_vc_issuance_verify_and_tamper(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "vc"}}, timeout=20.0)
    eid = r.json()["execution_id"]
    vc = httpx.post(srv.base_url + "/api/v1/execution/vc",
                    json={"execution_id": eid}).json()["vc"]
    assert vc["type"] == ["VerifiableCredential", "AgentFieldExecutionCredential"]
    assert vc["proof"]["proofValue"]
    ver = httpx.post(srv.base_url + "/api/v1/did/verify", json=vc).json()
    assert ver["valid"]
    # server-side verification incl. output-hash check
    got = httpx.get(srv.base_url + f"/api/v1/executions/{eid}/vc").json()
    assert got["verification"]["valid"]
    assert got["verification"]["checks"]["output_hash_matches"]
    # tamper -> invalid
    bad = json.loads(json.dumps(vc))
    bad["credentialSubject"]["execution"]["output_hash"] = "0" * 64
    ver = httpx.post(srv.base_url + "/api/v1/did/verify", json=bad).json()
    assert not ver["valid"]
    # vc chain for the run
    chain = httpx.get(srv.base_url +
                      f"/api/v1/did/workflow/{r.json()['run_id']}/vc-chain").json()
    assert chain["count"] >= 1 and chain["all_valid"]


def test_vc_comprehensive_verification(cp_server, greeting_agent):
    """Scored integrity/security/compliance report (reference parity:
    VerifyExecutionVCComprehensive, vc_service.go:926-1400)."""
    srv, cp = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "comp"}}, timeout=20.0)
    eid, run_id = r.json()["execution_id"], r.json()["run_id"]
    httpx.post(srv.base_url + "/api/v1/execution/vc",
               json={"execution_id": eid})
    rep = httpx.post(srv.base_url +
                     f"/api/ui/v1/executions/{eid}/verify-vc").json()
    assert rep["valid"] and rep["overall_score"] >= 90
    assert rep["integrity_checks"]["hash_validation"]
    assert rep["security_analysis"]["security_score"] == 100.0
    assert rep["compliance_checks"]["w3c_compliance"]
    assert rep["critical_issues"] == []
    # unknown execution -> scored-zero report, not a 500
    rep = httpx.post(srv.base_url +
                     "/api/ui/v1/executions/exec_nope/verify-vc").json()
    assert not rep["valid"] and rep["overall_score"] == 0
    assert rep["critical_issues"][0]["type"] == "vc_not_found"
    # tamper with the stored execution result: hashes + status must flag
    cp.storage._exec("UPDATE executions SET result='{\"x\": 1}' WHERE id=?",
                     (eid,))
    rep = httpx.post(srv.base_url +
                     f"/api/ui/v1/executions/{eid}/verify-vc").json()
    assert not rep["valid"]
    assert any(i["type"] == "output_hash_mismatch"
               for i in rep["critical_issues"])
    # chain-level report
    chain = httpx.get(
        srv.base_url +
        f"/api/v1/did/workflow/{run_id}/vc-chain/verify").json()
    assert chain["workflow_id"] == run_id
    assert eid in chain["components"]
    assert not chain["valid"]  # tampered component propagates


def test_cli_vc_verify_report(tmp_path, cp_server, greeting_agent):
    """`af vc verify --report` over an exported chain file (offline,
    reference: vc_verification_enhanced.go)."""
    import subprocess
    import sys as _sys
    from pathlib import Path
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "rep"}}, timeout=20.0)
    eid, run_id = r.json()["execution_id"], r.json()["run_id"]
    httpx.post(srv.base_url + "/api/v1/execution/vc",
               json={"execution_id": eid})
    export = httpx.get(srv.base_url +
                       f"/api/v1/did/export/vcs?workflow_id={run_id}").json()
    f = tmp_path / "chain.json"
    f.write_text(json.dumps(export))
    out = subprocess.run(
        [_sys.executable, "-m", "agentfield_amd", "vc", "verify", str(f),
         "--report"], capture_output=True, text=True,
        cwd=Path(__file__).resolve().parent.parent)
    assert out.returncode == 0, out.stdout + out.stderr
    rep = json.loads(out.stdout)
    assert rep["valid"] and rep["count"] == 1
    assert rep["overall_score"] >= 90
    comp = rep["components"][eid]
    assert comp["security_analysis"]["security_score"] == 100.0


def test_did_resolve(cp_server, greeting_agent):
    srv, _ = cp_server
    st = httpx.get(srv.base_url + "/api/v1/did/status").json()
    assert st["enabled"] and st["root_did"].startswith("did:key:z")
    doc = httpx.get(srv.base_url +
                    f"/api/v1/did/resolve/{st['root_did']}").json()
    assert doc["document"]["id"] == st["root_did"]


def test_node_status_transitions(cp_server, greeting_agent):
    srv, _ = cp_server
    ok = httpx.post(srv.base_url + "/api/v1/nodes/greeter/status",
                    json={"status": "inactive"})
    assert ok.status_code == 200
    bad = httpx.post(srv.base_url + "/api/v1/nodes/greeter/status",
                     json={"status": "bogus"})
    assert bad.status_code == 400
    httpx.post(srv.base_url + "/api/v1/nodes/greeter/status",
               json={"status": "active"})


def test_heartbeat_unknown_node_asks_reregister(cp_server):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/nodes/ghost/heartbeat", json={})
    assert r.status_code == 404 and r.json()["action"] == "re-register"


def test_ui_dashboard_and_reasoners(cp_server, greeting_agent):
    srv, _ = cp_server
    d = httpx.get(srv.base_url + "/api/ui/v1/dashboard/summary").json()
    assert d["nodes"]["total"] >= 1
    rs = httpx.get(srv.base_url + "/api/ui/v1/reasoners").json()["reasoners"]
    assert any(r["id"] == "greet" for r in rs)


def test_execution_sse_stream(cp_server, greeting_agent):
    srv, _ = cp_server
    got = []

    def listen():
        with httpx.stream("GET", srv.base_url + "/api/ui/v1/executions/events",
                          timeout=10.0) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:"):
                    got.append(json.loads(line[5:]))
                    return

    t = threading.Thread(target=listen, daemon=True)
    t.start()
    time.sleep(0.3)
    httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
               json={"input": {"name": "sse"}}, timeout=20.0)
    t.join(timeout=10.0)
    assert got and got[0]["terminal"]


def test_timeline_and_activity_and_metrics(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "tl"}}, timeout=20.0)
    eid = r.json()["execution_id"]
    tl = httpx.get(srv.base_url +
                   f"/api/ui/v1/executions/{eid}/timeline").json()
    assert tl["execution"]["execution_id"] == eid
    assert any(e["event"] == "completed" for e in tl["events"])
    act = httpx.get(srv.base_url + "/api/ui/v1/activity/recent").json()
    assert any(a["id"] == eid for a in act["activity"])
    m = httpx.get(srv.base_url +
                  "/api/ui/v1/reasoners/greeter/greet/metrics").json()
    assert m["executions"] >= 1 and m["p50_ms"] is not None
    assert m["success_rate"] > 0


def test_cancel_execution_endpoint(cp_server):
    srv, cp = cp_server
    cp.storage.create_execution({"id": "exec_c1", "run_id": "run_c1",
                                 "status": "running", "input": {}})
    r = httpx.post(srv.base_url + "/api/v1/executions/exec_c1/cancel")
    assert r.status_code == 200 and r.json()["status"] == "cancelled"
    r = httpx.post(srv.base_url + "/api/v1/executions/exec_c1/cancel")
    assert r.status_code == 409


def test_async_queue_backpressure_503():
    """Bounded async queue: saturation returns 503 + backpressure metric
    (reference C4: non-blocking submit, 503 'queue is full')."""
    import asyncio
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config

    cp = ControlPlane(Config(background_services=False,
                             async_workers=1, async_queue_capacity=2,
                             agent_timeout=5.0))
    srv = AppServer(create_app(cp)).start().wait_healthy()

    # an agent that hangs long enough to wedge the single worker
    slow = FastAPI()

    @slow.post("/reasoners/stall")
    async def stall(req: Request):
        await asyncio.sleep(3.0)
        return {"result": "late"}

    slow_srv = AppServer(slow).start()
    try:
        cp.storage.upsert_node({"id": "slowpoke",
                                "base_url": slow_srv.base_url,
                                "reasoners": [{"id": "stall"}]})
        codes = []
        for _ in range(6):
            r = httpx.post(srv.base_url + "/api/v1/execute/async/slowpoke.stall",
                           json={"input": {}}, timeout=5.0)
            codes.append(r.status_code)
        assert 202 in codes
        assert 503 in codes, f"expected backpressure, got {codes}"
        m = httpx.get(srv.base_url + "/metrics").text
        assert "agentfield_gateway_backpressure_total" in m
        line = [l for l in m.splitlines()
                if l.startswith("agentfield_gateway_backpressure_total ")][0]
        assert float(line.split()[-1]) >= 1
        # the rejected executions are failed, not lost
        failed = [c for c in codes if c == 503]
        assert len(failed) >= 1
    finally:
        slow_srv.stop()
        srv.stop()


def test_memory_sse_and_history(cp_server):
    srv, _ = cp_server
    got = []

    def listen():
        with httpx.stream("GET", srv.base_url + "/api/v1/memory/events/sse",
                          timeout=10.0) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:"):
                    got.append(json.loads(line[5:]))
                    return

    t = threading.Thread(target=listen, daemon=True)
    t.start()
    time.sleep(0.3)
    t0 = time.time()
    httpx.post(srv.base_url + "/api/v1/memory/set",
               json={"key": "sse_k", "value": 9, "scope": "global"})
    t.join(timeout=10.0)
    assert got and got[0]["key"] == "sse_k" and got[0]["op"] == "set"
    hist = httpx.get(srv.base_url + "/api/v1/memory/events/history",
                     params={"since": t0 - 1}).json()["events"]
    assert any(e["key"] == "sse_k" for e in hist)
    # vector delete roundtrip
    httpx.post(srv.base_url + "/api/v1/memory/vector/set",
               json={"key": "vd", "embedding": [1, 2], "scope": "global"})
    r = httpx.post(srv.base_url + "/api/v1/memory/vector/delete",
                   json={"key": "vd", "scope": "global"}).json()
    assert r["deleted"] is True


def test_include_router_prefix_rewriting(cp_server, greeting_agent):
    """AgentRouter composition: prefixed names register on the agent, stay
    locally callable, and resolve through the control plane's
    node.<dotted.reasoner> target (split on first dot only)."""
    from agentfield_amd.sdk import AgentRouter

    srv, _ = cp_server
    _agent_srv, app = greeting_agent

    billing = AgentRouter(prefix="billing")

    @billing.reasoner()
    def report(month: str):
        return {"month": month, "total": 42}

    inner = AgentRouter()

    @inner.skill()
    def audit(x: int):
        return x * 2

    billing.include_router(inner, prefix="ops")   # nested: billing.ops.audit
    locals_ = app.include_router(billing)
    assert set(locals_) == {"billing.report", "billing.ops.audit"}
    assert locals_["billing.report"]("jan") == {"month": "jan", "total": 42}

    # re-register so the control plane learns the new reasoners
    assert app.register()
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.billing.report",
                   json={"input": {"month": "feb"}}, timeout=20.0)
    assert r.status_code == 200, r.text
    assert r.json()["result"] == {"month": "feb", "total": 42}
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.billing.ops.audit",
                   json={"input": {"x": 21}}, timeout=20.0)
    assert r.status_code == 200, r.text
    assert r.json()["result"] == 42


def test_memory_on_change_sse_push(cp_server):
    """SDK on_change watcher receives events over the live SSE stream with
    push latency (not the 1 s polling fallback)."""
    from agentfield_amd.sdk.client import AgentFieldClient
    from agentfield_amd.sdk.memory import MemoryInterface

    srv, _ = cp_server
    mem = MemoryInterface(AgentFieldClient(srv.base_url), "watcher-node")
    seen = []

    @mem.on_change("watch_*")
    def _on(ev):
        seen.append(ev)

    try:
        # let the watcher connect its SSE stream
        wait_until(lambda: mem._watch_thread is not None and
                   mem._watch_thread.is_alive(), 5.0)
        time.sleep(0.5)
        httpx.post(srv.base_url + "/api/v1/memory/set",
                   json={"key": "watch_k1", "value": {"n": 1},
                         "scope": "global"})
        httpx.post(srv.base_url + "/api/v1/memory/set",
                   json={"key": "other_k", "value": 0, "scope": "global"})
        assert wait_until(lambda: len(seen) >= 1, 5.0)
        time.sleep(0.3)
        assert [e["key"] for e in seen] == ["watch_k1"]  # glob filtered
        assert seen[0]["op"] == "set" and seen[0]["value"] == {"n": 1}
    finally:
        mem.stop()


def test_ui_dashboard_served(cp_server):
    srv, _ = cp_server
    r = httpx.get(srv.base_url + "/")
    assert r.status_code == 200
    assert "agentfield-amd" in r.text and "text/html" in r.headers["content-type"]


def test_admin_grpc_execution_and_status(cp_server, greeting_agent):
    import socket
    from agentfield_amd.controlplane.admin_grpc import (AdminClient,
                                                        start_admin_grpc)
    srv, cp = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "grpc"}}, timeout=20.0)
    eid = r.json()["execution_id"]
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    server = start_admin_grpc(cp, port=port)
    try:
        cl = AdminClient(f"127.0.0.1:{port}")
        st = cl.server_status()
        assert st["status"] == "healthy" and st["nodes"] >= 1
        ex = cl.get_execution(eid)
        assert ex["execution"]["status"] == "completed"
    finally:
        server.stop(grace=0)


def test_node_actions_lease_protocol(cp_server, greeting_agent):
    """Action claim/ack leases (reference C7): pending actions are claimed
    with a lease, re-claimable after expiry, acked exactly once."""
    srv, cp = cp_server
    base = srv.base_url + "/api/v1/nodes/greeter"
    r = httpx.post(base + "/stop")
    assert r.status_code == 200
    aid = r.json()["action_id"]
    lc = httpx.get(base + "/lifecycle").json()
    assert lc["status"] == "stopping" and lc["pending_actions"] >= 1
    # claim with a LONG lease so the within-lease assertion can't race
    # lease expiry on a loaded machine
    acts = httpx.post(base + "/actions/claim",
                      json={"lease_s": 60}).json()["actions"]
    assert any(a["id"] == aid and a["action"] == "stop" for a in acts)
    # second claim within the lease: nothing new
    assert httpx.post(base + "/actions/claim",
                      json={"lease_s": 60}).json()["actions"] == []
    assert httpx.post(base + "/actions/ack",
                      json={"action_id": aid}).status_code == 200
    # double-ack rejected
    assert httpx.post(base + "/actions/ack",
                      json={"action_id": aid}).status_code == 409
    # expiry: a SECOND action claimed with a short lease becomes
    # claimable again once the lease lapses
    aid2 = httpx.post(base + "/stop").json()["action_id"]
    acts = httpx.post(base + "/actions/claim",
                      json={"lease_s": 0.2}).json()["actions"]
    assert any(a["id"] == aid2 for a in acts)
    time.sleep(0.5)  # lease expires -> claimable again
    acts2 = httpx.post(base + "/actions/claim",
                       json={"lease_s": 30}).json()["actions"]
    assert any(a["id"] == aid2 for a in acts2)
    assert httpx.post(base + "/actions/ack",
                      json={"action_id": aid2}).status_code == 200
    # restore status for other tests
    httpx.post(base + "/start")
    httpx.post(srv.base_url + "/api/v1/nodes/greeter/status",
               json={"status": "active"})


def test_node_bulk_status_and_refresh(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/nodes/status/bulk",
                   json={"ids": ["greeter", "missing"]}).json()
    assert r["statuses"]["greeter"]["status"]
    assert r["statuses"]["missing"] is None
    # live refresh probes the real agent server -> healthy
    r = httpx.post(srv.base_url +
                   "/api/v1/nodes/greeter/status/refresh").json()
    assert r["healthy"] is True and r["status"] == "active"


def test_legacy_direct_execution_routes(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/reasoners/greeter/greet",
                   json={"input": {"name": "direct"}}, timeout=20.0)
    assert r.status_code == 200, r.text
    assert r.json()["result"] == {"greeting": "hello direct"}
    r = httpx.post(srv.base_url + "/api/v1/skills/greeter/add",
                   json={"input": {"a": 2, "b": 5}}, timeout=20.0)
    assert r.status_code == 200 and r.json()["result"] == 7


def test_execution_notes(cp_server, greeting_agent):
    srv, _ = cp_server
    r = httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
                   json={"input": {"name": "note-me"}}, timeout=20.0)
    eid = r.json()["execution_id"]
    r = httpx.post(srv.base_url + f"/api/v1/executions/{eid}/notes",
                   json={"note": "checked by ops", "author": "alice"})
    assert r.status_code == 200
    notes = httpx.get(srv.base_url +
                      f"/api/v1/executions/{eid}/notes").json()["notes"]
    assert len(notes) == 1 and notes[0]["note"] == "checked by ops"
    assert notes[0]["author"] == "alice"
    # unknown execution 404s
    assert httpx.post(srv.base_url + "/api/v1/executions/exec_nope/notes",
                      json={"note": "x"}).status_code == 404


def test_reasoner_sse_stream(cp_server, greeting_agent):
    srv, _ = cp_server
    got = []

    def listen():
        with httpx.stream("GET", srv.base_url + "/api/ui/v1/reasoners/events",
                          timeout=10.0) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:"):
                    got.append(json.loads(line[5:]))
                    return

    t = threading.Thread(target=listen, daemon=True)
    t.start()
    time.sleep(0.3)
    httpx.post(srv.base_url + "/api/v1/execute/greeter.greet",
               json={"input": {"name": "sse"}}, timeout=20.0)
    t.join(timeout=10.0)
    assert got and got[0]["type"] == "reasoner.execution"
    assert got[0]["reasoner_id"] == "greet"
    assert got[0]["status"] == "completed"


def test_sdk_action_handler_roundtrip(cp_server, greeting_agent):
    """Agent-side claim/ack: a custom @on_action handler runs on heartbeat
    and the action is acked with its outcome."""
    srv, cp = cp_server
    _agent_srv, app = greeting_agent
    seen = []

    @app.on_action("rotate-keys")
    def rotate(payload):
        seen.append(payload)

    cp.storage.enqueue_action("greeter", "rotate-keys", {"reason": "test"})
    cp.storage.enqueue_action("greeter", "unknown-action")
    app._drain_actions()  # what the heartbeat loop runs
    assert seen == [{"reason": "test"}]
    rows = cp.storage._q(
        "SELECT action, status, ack_status FROM node_actions "
        "WHERE node_id='greeter' AND action IN ('rotate-keys','unknown-action')")
    by = {r["action"]: r for r in rows}
    assert by["rotate-keys"]["status"] == "acked"
    assert by["rotate-keys"]["ack_status"] == "done"
    assert by["unknown-action"]["ack_status"] == "ignored"


def test_ui_config_sanitized(cp_server):
    srv, _ = cp_server
    r = httpx.get(srv.base_url + "/api/ui/v1/config").json()
    assert "sync_timeout" in r["config"]
    joined = json.dumps(r["config"]).lower()
    assert "secret" not in joined and "keystore" not in joined


def test_ui_dashboard_data_contracts(cp_server, greeting_agent):
    """Every endpoint+field the embedded dashboard's JS reads must exist
    (pinned so UI and API can't drift apart)."""
    srv, _ = cp_server
    base = srv.base_url
    r = httpx.post(base + "/api/v1/execute/greeter.relay",
                   json={"input": {"name": "ui"}}, timeout=30.0).json()
    run_id = r["run_id"]
    html = httpx.get(base + "/").text
    for ep in ("/api/ui/v1/dashboard/summary", "/api/ui/v1/nodes",
               "/api/ui/v1/executions", "/api/ui/v2/workflow-runs",
               "/api/ui/v1/config", "/api/v1/did/status",
               "/api/ui/v1/executions/events"):
        assert ep in html, ep
    d = httpx.get(base + "/api/ui/v1/dashboard/summary").json()
    assert {"nodes", "executions", "uptime_s"} <= set(d)
    nodes = httpx.get(base + "/api/ui/v1/nodes").json()["nodes"]
    assert {"id", "status", "reasoners", "skills",
            "last_heartbeat"} <= set(nodes[0])
    ex = httpx.get(base + "/api/ui/v1/executions?limit=5").json()["executions"]
    assert {"id", "node_id", "reasoner_id", "status", "run_id"} <= set(ex[0])
    dag = httpx.get(base + f"/api/ui/v1/workflows/{run_id}/dag").json()
    assert {"execution_id", "node_id", "reasoner_id", "status",
            "parent_execution_id"} <= set(dag["nodes"][0])
    tl = httpx.get(base +
                   f"/api/ui/v1/executions/{ex[0]['id']}/timeline").json()
    assert all({"at", "event"} <= set(e) for e in tl["events"])
    m = httpx.get(base + "/api/ui/v1/reasoners/greeter/greet/metrics").json()
    assert {"executions", "success_rate", "p50_ms"} <= set(m)


def test_distributed_locks(cp_server):
    """Lease locks: exclusive across owners, reentrant for the holder,
    stealable only after expiry, releasable only by the holder."""
    import time as _t

    import httpx
    srv, _cp = cp_server
    url = srv.base_url

    def op(o, **body):
        return httpx.post(f"{url}/api/v1/locks/{o}", json=body,
                          timeout=10.0).json()

    assert op("acquire", name="L", owner="a", ttl_s=5.0)["acquired"]
    assert op("acquire", name="L", owner="a", ttl_s=5.0)["acquired"]  # reentrant
    r = op("acquire", name="L", owner="b", ttl_s=5.0)
    assert not r["acquired"] and r["holder"] == "a"
    assert not op("release", name="L", owner="b")["released"]
    assert op("refresh", name="L", owner="a", ttl_s=5.0)["refreshed"]
    assert op("release", name="L", owner="a")["released"]
    assert op("acquire", name="L", owner="b", ttl_s=0.2)["acquired"]
    _t.sleep(0.35)  # expire b's lease
    assert op("acquire", name="L", owner="c", ttl_s=5.0)["acquired"]
    assert not op("refresh", name="L", owner="b")["refreshed"]
    op("release", name="L", owner="c")


def test_sdk_lock_context_manager(cp_server):
    from agentfield_amd.sdk.client import AgentFieldClient
    from agentfield_amd.sdk.memory import DistributedLock
    srv, _cp = cp_server
    c1 = AgentFieldClient(srv.base_url)
    c2 = AgentFieldClient(srv.base_url)
    with DistributedLock(c1, "job", owner="n1", ttl_s=5.0, timeout_s=1.0):
        l2 = DistributedLock(c2, "job", owner="n2", ttl_s=5.0,
                             timeout_s=0.3, poll_s=0.05)
        assert not l2.acquire()
    # released -> n2 can take it
    assert DistributedLock(c2, "job", owner="n2", ttl_s=5.0,
                           timeout_s=1.0).acquire()


def test_webhook_burst_poller_race():
    """Burst of async executions with webhooks while the retry POLLER
    runs: the poller must never claim a webhook whose execution has not
    completed (that orphaned one whole batch as 'inflight' forever —
    notify() skips inflight rows), and every webhook must deliver."""
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config as CPConfig

    cp = ControlPlane(CPConfig(background_services=True, sync_timeout=30.0,
                               db_path=":memory:"))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    # make the poller race hard: poll every 0.05s instead of 5s
    cp.webhooks.poll_interval = 0.05
    agent = Agent("burst", agentfield_url=srv.base_url, auto_register=False)

    @agent.reasoner()
    def slow(x: int = 0):
        time.sleep(0.2)  # executions outlive several poller cycles
        return {"x": x}

    asrv = AppServer(agent).start()
    agent.base_url = asrv.base_url
    assert agent.register()

    hits = []
    hook = FastAPI()

    @hook.post("/hook")
    async def recv(req: Request):
        hits.append(await req.body())
        return {"ok": True}

    hsrv = AppServer(hook).start()
    N = 96
    with httpx.Client(timeout=30.0) as c:
        for i in range(N):
            r = c.post(srv.base_url + "/api/v1/execute/async/burst.slow",
                       json={"input": {"x": i},
                             "webhook": {"url": hsrv.base_url + "/hook",
                                         "secret": "s"}})
            assert r.status_code == 202
    wait_until(lambda: len(hits) >= N, timeout=90.0)

    def all_marked():
        rows = cp.storage._q("SELECT status, COUNT(*) c FROM "
                             "execution_webhooks GROUP BY status", ())
        return rows == [{"status": "delivered", "c": N}]

    wait_until(all_marked, timeout=10.0)  # status write can lag the hook
    srv.stop()


def test_webhook_stale_inflight_reclaimed():
    """A claim whose worker died (claimed inflight, never delivered)
    becomes due again after the stale window and CAN be re-claimed —
    crash-safe delivery."""
    from agentfield_amd.controlplane.storage import Storage
    st = Storage(":memory:")
    st.register_webhook("e1", "http://x/hook", "sec", {})
    st.stage_webhook_payload("e1", {"event": "execution.completed"})
    # claim with a tiny stale window, simulate worker death (no attempt)
    assert st.try_mark_webhook_inflight("e1", stale_s=0.05)
    assert not st.try_mark_webhook_inflight("e1", stale_s=0.05)  # held
    assert st.due_webhooks() == []          # not yet stale
    time.sleep(0.1)
    due = st.due_webhooks()
    assert [d["execution_id"] for d in due] == ["e1"]  # stale -> due
    assert st.try_mark_webhook_inflight("e1")          # re-claimable
    st.webhook_attempted("e1", True, 200, None, 0.0, 5)
    assert st.get_webhook("e1")["status"] == "delivered"
    st.close()


def test_step_retry_on_transient_failure():
    """A 503 from the agent (restarting under the process manager) is
    retried up to step_retries times with backoff and counted in
    agentfield_step_retries_total; a 500 (agent-side bug) is NOT."""
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config as CPConfig

    cp = ControlPlane(CPConfig(background_services=False, sync_timeout=20.0,
                               step_retries=3, step_retry_backoff_s=0.01))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    flaky_hits = {"n": 0}
    buggy_hits = {"n": 0}
    agent = FastAPI()

    from fastapi.responses import Response as FResponse

    @agent.post("/reasoners/flaky")
    async def flaky(req: Request):
        flaky_hits["n"] += 1
        if flaky_hits["n"] < 3:
            return FResponse(status_code=503)
        return {"result": {"ok": True}}

    @agent.post("/reasoners/buggy")
    async def buggy(req: Request):
        buggy_hits["n"] += 1
        return FResponse(status_code=500)

    asrv = AppServer(agent).start()
    httpx.post(srv.base_url + "/api/v1/nodes/register", json={
        "node_id": "fl", "base_url": asrv.base_url,
        "reasoners": [{"id": "flaky"}, {"id": "buggy"}]}, timeout=10.0)

    r = httpx.post(srv.base_url + "/api/v1/execute/fl.flaky",
                   json={"input": {}}, timeout=30.0)
    assert r.status_code == 200, r.text
    assert flaky_hits["n"] == 3  # two transient 503s, then success

    before = buggy_hits["n"]
    r = httpx.post(srv.base_url + "/api/v1/execute/fl.buggy",
                   json={"input": {}}, timeout=30.0)
    body = r.json()
    assert r.status_code >= 400 or body.get("status") == "failed", body
    assert buggy_hits["n"] == before + 1  # 500 is terminal, no retry
    srv.stop()


def test_engine_metrics_ride_heartbeats():
    """Agents hosting an in-process engine report it in heartbeats; the
    control plane folds the snapshot into agentfield_engine_* metrics
    (token counters advance by deltas, gauges labeled per node)."""
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config as CPConfig

    cp = ControlPlane(CPConfig(background_services=False))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    httpx.post(srv.base_url + "/api/v1/nodes/register", json={
        "node_id": "gpu0", "base_url": "http://127.0.0.1:1",
        "reasoners": []}, timeout=10.0)
    for delta in (100, 50):
        r = httpx.post(srv.base_url + "/api/v1/nodes/gpu0/heartbeat", json={
            "status": "active",
            "engine": {"running": 7, "kv_free_pages": 1234,
                       "prefill_tokens_delta": delta * 8,
                       "decode_tokens_delta": delta}}, timeout=10.0)
        assert r.status_code == 200
    text = httpx.get(srv.base_url + "/metrics", timeout=10.0).text
    assert 'agentfield_engine_tokens_total{kind="decode",node="gpu0"} 150' \
        in text
    assert 'agentfield_engine_batch_occupancy{node="gpu0"} 7' in text
    assert 'agentfield_engine_kv_free_pages{node="gpu0"} 1234' in text
    # dashboard surfaces the latest snapshot per node
    summ = httpx.get(srv.base_url + "/api/ui/v1/dashboard/summary",
                     timeout=10.0).json()
    assert summ["engines"]["gpu0"] == {"running": 7,
                                       "kv_free_pages": 1234}
    srv.stop()
