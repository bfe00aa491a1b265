"""Failure detection / recovery semantics: presence
sweeps, stale-execution cleanup, agent-death handling, webhook warm start."""
import time

import httpx

from agentfield_amd.controlplane import ControlPlane, create_app
from agentfield_amd.controlplane.server import Config
from agentfield_amd.controlplane.services import (ExecutionCleanup,
                                                  PresenceManager)
from agentfield_amd.controlplane.storage import Storage
from agentfield_amd.controlplane.events import Buses

from helpers import AppServer, wait_until


def test_presence_sweep_marks_inactive_and_evicts():
    st = Storage(":memory:")
    buses = Buses()
    pm = PresenceManager(st, buses, ttl=10.0, hard_evict=100.0)
    st.upsert_node({"id": "n1", "base_url": "http://x"})
    st.set_node_status("n1", "active")
    now = time.time()
    assert pm.sweep_once(now=now) == []            # fresh heartbeat
    assert pm.sweep_once(now=now + 20) == ["n1"]   # ttl passed -> inactive
    assert st.get_node("n1")["status"] == "inactive"
    assert pm.sweep_once(now=now + 200) == ["n1"]  # hard evict -> stopped
    assert st.get_node("n1")["status"] == "stopped"


def test_cleanup_marks_stale_running_and_deletes_old():
    st = Storage(":memory:")
    st.create_execution({"id": "exec_stale", "status": "running"})
    # stale (started 4000s ago) but still within retention
    st._exec("UPDATE executions SET started_at=? WHERE id=?",
             (time.time() - 4000, "exec_stale"))
    cl = ExecutionCleanup(st, stale_s=1800, retention_s=86400)
    out = cl.run_once()
    assert out["stale_marked"] == 1 and out["deleted"] == 0
    assert st.get_execution("exec_stale")["status"] == "failed"
    # age it past retention -> deleted
    st._exec("UPDATE executions SET created_at=? WHERE id=?",
             (time.time() - 200000, "exec_stale"))
    out = cl.run_once()
    assert out["deleted"] == 1
    assert st.get_execution("exec_stale") is None


def test_execute_against_dead_agent_fails_fast():
    cp = ControlPlane(Config(background_services=False, agent_timeout=2.0))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    try:
        cp.storage.upsert_node({"id": "ghost",
                                "base_url": "http://127.0.0.1:9",
                                "reasoners": [{"id": "r"}]})
        r = httpx.post(srv.base_url + "/api/v1/execute/ghost.r",
                       json={"input": {}}, timeout=30.0)
        body = r.json()
        assert body["status"] == "failed"
        assert "unreachable" in body["error_message"]
    finally:
        srv.stop()


def test_webhook_warm_start_redelivers(tmp_path):
    """Webhooks pending in the DB survive a control-plane restart: the new
    dispatcher's poller re-drives them (reference warm-start scan)."""
    db = str(tmp_path / "wh.db")
    st1 = Storage(db)
    st1.create_execution({"id": "exec_w", "run_id": "run_w", "input": {}})
    st1.register_webhook("exec_w", "http://will-be-replaced", "")
    st1.update_execution_result("exec_w", "completed", {"ok": 1})
    st1.stage_webhook_payload("exec_w", {"event": "execution.completed",
                                         "execution_id": "exec_w",
                                         "status": "completed"})
    st1.close()

    hits = []
    from fastapi import FastAPI, Request
    hook = FastAPI()

    @hook.post("/h")
    async def recv(req: Request):
        hits.append(await req.json())
        return {"ok": True}

    hook_srv = AppServer(hook).start()
    try:
        st2 = Storage(db)
        st2._exec("UPDATE execution_webhooks SET url=? WHERE execution_id=?",
                  (hook_srv.base_url + "/h", "exec_w"))
        cp = ControlPlane.__new__(ControlPlane)  # only need the dispatcher
        from agentfield_amd.controlplane.services import (Metrics,
                                                          WebhookDispatcher)
        disp = WebhookDispatcher(st2, Metrics(), poll_interval=0.2)

        async def run():
            import asyncio
            await disp.start()
            for _ in range(300):
                if hits:
                    break
                await asyncio.sleep(0.1)
            await disp.stop()
        import asyncio
        asyncio.run(run())
        assert hits and hits[0]["execution_id"] == "exec_w"
        assert st2.get_webhook("exec_w")["status"] == "delivered"
    finally:
        hook_srv.stop()
