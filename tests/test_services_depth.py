"""Direct unit coverage for HealthMonitor.probe_all and
ExecutionCleanup.run_once (flagged unit-light in round 1)."""
import asyncio
import sys
import time
from pathlib import Path

from fastapi import FastAPI

sys.path.insert(0, str(Path(__file__).resolve().parent))
from helpers import AppServer

from agentfield_amd.controlplane.events import Buses
from agentfield_amd.controlplane.services import (ExecutionCleanup,
                                                  HealthMonitor)
from agentfield_amd.controlplane.storage import Storage


def test_health_monitor_probe_transitions():
    """active+unreachable -> unhealthy; unhealthy+reachable -> active;
    stopped nodes are never probed."""
    st = Storage(":memory:")
    buses = Buses()
    events = []
    sid, q = buses.node.subscribe()

    healthy_app = FastAPI()

    @healthy_app.get("/health")
    async def health():
        return {"status": "healthy"}

    srv = AppServer(healthy_app).start()
    try:
        st.upsert_node({"id": "up", "base_url": srv.base_url,
                        "status": "active"})
        st.set_node_status("up", "unhealthy")   # will recover
        st.upsert_node({"id": "down", "base_url": "http://127.0.0.1:9",
                        "status": "active"})
        st.set_node_status("down", "active")
        st.upsert_node({"id": "stopped", "base_url": "http://127.0.0.1:9",
                        "status": "active"})
        st.set_node_status("stopped", "stopped")

        hm = HealthMonitor(st, buses, interval=999, timeout=2.0)
        asyncio.run(hm.probe_all())

        assert st.get_node("up")["status"] == "active"
        assert st.get_node("down")["status"] == "unhealthy"
        assert st.get_node("stopped")["status"] == "stopped"  # skipped
        kinds = set()
        while not q.empty():
            kinds.add(q.get_nowait()["type"])
        assert kinds == {"node.active", "node.unhealthy"}
        # second probe: states already settled -> no duplicate events
        asyncio.run(hm.probe_all())
        assert st.get_node("down")["status"] == "unhealthy"
        dup = []
        while not q.empty():
            dup.append(q.get_nowait())
        assert dup == []
    finally:
        buses.node.unsubscribe(sid)
        srv.stop()


def test_execution_cleanup_retention_and_stale():
    st = Storage(":memory:")
    now = time.time()
    # old terminal execution -> deleted
    st.create_execution({"id": "old_done", "run_id": "r1",
                         "status": "completed", "input": {}})
    st._exec("UPDATE executions SET created_at=?, status='completed' "
             "WHERE id='old_done'", (now - 7200,))
    # old RUNNING execution, recent start -> kept (not stale yet)
    st.create_execution({"id": "young_run", "run_id": "r2",
                         "status": "running", "input": {}})
    # old running past stale threshold -> marked failed, NOT deleted
    st.create_execution({"id": "stale_run", "run_id": "r3",
                         "status": "running", "input": {}})
    st._exec("UPDATE executions SET started_at=? WHERE id='stale_run'",
             (now - 7200,))
    # recent terminal -> kept
    st.create_execution({"id": "fresh_done", "run_id": "r4",
                         "status": "completed", "input": {}})
    st._exec("UPDATE executions SET status='completed' "
             "WHERE id='fresh_done'",)

    ec = ExecutionCleanup(st, retention_s=3600, stale_s=1800,
                          interval=999, batch=10)
    out = ec.run_once()
    assert out["stale_marked"] == 1
    assert out["deleted"] >= 1
    assert st.get_execution("old_done") is None
    assert st.get_execution("young_run")["status"] == "running"
    stale = st.get_execution("stale_run")
    assert stale["status"] == "failed"
    assert "stale" in stale["error_message"]
    assert st.get_execution("fresh_done") is not None
    # second pass: nothing new to mark; stale_run survives until its
    # created_at passes retention
    out2 = ec.run_once()
    assert out2["stale_marked"] == 0
    assert st.get_execution("stale_run") is not None
    st._exec("UPDATE executions SET created_at=? WHERE id='stale_run'",
             (now - 7200,))
    assert ec.run_once()["deleted"] == 1
    assert st.get_execution("stale_run") is None


def test_execution_cleanup_batch_limit():
    st = Storage(":memory:")
    now = time.time()
    for i in range(25):
        st.create_execution({"id": f"e{i}", "run_id": "r",
                             "status": "completed", "input": {}})
    st._exec("UPDATE executions SET created_at=?, status='completed'",
             (now - 7200,))
    ec = ExecutionCleanup(st, retention_s=3600, stale_s=1800,
                          interval=999, batch=10)
    assert ec.run_once()["deleted"] == 10  # bounded per pass
    assert ec.run_once()["deleted"] == 10
    assert ec.run_once()["deleted"] == 5
