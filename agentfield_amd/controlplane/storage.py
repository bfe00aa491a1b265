"""SQLite-backed storage for the control plane.

Deliberately NOT a clone of the reference's ~120-method StorageProvider
(SURVEY.md §7 hard-part 4): this is the minimal interface the handlers
actually use, grown as handlers grew.  WAL mode + a process-wide lock keeps
it safe under the async server; an in-memory mode (":memory:") backs tests.
"""
from __future__ import annotations

import json
import sqlite3
import threading
import time
from pathlib import Path

import numpy as np


def now() -> float:
    return time.time()


_SCHEMA = """
CREATE TABLE IF NOT EXISTS agent_nodes (
  id TEXT PRIMARY KEY, team_id TEXT, base_url TEXT, version TEXT,
  deployment_type TEXT DEFAULT 'long_running', status TEXT DEFAULT 'registered',
  reasoners TEXT DEFAULT '[]', skills TEXT DEFAULT '[]',
  metadata TEXT DEFAULT '{}', did TEXT,
  registered_at REAL, last_heartbeat REAL, last_status_change REAL
);
CREATE TABLE IF NOT EXISTS executions (
  id TEXT PRIMARY KEY, run_id TEXT, parent_execution_id TEXT,
  node_id TEXT, reasoner_id TEXT, target_type TEXT DEFAULT 'reasoner',
  status TEXT DEFAULT 'pending', input TEXT, result TEXT, error_message TEXT,
  input_uri TEXT, result_uri TEXT,
  session_id TEXT, actor_id TEXT, depth INTEGER DEFAULT 0,
  created_at REAL, started_at REAL, finished_at REAL, duration_ms REAL,
  webhook_registered INTEGER DEFAULT 0
);
CREATE INDEX IF NOT EXISTS idx_exec_run ON executions(run_id);
CREATE INDEX IF NOT EXISTS idx_exec_parent ON executions(parent_execution_id);
CREATE INDEX IF NOT EXISTS idx_exec_status ON executions(status, created_at);
CREATE TABLE IF NOT EXISTS execution_webhooks (
  execution_id TEXT PRIMARY KEY, url TEXT, secret TEXT, headers TEXT,
  status TEXT DEFAULT 'pending', attempts INTEGER DEFAULT 0,
  next_attempt_at REAL, last_error TEXT, created_at REAL, delivered_at REAL,
  payload TEXT
);
CREATE TABLE IF NOT EXISTS webhook_events (
  id INTEGER PRIMARY KEY AUTOINCREMENT, execution_id TEXT, attempt INTEGER,
  status_code INTEGER, error TEXT, at REAL
);
CREATE TABLE IF NOT EXISTS memories (
  scope TEXT, scope_id TEXT, key TEXT, value TEXT, updated_at REAL,
  PRIMARY KEY (scope, scope_id, key)
);
CREATE TABLE IF NOT EXISTS vectors (
  scope TEXT, scope_id TEXT, key TEXT, embedding BLOB, metadata TEXT,
  updated_at REAL, PRIMARY KEY (scope, scope_id, key)
);
CREATE TABLE IF NOT EXISTS memory_events (
  id INTEGER PRIMARY KEY AUTOINCREMENT, scope TEXT, scope_id TEXT, key TEXT,
  op TEXT, value TEXT, at REAL
);
CREATE TABLE IF NOT EXISTS dids (
  did TEXT PRIMARY KEY, kind TEXT, subject_id TEXT, parent_did TEXT,
  public_key_b64 TEXT, document TEXT, derivation_index INTEGER, created_at REAL
);
CREATE TABLE IF NOT EXISTS vcs (
  id TEXT PRIMARY KEY, execution_id TEXT, run_id TEXT, issuer_did TEXT,
  document TEXT, created_at REAL
);
CREATE INDEX IF NOT EXISTS idx_vcs_run ON vcs(run_id);
CREATE TABLE IF NOT EXISTS workflow_runs (
  run_id TEXT PRIMARY KEY, status TEXT, root_execution_id TEXT,
  started_at REAL, finished_at REAL, metadata TEXT DEFAULT '{}'
);
CREATE TABLE IF NOT EXISTS node_actions (
  id INTEGER PRIMARY KEY AUTOINCREMENT, node_id TEXT, action TEXT,
  payload TEXT DEFAULT '{}', status TEXT DEFAULT 'pending',
  lease_expires_at REAL DEFAULT 0, created_at REAL, acked_at REAL,
  ack_status TEXT
);
CREATE INDEX IF NOT EXISTS idx_actions_node ON node_actions(node_id, status);
CREATE TABLE IF NOT EXISTS execution_notes (
  id INTEGER PRIMARY KEY AUTOINCREMENT, execution_id TEXT, author TEXT,
  note TEXT, created_at REAL
);
CREATE INDEX IF NOT EXISTS idx_notes_exec ON execution_notes(execution_id);
CREATE TABLE IF NOT EXISTS dist_locks (
  name TEXT PRIMARY KEY, owner TEXT, expires_at REAL
);
"""


class Storage:
    def __init__(self, path: str = ":memory:"):
        if path != ":memory:":
            Path(path).parent.mkdir(parents=True, exist_ok=True)
        self._db = sqlite3.connect(path, check_same_thread=False,
                                   timeout=10.0)
        self._db.row_factory = sqlite3.Row
        self._lock = threading.RLock()
        with self._lock:
            if path != ":memory:":
                self._db.execute("PRAGMA journal_mode=WAL")
                # multi-worker planes share one WAL file: writers from
                # other processes serialize on the WAL write lock; wait
                # instead of surfacing SQLITE_BUSY
                self._db.execute("PRAGMA busy_timeout=10000")
            self._db.execute("PRAGMA synchronous=NORMAL")
            self._db.executescript(_SCHEMA)
            self._db.commit()

    def _exec(self, sql: str, args=()):
        with self._lock:
            cur = self._db.execute(sql, args)
            self._db.commit()
            return cur

    def _q(self, sql: str, args=()):
        with self._lock:
            return [dict(r) for r in self._db.execute(sql, args).fetchall()]

    def _q1(self, sql: str, args=()):
        rows = self._q(sql, args)
        return rows[0] if rows else None

    # ---------------- nodes ----------------
    def upsert_node(self, node: dict) -> None:
        self._exec(
            """INSERT INTO agent_nodes
               (id, team_id, base_url, version, deployment_type, status,
                reasoners, skills, metadata, registered_at, last_heartbeat,
                last_status_change)
               VALUES (?,?,?,?,?,?,?,?,?,?,?,?)
               ON CONFLICT(id) DO UPDATE SET
                 base_url=excluded.base_url, version=excluded.version,
                 deployment_type=excluded.deployment_type,
                 reasoners=excluded.reasoners, skills=excluded.skills,
                 metadata=excluded.metadata, last_heartbeat=excluded.last_heartbeat""",
            (node["id"], node.get("team_id"), node.get("base_url"),
             node.get("version"), node.get("deployment_type", "long_running"),
             node.get("status", "registered"),
             json.dumps(node.get("reasoners", [])),
             json.dumps(node.get("skills", [])),
             json.dumps(node.get("metadata", {})), now(), now(), now()))

    def get_node(self, node_id: str) -> dict | None:
        r = self._q1("SELECT * FROM agent_nodes WHERE id=?", (node_id,))
        return self._node_row(r) if r else None

    def list_nodes(self) -> list[dict]:
        return [self._node_row(r) for r in self._q("SELECT * FROM agent_nodes")]

    def delete_node(self, node_id: str) -> None:
        self._exec("DELETE FROM agent_nodes WHERE id=?", (node_id,))

    @staticmethod
    def _node_row(r: dict) -> dict:
        r = dict(r)
        for k in ("reasoners", "skills", "metadata"):
            r[k] = json.loads(r[k] or "null") or ([] if k != "metadata" else {})
        return r

    def set_node_status(self, node_id: str, status: str) -> None:
        self._exec("UPDATE agent_nodes SET status=?, last_status_change=? WHERE id=?",
                   (status, now(), node_id))

    def set_node_metadata(self, node_id: str, metadata: dict) -> None:
        self._exec("UPDATE agent_nodes SET metadata=? WHERE id=?",
                   (json.dumps(metadata), node_id))

    def touch_heartbeat(self, node_id: str, status: str | None = None) -> None:
        if status:
            self._exec("UPDATE agent_nodes SET last_heartbeat=?, status=? WHERE id=?",
                       (now(), status, node_id))
        else:
            self._exec("UPDATE agent_nodes SET last_heartbeat=? WHERE id=?",
                       (now(), node_id))

    # ---------------- executions ----------------
    def create_execution(self, rec: dict) -> None:
        self._exec(
            """INSERT OR REPLACE INTO executions
               (id, run_id, parent_execution_id, node_id, reasoner_id,
                target_type, status, input, session_id, actor_id, depth,
                created_at, started_at, webhook_registered)
               VALUES (?,?,?,?,?,?,?,?,?,?,?,?,?,?)""",
            (rec["id"], rec.get("run_id"), rec.get("parent_execution_id"),
             rec.get("node_id"), rec.get("reasoner_id"),
             rec.get("target_type", "reasoner"), rec.get("status", "running"),
             json.dumps(rec.get("input")), rec.get("session_id"),
             rec.get("actor_id"), rec.get("depth", 0), now(), now(),
             1 if rec.get("webhook_registered") else 0))

    def get_execution(self, exec_id: str) -> dict | None:
        r = self._q1("SELECT * FROM executions WHERE id=?", (exec_id,))
        return self._exec_row(r) if r else None

    @staticmethod
    def _exec_row(r: dict) -> dict:
        r = dict(r)
        for k in ("input", "result"):
            if r.get(k):
                try:
                    r[k] = json.loads(r[k])
                except (TypeError, ValueError):
                    pass
        return r

    def update_execution_result(self, exec_id: str, status: str,
                                result=None, error: str | None = None,
                                duration_ms: float | None = None) -> bool:
        """Terminal states are write-once: a late timeout/duplicate callback
        must not overwrite a completed record.  Returns False if the row was
        already terminal."""
        cur = self._exec(
            """UPDATE executions SET status=?, result=?, error_message=?,
               finished_at=?, duration_ms=COALESCE(?, (?-started_at)*1000.0)
               WHERE id=? AND status NOT IN
                 ('completed','failed','timeout','cancelled')""",
            (status, json.dumps(result) if result is not None else None,
             error, now(), duration_ms, now(), exec_id))
        return cur.rowcount > 0

    def upsert_workflow_event(self, ev: dict) -> bool:
        """SDK-pushed nested-call events (A.3): create-if-missing.  Returns
        True when this ingestion OWNS the record's lifecycle (it created it,
        i.e. an in-process nested call): control-plane-dispatched executions
        (exec_* ids created by prepare_execution) are finalized by the agent
        status callback, never by trace events — otherwise the trace event
        races the callback and swallows its result/webhook."""
        if not self._q1("SELECT id FROM executions WHERE id=?", (ev["execution_id"],)):
            self.create_execution({
                "id": ev["execution_id"], "run_id": ev.get("run_id") or ev.get("workflow_id"),
                "parent_execution_id": ev.get("parent_execution_id"),
                "node_id": ev.get("agent_node_id"), "reasoner_id": ev.get("reasoner_id"),
                "status": ev.get("status", "running"), "input": ev.get("input_data"),
            })
            return True
        return not str(ev["execution_id"]).startswith("exec_")

    def executions_by_run(self, run_id: str) -> list[dict]:
        return [self._exec_row(r) for r in self._q(
            "SELECT * FROM executions WHERE run_id=? ORDER BY created_at", (run_id,))]

    def list_executions(self, limit: int = 100, node_id: str | None = None,
                        status: str | None = None) -> list[dict]:
        sql = "SELECT * FROM executions"
        conds, args = [], []
        if node_id:
            conds.append("node_id=?")
            args.append(node_id)
        if status:
            conds.append("status=?")
            args.append(status)
        if conds:
            sql += " WHERE " + " AND ".join(conds)
        sql += " ORDER BY created_at DESC LIMIT ?"
        args.append(limit)
        return [self._exec_row(r) for r in self._q(sql, tuple(args))]

    def batch_status(self, ids: list[str]) -> dict[str, dict]:
        if not ids:
            return {}
        marks = ",".join("?" for _ in ids)
        rows = self._q(f"SELECT * FROM executions WHERE id IN ({marks})", tuple(ids))
        return {r["id"]: self._exec_row(r) for r in rows}

    def mark_stale_running(self, older_than_s: float) -> int:
        cur = self._exec(
            """UPDATE executions SET status='failed',
               error_message='marked stale by cleanup', finished_at=?
               WHERE status='running' AND started_at < ?""",
            (now(), now() - older_than_s))
        return cur.rowcount

    def delete_old_executions(self, older_than_s: float, batch: int = 100) -> int:
        cur = self._exec(
            """DELETE FROM executions WHERE id IN (
                 SELECT id FROM executions WHERE created_at < ? AND
                 status IN ('completed','failed','timeout','cancelled') LIMIT ?)""",
            (now() - older_than_s, batch))
        return cur.rowcount

    # ---------------- webhooks ----------------
    def register_webhook(self, execution_id: str, url: str, secret: str = "",
                         headers: dict | None = None) -> None:
        self._exec(
            """INSERT OR REPLACE INTO execution_webhooks
               (execution_id, url, secret, headers, status, attempts,
                next_attempt_at, created_at) VALUES (?,?,?,?, 'pending', 0, ?, ?)""",
            (execution_id, url, secret, json.dumps(headers or {}), now(), now()))

    def get_webhook(self, execution_id: str) -> dict | None:
        r = self._q1("SELECT * FROM execution_webhooks WHERE execution_id=?",
                     (execution_id,))
        if r:
            r = dict(r)
            r["headers"] = json.loads(r["headers"] or "{}")
        return r

    def stage_webhook_payload(self, execution_id: str, payload: dict) -> None:
        self._exec("UPDATE execution_webhooks SET payload=? WHERE execution_id=?",
                   (json.dumps(payload), execution_id))

    def try_mark_webhook_inflight(self, execution_id: str,
                                  stale_s: float = 60.0) -> bool:
        """Idempotent in-flight claim (webhook_dispatcher.go:196).  The
        claim stamps next_attempt_at = now + stale_s so a claim whose
        worker died (process crash, orphaned queue item) becomes
        reclaimable after the window instead of wedging forever."""
        t = now()
        cur = self._exec(
            """UPDATE execution_webhooks
               SET status='inflight', next_attempt_at=?
               WHERE execution_id=? AND
                     (status IN ('pending','retry')
                      OR (status='inflight' AND next_attempt_at <= ?))""",
            (t + stale_s, execution_id, t))
        return cur.rowcount > 0

    def release_webhook_claim(self, execution_id: str) -> None:
        """Undo an in-flight claim that cannot proceed (e.g. the
        execution has not completed yet, so no payload is staged)."""
        self._exec(
            """UPDATE execution_webhooks SET status='pending',
               next_attempt_at=? WHERE execution_id=? AND status='inflight'""",
            (now(), execution_id))

    def webhook_attempted(self, execution_id: str, ok: bool, status_code: int,
                          error: str | None, backoff_s: float,
                          max_attempts: int) -> None:
        wh = self.get_webhook(execution_id)
        if not wh:
            return
        attempts = wh["attempts"] + 1
        if ok:
            st, nxt = "delivered", None
        elif attempts >= max_attempts:
            st, nxt = "failed", None
        else:
            st, nxt = "retry", now() + backoff_s
        self._exec(
            """UPDATE execution_webhooks SET status=?, attempts=?,
               next_attempt_at=?, last_error=?, delivered_at=? WHERE execution_id=?""",
            (st, attempts, nxt, error, now() if ok else None, execution_id))
        self._exec(
            "INSERT INTO webhook_events (execution_id, attempt, status_code, error, at)"
            " VALUES (?,?,?,?,?)", (execution_id, attempts, status_code, error, now()))

    def due_webhooks(self, batch: int = 64) -> list[dict]:
        # payload IS NOT NULL: a pending webhook of a still-RUNNING
        # execution is not deliverable — claiming it pre-completion
        # orphaned it forever (notify() skips inflight rows; measured:
        # exactly one poller batch of 64 wedged under a 384-burst).
        # Stale inflight claims (worker death) become due again via the
        # claim's next_attempt_at stamp.
        rows = self._q(
            """SELECT * FROM execution_webhooks
               WHERE payload IS NOT NULL AND next_attempt_at <= ?
                 AND status IN ('pending','retry','inflight')
               ORDER BY next_attempt_at LIMIT ?""", (now(), batch))
        for r in rows:
            r["headers"] = json.loads(r["headers"] or "{}")
        return rows

    def webhook_history(self, execution_id: str) -> list[dict]:
        return self._q("SELECT * FROM webhook_events WHERE execution_id=? ORDER BY at",
                       (execution_id,))

    # ---------------- memory (KV) ----------------
    def memory_set(self, scope: str, scope_id: str, key: str, value) -> None:
        self._exec(
            """INSERT INTO memories (scope, scope_id, key, value, updated_at)
               VALUES (?,?,?,?,?) ON CONFLICT(scope, scope_id, key)
               DO UPDATE SET value=excluded.value, updated_at=excluded.updated_at""",
            (scope, scope_id, key, json.dumps(value), now()))
        self._exec(
            "INSERT INTO memory_events (scope, scope_id, key, op, value, at)"
            " VALUES (?,?,?,?,?,?)", (scope, scope_id, key, "set",
                                      json.dumps(value), now()))

    def memory_get(self, scope: str, scope_id: str, key: str):
        r = self._q1("SELECT value FROM memories WHERE scope=? AND scope_id=? AND key=?",
                     (scope, scope_id, key))
        return json.loads(r["value"]) if r else None

    def memory_delete(self, scope: str, scope_id: str, key: str) -> bool:
        cur = self._exec("DELETE FROM memories WHERE scope=? AND scope_id=? AND key=?",
                         (scope, scope_id, key))
        if cur.rowcount:
            self._exec(
                "INSERT INTO memory_events (scope, scope_id, key, op, value, at)"
                " VALUES (?,?,?,?,?,?)", (scope, scope_id, key, "delete", None, now()))
        return cur.rowcount > 0

    def memory_list(self, scope: str, scope_id: str, prefix: str = "") -> list[str]:
        rows = self._q(
            "SELECT key FROM memories WHERE scope=? AND scope_id=? AND key LIKE ?"
            " ORDER BY key", (scope, scope_id, prefix + "%"))
        return [r["key"] for r in rows]

    def memory_events_since(self, since: float, scope: str | None = None,
                            limit: int = 200) -> list[dict]:
        if scope:
            return self._q(
                "SELECT * FROM memory_events WHERE at>? AND scope=? ORDER BY at LIMIT ?",
                (since, scope, limit))
        return self._q("SELECT * FROM memory_events WHERE at>? ORDER BY at LIMIT ?",
                       (since, limit))

    # ---------------- vector memory ----------------
    def vector_set(self, scope: str, scope_id: str, key: str,
                   embedding: list[float], metadata: dict | None = None) -> None:
        emb = np.asarray(embedding, dtype=np.float32).tobytes()
        self._exec(
            """INSERT INTO vectors (scope, scope_id, key, embedding, metadata, updated_at)
               VALUES (?,?,?,?,?,?) ON CONFLICT(scope, scope_id, key) DO UPDATE SET
               embedding=excluded.embedding, metadata=excluded.metadata,
               updated_at=excluded.updated_at""",
            (scope, scope_id, key, emb, json.dumps(metadata or {}), now()))

    def vector_delete(self, scope: str, scope_id: str, key: str) -> bool:
        cur = self._exec("DELETE FROM vectors WHERE scope=? AND scope_id=? AND key=?",
                         (scope, scope_id, key))
        return cur.rowcount > 0

    def vector_search(self, scope: str, scope_id: str, query: list[float],
                      top_k: int = 5, metric: str = "cosine",
                      filters: dict | None = None) -> list[dict]:
        rows = self._q("SELECT * FROM vectors WHERE scope=? AND scope_id=?",
                       (scope, scope_id))
        if not rows:
            return []
        q = np.asarray(query, dtype=np.float32)
        out = []
        for r in rows:
            meta = json.loads(r["metadata"] or "{}")
            if filters and any(meta.get(k) != v for k, v in filters.items()):
                continue
            e = np.frombuffer(r["embedding"], dtype=np.float32)
            if e.shape != q.shape:
                continue
            if metric == "dot":
                score = float(e @ q)
            elif metric == "l2":
                score = -float(np.linalg.norm(e - q))
            else:
                denom = float(np.linalg.norm(e) * np.linalg.norm(q)) or 1e-12
                score = float(e @ q) / denom
            out.append({"key": r["key"], "score": score, "metadata": meta})
        out.sort(key=lambda x: -x["score"])
        return out[:top_k]

    # ---------------- DID / VC ----------------
    def put_did(self, rec: dict) -> None:
        self._exec(
            """INSERT OR REPLACE INTO dids
               (did, kind, subject_id, parent_did, public_key_b64, document,
                derivation_index, created_at) VALUES (?,?,?,?,?,?,?,?)""",
            (rec["did"], rec.get("kind"), rec.get("subject_id"),
             rec.get("parent_did"), rec.get("public_key_b64"),
             json.dumps(rec.get("document", {})), rec.get("derivation_index", 0),
             now()))

    def get_did(self, did: str) -> dict | None:
        r = self._q1("SELECT * FROM dids WHERE did=?", (did,))
        if r:
            r = dict(r)
            r["document"] = json.loads(r["document"] or "{}")
        return r

    def did_for_subject(self, kind: str, subject_id: str) -> dict | None:
        r = self._q1("SELECT * FROM dids WHERE kind=? AND subject_id=?",
                     (kind, subject_id))
        if r:
            r = dict(r)
            r["document"] = json.loads(r["document"] or "{}")
        return r

    def list_dids(self, kind: str | None = None) -> list[dict]:
        rows = self._q("SELECT * FROM dids WHERE kind=?", (kind,)) if kind \
            else self._q("SELECT * FROM dids")
        for r in rows:
            r["document"] = json.loads(r["document"] or "{}")
        return rows

    def max_derivation_index(self) -> int:
        r = self._q1("SELECT MAX(derivation_index) AS m FROM dids")
        return int(r["m"] or 0)

    def put_vc(self, vc_id: str, execution_id: str, run_id: str,
               issuer_did: str, document: dict) -> None:
        self._exec(
            "INSERT OR REPLACE INTO vcs (id, execution_id, run_id, issuer_did,"
            " document, created_at) VALUES (?,?,?,?,?,?)",
            (vc_id, execution_id, run_id, issuer_did, json.dumps(document), now()))

    def get_vc(self, vc_id: str) -> dict | None:
        r = self._q1("SELECT * FROM vcs WHERE id=?", (vc_id,))
        if r:
            r = dict(r)
            r["document"] = json.loads(r["document"])
        return r

    def vc_for_execution(self, execution_id: str) -> dict | None:
        r = self._q1("SELECT * FROM vcs WHERE execution_id=?", (execution_id,))
        if r:
            r = dict(r)
            r["document"] = json.loads(r["document"])
        return r

    def vcs_for_run(self, run_id: str) -> list[dict]:
        rows = self._q("SELECT * FROM vcs WHERE run_id=? ORDER BY created_at",
                       (run_id,))
        for r in rows:
            r["document"] = json.loads(r["document"])
        return rows

    # ---------------- workflow runs (v2) ----------------
    # ------------------------------------------- node action lease queue
    # (reference: nodes actions claim/ack lease protocol, server.go:865-867)
    def enqueue_action(self, node_id: str, action: str,
                       payload: dict | None = None) -> int:
        with self._lock:
            cur = self._db.execute(
                "INSERT INTO node_actions (node_id, action, payload, "
                "created_at) VALUES (?,?,?,?)",
                (node_id, action, json.dumps(payload or {}), now()))
            self._db.commit()
            return int(cur.lastrowid)

    def claim_actions(self, node_id: str, lease_s: float = 30.0,
                      limit: int = 16) -> list[dict]:
        """Claim pending (or lease-expired claimed) actions for a node.
        Claims are leases: unacked actions become claimable again after
        lease_s, so a crashed agent never strands an action."""
        t = now()
        with self._lock:
            rows = self._q(
                "SELECT * FROM node_actions WHERE node_id=? AND "
                "(status='pending' OR (status='claimed' AND "
                "lease_expires_at<?)) ORDER BY id LIMIT ?",
                (node_id, t, limit))
            for r in rows:
                self._db.execute(
                    "UPDATE node_actions SET status='claimed', "
                    "lease_expires_at=? WHERE id=?",
                    (t + lease_s, r["id"]))
            self._db.commit()
        for r in rows:
            r["payload"] = json.loads(r["payload"] or "{}")
            r["status"] = "claimed"
        return rows

    def ack_action(self, action_id: int, status: str = "done") -> bool:
        with self._lock:
            cur = self._db.execute(
                "UPDATE node_actions SET status='acked', acked_at=?, "
                "ack_status=? WHERE id=? AND status='claimed'",
                (now(), status, action_id))
            self._db.commit()
            return cur.rowcount > 0

    def pending_actions(self, node_id: str) -> int:
        r = self._q("SELECT COUNT(*) AS n FROM node_actions WHERE node_id=? "
                    "AND status!='acked'", (node_id,))
        return int(r[0]["n"])

    # ------------------------------------------------- execution notes
    def add_note(self, execution_id: str, note: str,
                 author: str = "") -> int:
        with self._lock:
            cur = self._db.execute(
                "INSERT INTO execution_notes (execution_id, author, note, "
                "created_at) VALUES (?,?,?,?)",
                (execution_id, author, note, now()))
            self._db.commit()
            return int(cur.lastrowid)

    def notes_for(self, execution_id: str) -> list[dict]:
        return self._q("SELECT * FROM execution_notes WHERE execution_id=? "
                       "ORDER BY id", (execution_id,))

    def upsert_run(self, run_id: str, status: str,
                   root_execution_id: str | None = None) -> None:
        self._exec(
            """INSERT INTO workflow_runs (run_id, status, root_execution_id, started_at)
               VALUES (?,?,?,?) ON CONFLICT(run_id) DO UPDATE SET status=excluded.status,
               finished_at=CASE WHEN excluded.status IN
                 ('completed','failed','timeout','cancelled') THEN ? ELSE NULL END""",
            (run_id, status, root_execution_id, now(), now()))

    # ---------------- hot-path composites (single lock + single commit) --
    # The execute hot loop is 2-4 SQL round trips per
    # request when issued one-by-one; grouping them under one lock/commit
    # mirrors the reference's serialized completion queue batching
    # (execute.go:1404-1439) without a worker hop.
    def create_execution_full(self, rec: dict, run_id: str,
                              webhook: dict | None = None) -> None:
        with self._lock:
            self._db.execute(
                """INSERT OR REPLACE INTO executions
                   (id, run_id, parent_execution_id, node_id, reasoner_id,
                    target_type, status, input, session_id, actor_id, depth,
                    created_at, started_at, webhook_registered)
                   VALUES (?,?,?,?,?,?,?,?,?,?,?,?,?,?)""",
                (rec["id"], rec.get("run_id"), rec.get("parent_execution_id"),
                 rec.get("node_id"), rec.get("reasoner_id"),
                 rec.get("target_type", "reasoner"),
                 rec.get("status", "running"),
                 json.dumps(rec.get("input")), rec.get("session_id"),
                 rec.get("actor_id"), rec.get("depth", 0), now(), now(),
                 1 if rec.get("webhook_registered") else 0))
            self._db.execute(
                """INSERT INTO workflow_runs
                   (run_id, status, root_execution_id, started_at)
                   VALUES (?,?,?,?) ON CONFLICT(run_id) DO UPDATE SET
                   status=excluded.status, finished_at=NULL""",
                (run_id, "running", rec["id"], now()))
            if webhook:
                self._db.execute(
                    """INSERT OR REPLACE INTO execution_webhooks
                       (execution_id, url, secret, headers, status, attempts,
                        next_attempt_at, created_at)
                       VALUES (?,?,?,?, 'pending', 0, ?, ?)""",
                    (rec["id"], webhook["url"], webhook.get("secret", ""),
                     json.dumps(webhook.get("headers") or {}), now(), now()))
            self._db.commit()

    def finalize_execution(self, exec_id: str, status: str, result=None,
                           error: str | None = None,
                           duration_ms: float | None = None,
                           aggregate=None) -> tuple[bool, dict | None]:
        """Terminal write + row fetch + run-status roll-up in ONE
        lock/commit.  `aggregate` folds sibling statuses into the run
        status (workflow.aggregate_status).  Returns (applied, record);
        applied=False when the row was already terminal (write-once)."""
        with self._lock:
            cur = self._db.execute(
                """UPDATE executions SET status=?, result=?, error_message=?,
                   finished_at=?, duration_ms=COALESCE(?, (?-started_at)*1000.0)
                   WHERE id=? AND status NOT IN
                     ('completed','failed','timeout','cancelled')""",
                (status, json.dumps(result) if result is not None else None,
                 error, now(), duration_ms, now(), exec_id))
            applied = cur.rowcount > 0
            r = self._db.execute("SELECT * FROM executions WHERE id=?",
                                 (exec_id,)).fetchone()
            rec = self._exec_row(dict(r)) if r else None
            if applied and rec and rec.get("run_id") and aggregate:
                sibs = self._db.execute(
                    "SELECT status FROM executions WHERE run_id=?",
                    (rec["run_id"],)).fetchall()
                run_status = aggregate([s["status"] for s in sibs])
                self._db.execute(
                    """INSERT INTO workflow_runs (run_id, status, started_at)
                       VALUES (?,?,?) ON CONFLICT(run_id) DO UPDATE SET
                       status=excluded.status,
                       finished_at=CASE WHEN excluded.status IN
                         ('completed','failed','timeout','cancelled')
                         THEN ? ELSE NULL END""",
                    (rec["run_id"], run_status, now(), now()))
            if applied:
                self._db.commit()
        return applied, rec

    def get_run(self, run_id: str) -> dict | None:
        return self._q1("SELECT * FROM workflow_runs WHERE run_id=?", (run_id,))

    def list_runs(self, limit: int = 50) -> list[dict]:
        return self._q("SELECT * FROM workflow_runs ORDER BY started_at DESC LIMIT ?",
                       (limit,))

    # ---------------------------------------------------- distributed locks
    # Lease-based mutual exclusion across processes sharing the DB
    # (reference: storage locks with expiry — local.go's lock table).
    # acquire is a single atomic UPSERT guarded on expiry/ownership, so
    # concurrent workers over the same WAL cannot both win.
    def lock_acquire(self, name: str, owner: str, ttl_s: float = 30.0) -> bool:
        now = time.time()
        with self._lock:
            cur = self._db.execute(
                """INSERT INTO dist_locks(name, owner, expires_at)
                   VALUES(?,?,?)
                   ON CONFLICT(name) DO UPDATE
                     SET owner=excluded.owner, expires_at=excluded.expires_at
                     WHERE dist_locks.expires_at < ?
                        OR dist_locks.owner = excluded.owner""",
                (name, owner, now + ttl_s, now))
            self._db.commit()
            if cur.rowcount > 0:
                return True
        return False

    def lock_release(self, name: str, owner: str) -> bool:
        with self._lock:
            cur = self._db.execute(
                "DELETE FROM dist_locks WHERE name=? AND owner=?",
                (name, owner))
            self._db.commit()
            return cur.rowcount > 0

    def lock_refresh(self, name: str, owner: str, ttl_s: float = 30.0) -> bool:
        with self._lock:
            cur = self._db.execute(
                "UPDATE dist_locks SET expires_at=? WHERE name=? AND owner=?",
                (time.time() + ttl_s, name, owner))
            self._db.commit()
            return cur.rowcount > 0

    def lock_holder(self, name: str) -> str | None:
        row = self._q1("SELECT owner, expires_at FROM dist_locks WHERE name=?",
                       (name,))
        if row and row["expires_at"] > time.time():
            return row["owner"]
        return None

    def close(self):
        with self._lock:
            self._db.close()
