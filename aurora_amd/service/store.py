"""Persistence layer: SQLite with enforced org scoping.

The reference runs Postgres with FORCE ROW LEVEL SECURITY and per-connection
``myapp.current_org_id`` session vars (reference: utils/db/connection_pool.py:185,
AGENTS.md:145-153).  This environment has no Postgres server, so the same
tenant-isolation contract is enforced structurally: every tenant table
carries org_id, and the only query surface is OrgStore, which is *bound* to
an org at construction and injects the org predicate into every statement.
AdminStore (the get_admin_connection analog) is separate and explicit.

Schema mirrors the reference's tables: incidents,
alerts, chat_sessions, llm_context_history, incident_thoughts,
execution_steps (waterfall), citations, suggestions, findings, postmortems,
artifacts, actions, lifecycle_events, llm_usage, command_policies,
audit_events, kb documents/chunks, connector tokens, graph nodes/edges,
task queue.
"""
from __future__ import annotations

import json
import os
import sqlite3
import threading
import time
import uuid
from typing import Any, Dict, Iterable, List, Optional

_SCHEMA = """
PRAGMA journal_mode=WAL;
CREATE TABLE IF NOT EXISTS orgs (
  id TEXT PRIMARY KEY, name TEXT NOT NULL, created_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS users (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, email TEXT UNIQUE NOT NULL,
  role TEXT NOT NULL DEFAULT 'member', created_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS incidents (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, user_id TEXT,
  title TEXT NOT NULL, source TEXT NOT NULL, severity TEXT,
  status TEXT NOT NULL DEFAULT 'open',
  aurora_status TEXT NOT NULL DEFAULT 'pending',
  payload TEXT, summary TEXT, root_cause TEXT,
  aurora_chat_session_id TEXT, rca_task_id TEXT,
  created_at REAL NOT NULL, updated_at REAL NOT NULL,
  investigation_started_at REAL, resolved_at REAL);
CREATE INDEX IF NOT EXISTS idx_incidents_org ON incidents(org_id, created_at);
CREATE TABLE IF NOT EXISTS alerts (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, incident_id TEXT,
  source TEXT NOT NULL, fingerprint TEXT, payload TEXT, created_at REAL NOT NULL);
CREATE INDEX IF NOT EXISTS idx_alerts_org_created ON alerts(org_id, created_at);
CREATE TABLE IF NOT EXISTS chat_sessions (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, user_id TEXT,
  messages TEXT NOT NULL DEFAULT '[]', is_background INTEGER NOT NULL DEFAULT 0,
  incident_id TEXT, created_at REAL NOT NULL, updated_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS llm_context_history (
  session_id TEXT PRIMARY KEY, org_id TEXT NOT NULL,
  messages TEXT NOT NULL, updated_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS incident_thoughts (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  incident_id TEXT NOT NULL, content TEXT NOT NULL, ts REAL NOT NULL);
CREATE INDEX IF NOT EXISTS idx_thoughts_org_incident
  ON incident_thoughts(org_id, incident_id);
CREATE TABLE IF NOT EXISTS execution_steps (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, incident_id TEXT,
  session_id TEXT, tool TEXT NOT NULL, args TEXT, output_excerpt TEXT,
  status TEXT NOT NULL, error TEXT, started_at REAL NOT NULL, finished_at REAL);
CREATE INDEX IF NOT EXISTS idx_steps_incident ON execution_steps(org_id, incident_id);
CREATE TABLE IF NOT EXISTS incident_citations (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  incident_id TEXT NOT NULL, claim TEXT NOT NULL, source_tool TEXT,
  evidence TEXT, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS incident_suggestions (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  incident_id TEXT NOT NULL, title TEXT NOT NULL, command TEXT,
  rationale TEXT, safe INTEGER NOT NULL DEFAULT 0, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS findings (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, incident_id TEXT NOT NULL,
  role TEXT NOT NULL, wave INTEGER NOT NULL DEFAULT 0,
  status TEXT NOT NULL DEFAULT 'pending', content TEXT, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS postmortems (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, incident_id TEXT NOT NULL,
  content TEXT NOT NULL, created_at REAL NOT NULL, updated_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS artifacts (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, session_id TEXT,
  name TEXT NOT NULL, content TEXT NOT NULL, updated_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS actions (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, name TEXT NOT NULL,
  trigger TEXT NOT NULL, config TEXT, enabled INTEGER NOT NULL DEFAULT 1,
  schedule TEXT, last_run REAL, created_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS lifecycle_events (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  incident_id TEXT NOT NULL, event TEXT NOT NULL, data TEXT, ts REAL NOT NULL);
CREATE INDEX IF NOT EXISTS idx_lifecycle_org_incident
  ON lifecycle_events(org_id, incident_id);
CREATE TABLE IF NOT EXISTS llm_usage (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  model TEXT NOT NULL, prompt_tokens INTEGER NOT NULL,
  completion_tokens INTEGER NOT NULL, response_ms REAL, purpose TEXT,
  session_id TEXT, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS command_policies (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, pattern TEXT NOT NULL,
  action TEXT NOT NULL CHECK(action IN ('allow','deny')),
  created_by TEXT, created_at REAL NOT NULL);
CREATE TABLE IF NOT EXISTS audit_events (
  id INTEGER PRIMARY KEY AUTOINCREMENT, org_id TEXT NOT NULL,
  kind TEXT NOT NULL, detail TEXT, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS kb_documents (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, user_id TEXT,
  name TEXT NOT NULL, content TEXT, ts REAL NOT NULL);
CREATE TABLE IF NOT EXISTS kb_chunks (
  id TEXT PRIMARY KEY, org_id TEXT NOT NULL, doc_id TEXT NOT NULL,
  seq INTEGER NOT NULL, text TEXT NOT NULL);
CREATE TABLE IF NOT EXISTS connector_tokens (
  org_id TEXT NOT NULL, user_id TEXT NOT NULL, provider TEXT NOT NULL,
  data TEXT NOT NULL, updated_at REAL NOT NULL,
  PRIMARY KEY (org_id, user_id, provider));
CREATE TABLE IF NOT EXISTS org_tool_permissions (
  org_id TEXT NOT NULL, tool TEXT NOT NULL, allowed INTEGER NOT NULL,
  PRIMARY KEY (org_id, tool));
CREATE TABLE IF NOT EXISTS graph_nodes (
  org_id TEXT NOT NULL, id TEXT NOT NULL, kind TEXT NOT NULL,
  props TEXT, updated_at REAL NOT NULL, PRIMARY KEY (org_id, id));
CREATE TABLE IF NOT EXISTS graph_edges (
  org_id TEXT NOT NULL, src TEXT NOT NULL, dst TEXT NOT NULL,
  kind TEXT NOT NULL, props TEXT, updated_at REAL NOT NULL,
  PRIMARY KEY (org_id, src, dst, kind));
CREATE TABLE IF NOT EXISTS task_queue (
  id TEXT PRIMARY KEY, name TEXT NOT NULL, org_id TEXT, args TEXT,
  status TEXT NOT NULL DEFAULT 'pending', dedup_key TEXT UNIQUE,
  eta REAL NOT NULL, attempts INTEGER NOT NULL DEFAULT 0,
  max_attempts INTEGER NOT NULL DEFAULT 3, result TEXT,
  created_at REAL NOT NULL, updated_at REAL NOT NULL);
CREATE INDEX IF NOT EXISTS idx_tasks_status ON task_queue(status, eta);
"""

# tables whose rows are tenant-scoped (org predicate is mandatory)
TENANT_TABLES = {
    "incidents", "alerts", "chat_sessions", "llm_context_history",
    "incident_thoughts", "execution_steps", "incident_citations",
    "incident_suggestions", "findings", "postmortems", "artifacts", "actions",
    "lifecycle_events", "llm_usage", "command_policies", "audit_events",
    "kb_documents", "kb_chunks", "connector_tokens", "org_tool_permissions",
    "graph_nodes", "graph_edges", "users",
}


class Database:
    """Process-wide SQLite handle (thread-safe via a lock; WAL mode)."""

    def __init__(self, path: str = ":memory:"):
        self.path = path
        self._lock = threading.RLock()
        self._conn = sqlite3.connect(path, check_same_thread=False)
        self._conn.row_factory = sqlite3.Row
        with self._lock:
            self._conn.executescript(_SCHEMA)
            self._conn.commit()

    def execute(self, sql: str, params: Iterable = ()) -> sqlite3.Cursor:
        with self._lock:
            cur = self._conn.execute(sql, tuple(params))
            self._conn.commit()
            return cur

    def query(self, sql: str, params: Iterable = ()) -> List[sqlite3.Row]:
        with self._lock:
            return self._conn.execute(sql, tuple(params)).fetchall()

    def close(self) -> None:
        with self._lock:
            self._conn.close()


def _now() -> float:
    return time.time()


def new_id(prefix: str = "") -> str:
    return (prefix + "-" if prefix else "") + uuid.uuid4().hex[:20]


def _row(r: Optional[sqlite3.Row]) -> Optional[Dict[str, Any]]:
    return dict(r) if r is not None else None


class OrgStore:
    """All tenant data access, bound to one org (RLS analog)."""

    def __init__(self, db: Database, org_id: str):
        assert org_id, "OrgStore requires an org id"
        self.db = db
        self.org_id = org_id

    # ---- generic helpers (org predicate injected) ----
    def _insert(self, table: str, row: Dict[str, Any]) -> None:
        assert table in TENANT_TABLES, table
        row = dict(row, org_id=self.org_id)
        cols = ",".join(row)
        ph = ",".join("?" for _ in row)
        self.db.execute(f"INSERT OR REPLACE INTO {table} ({cols}) VALUES ({ph})",
                        list(row.values()))

    def _select(self, table: str, where: str = "1=1", params: Iterable = (),
                order: str = "", limit: int = 0) -> List[Dict[str, Any]]:
        assert table in TENANT_TABLES, table
        sql = f"SELECT * FROM {table} WHERE org_id=? AND ({where})"
        if order:
            sql += f" ORDER BY {order}"
        if limit:
            sql += f" LIMIT {limit}"
        return [dict(r) for r in self.db.query(sql, [self.org_id, *params])]

    def _update(self, table: str, sets: Dict[str, Any], where: str,
                params: Iterable = ()) -> int:
        assert table in TENANT_TABLES, table
        assign = ",".join(f"{k}=?" for k in sets)
        cur = self.db.execute(
            f"UPDATE {table} SET {assign} WHERE org_id=? AND ({where})",
            [*sets.values(), self.org_id, *params])
        return cur.rowcount

    def _delete(self, table: str, where: str, params: Iterable = ()) -> int:
        assert table in TENANT_TABLES, table
        cur = self.db.execute(
            f"DELETE FROM {table} WHERE org_id=? AND ({where})",
            [self.org_id, *params])
        return cur.rowcount

    # ---- incidents ----
    def create_incident(self, title: str, source: str, payload: Optional[dict] = None,
                        user_id: Optional[str] = None, severity: Optional[str] = None,
                        incident_id: Optional[str] = None) -> str:
        iid = incident_id or new_id("inc")
        t = _now()
        self._insert("incidents", dict(
            id=iid, user_id=user_id, title=title, source=source,
            severity=severity, payload=json.dumps(payload or {}),
            created_at=t, updated_at=t))
        return iid

    def get_incident(self, incident_id: str) -> Optional[Dict[str, Any]]:
        rows = self._select("incidents", "id=?", [incident_id])
        return rows[0] if rows else None

    def list_incidents(self, status: Optional[str] = None, limit: int = 100) -> List[Dict[str, Any]]:
        if status:
            return self._select("incidents", "status=?", [status],
                                order="created_at DESC", limit=limit)
        return self._select("incidents", order="created_at DESC", limit=limit)

    def update_incident(self, incident_id: str, **sets) -> int:
        sets["updated_at"] = _now()
        if "payload" in sets and isinstance(sets["payload"], dict):
            sets["payload"] = json.dumps(sets["payload"])
        return self._update("incidents", sets, "id=?", [incident_id])

    def touch_incident(self, incident_id: str) -> None:
        """Heartbeat: the liveness signal the stale-investigation reaper
        checks (reference: task.py:2350 _is_task_dead via updated_at)."""
        self._update("incidents", {"updated_at": _now()}, "id=?", [incident_id])

    # ---- alerts / correlation ----
    def add_alert(self, source: str, payload: dict, fingerprint: str = "",
                  incident_id: Optional[str] = None) -> str:
        aid = new_id("alr")
        self._insert("alerts", dict(id=aid, incident_id=incident_id, source=source,
                                    fingerprint=fingerprint,
                                    payload=json.dumps(payload), created_at=_now()))
        return aid

    def recent_alerts(self, window_s: float = 900.0) -> List[Dict[str, Any]]:
        return self._select("alerts", "created_at > ?", [_now() - window_s],
                            order="created_at DESC", limit=200)

    # ---- chat sessions / context ----
    def create_session(self, user_id: Optional[str] = None, is_background: bool = False,
                       incident_id: Optional[str] = None,
                       session_id: Optional[str] = None) -> str:
        sid = session_id or new_id("sess")
        t = _now()
        self._insert("chat_sessions", dict(
            id=sid, user_id=user_id, is_background=int(is_background),
            incident_id=incident_id, created_at=t, updated_at=t))
        return sid

    def get_session(self, session_id: str) -> Optional[Dict[str, Any]]:
        rows = self._select("chat_sessions", "id=?", [session_id])
        return rows[0] if rows else None

    def append_session_messages(self, session_id: str, msgs: List[dict]) -> None:
        # append-only UI transcript (reference: workflow.py:1818)
        row = self.get_session(session_id)
        if row is None:
            return
        cur = json.loads(row["messages"])
        cur.extend(msgs)
        self._update("chat_sessions",
                     {"messages": json.dumps(cur), "updated_at": _now()},
                     "id=?", [session_id])

    def save_context_history(self, session_id: str, messages: List[dict]) -> None:
        # LLM-facing context, separate from the UI transcript
        # (reference: llm_context_manager.py:158-165)
        self._insert("llm_context_history", dict(
            session_id=session_id, messages=json.dumps(messages), updated_at=_now()))

    def load_context_history(self, session_id: str) -> List[dict]:
        rows = self._select("llm_context_history", "session_id=?", [session_id])
        return json.loads(rows[0]["messages"]) if rows else []

    # ---- thoughts / waterfall / citations / suggestions / findings ----
    def add_thought(self, incident_id: str, content: str) -> None:
        self._insert("incident_thoughts", dict(
            incident_id=incident_id, content=content, ts=_now()))
        self.touch_incident(incident_id)

    def thoughts(self, incident_id: str) -> List[Dict[str, Any]]:
        return self._select("incident_thoughts", "incident_id=?", [incident_id], order="ts")

    def step_start(self, tool: str, args: str, incident_id: Optional[str],
                   session_id: Optional[str]) -> str:
        sid = new_id("step")
        self._insert("execution_steps", dict(
            id=sid, incident_id=incident_id, session_id=session_id, tool=tool,
            args=args[:2000], status="running", started_at=_now()))
        return sid

    def step_end(self, step_id: str, status: str, output_excerpt: str = "",
                 error: Optional[str] = None) -> None:
        self._update("execution_steps",
                     {"status": status, "output_excerpt": output_excerpt[:4000],
                      "error": error, "finished_at": _now()}, "id=?", [step_id])

    def steps(self, incident_id: str) -> List[Dict[str, Any]]:
        return self._select("execution_steps", "incident_id=?", [incident_id],
                            order="started_at")

    def add_citation(self, incident_id: str, claim: str, source_tool: str,
                     evidence: str) -> None:
        self._insert("incident_citations", dict(
            incident_id=incident_id, claim=claim, source_tool=source_tool,
            evidence=evidence[:2000], ts=_now()))

    def citations(self, incident_id: str) -> List[Dict[str, Any]]:
        return self._select("incident_citations", "incident_id=?", [incident_id], order="ts")

    def add_suggestion(self, incident_id: str, title: str, command: str,
                       rationale: str, safe: bool) -> None:
        self._insert("incident_suggestions", dict(
            incident_id=incident_id, title=title, command=command,
            rationale=rationale, safe=int(safe), ts=_now()))

    def suggestions(self, incident_id: str) -> List[Dict[str, Any]]:
        return self._select("incident_suggestions", "incident_id=?", [incident_id], order="ts")

    def upsert_finding(self, incident_id: str, role: str, wave: int,
                       status: str, content: str = "",
                       finding_id: Optional[str] = None) -> str:
        fid = finding_id or new_id("fnd")
        self._insert("findings", dict(id=fid, incident_id=incident_id, role=role,
                                      wave=wave, status=status, content=content,
                                      ts=_now()))
        return fid

    def findings(self, incident_id: str, max_wave: Optional[int] = None) -> List[Dict[str, Any]]:
        if max_wave is not None:
            return self._select("findings", "incident_id=? AND wave<=?",
                                [incident_id, max_wave], order="ts")
        return self._select("findings", "incident_id=?", [incident_id], order="ts")

    # ---- postmortems / artifacts ----
    def save_postmortem(self, incident_id: str, content: str) -> str:
        existing = self._select("postmortems", "incident_id=?", [incident_id])
        t = _now()
        if existing:
            self._update("postmortems", {"content": content, "updated_at": t},
                         "incident_id=?", [incident_id])
            return existing[0]["id"]
        pid = new_id("pm")
        self._insert("postmortems", dict(id=pid, incident_id=incident_id,
                                         content=content, created_at=t, updated_at=t))
        return pid

    def get_postmortem(self, incident_id: str) -> Optional[Dict[str, Any]]:
        rows = self._select("postmortems", "incident_id=?", [incident_id])
        return rows[0] if rows else None

    def save_artifact(self, name: str, content: str,
                      session_id: Optional[str] = None) -> str:
        rows = self._select("artifacts", "name=? AND session_id IS ?", [name, session_id])
        if rows:
            self._update("artifacts", {"content": content, "updated_at": _now()},
                         "id=?", [rows[0]["id"]])
            return rows[0]["id"]
        aid = new_id("art")
        self._insert("artifacts", dict(id=aid, session_id=session_id, name=name,
                                       content=content, updated_at=_now()))
        return aid

    def artifacts(self, session_id: Optional[str] = None) -> List[Dict[str, Any]]:
        if session_id:
            return self._select("artifacts", "session_id=?", [session_id], order="name")
        return self._select("artifacts", order="name")

    def get_artifact(self, name: str) -> Optional[Dict[str, Any]]:
        rows = self._select("artifacts", "name=?", [name])
        return rows[0] if rows else None

    # ---- lifecycle / usage / audit ----
    def lifecycle(self, incident_id: str, event: str, data: Optional[dict] = None) -> None:
        self._insert("lifecycle_events", dict(
            incident_id=incident_id, event=event,
            data=json.dumps(data or {}), ts=_now()))

    def lifecycle_events(self, incident_id: str) -> List[Dict[str, Any]]:
        return self._select("lifecycle_events", "incident_id=?", [incident_id], order="ts")

    def track_usage(self, model: str, prompt_tokens: int, completion_tokens: int,
                    response_ms: float, purpose: str = "",
                    session_id: Optional[str] = None) -> None:
        self._insert("llm_usage", dict(
            model=model, prompt_tokens=prompt_tokens,
            completion_tokens=completion_tokens, response_ms=response_ms,
            purpose=purpose, session_id=session_id, ts=_now()))

    def usage_summary(self) -> Dict[str, Any]:
        rows = self.db.query(
            "SELECT model, COUNT(*) n, SUM(prompt_tokens) pt, "
            "SUM(completion_tokens) ct FROM llm_usage WHERE org_id=? GROUP BY model",
            [self.org_id])
        return {r["model"]: dict(r) for r in rows}

    def audit(self, kind: str, detail: str) -> None:
        self._insert("audit_events", dict(kind=kind, detail=detail[:4000], ts=_now()))

    def audit_events(self, limit: int = 100) -> List[Dict[str, Any]]:
        return self._select("audit_events", order="ts DESC", limit=limit)

    # ---- command policies / tool permissions ----
    def add_command_policy(self, pattern: str, action: str,
                           created_by: Optional[str] = None) -> str:
        pid = new_id("pol")
        self._insert("command_policies", dict(id=pid, pattern=pattern, action=action,
                                              created_by=created_by, created_at=_now()))
        return pid

    def command_policies(self) -> List[Dict[str, Any]]:
        return self._select("command_policies", order="created_at")

    def delete_command_policy(self, policy_id: str) -> int:
        return self._delete("command_policies", "id=?", [policy_id])

    def set_tool_permission(self, tool: str, allowed: bool) -> None:
        self._insert("org_tool_permissions", dict(tool=tool, allowed=int(allowed)))

    def tool_permissions(self) -> Dict[str, bool]:
        return {r["tool"]: bool(r["allowed"])
                for r in self._select("org_tool_permissions")}

    # ---- knowledge base ----
    def add_kb_document(self, name: str, content: str,
                        user_id: Optional[str] = None) -> str:
        did = new_id("doc")
        self._insert("kb_documents", dict(id=did, user_id=user_id, name=name,
                                          content=content, ts=_now()))
        return did

    def kb_documents(self) -> List[Dict[str, Any]]:
        return self._select("kb_documents", order="ts DESC")

    def delete_kb_document(self, doc_id: str) -> None:
        self._delete("kb_documents", "id=?", [doc_id])
        self._delete("kb_chunks", "doc_id=?", [doc_id])

    def add_kb_chunk(self, doc_id: str, seq: int, text: str) -> str:
        cid = f"{doc_id}:{seq}"
        self._insert("kb_chunks", dict(id=cid, doc_id=doc_id, seq=seq, text=text))
        return cid

    def kb_chunks(self, doc_id: Optional[str] = None) -> List[Dict[str, Any]]:
        if doc_id:
            return self._select("kb_chunks", "doc_id=?", [doc_id], order="seq")
        return self._select("kb_chunks")

    # ---- connector tokens (Vault analog lives in utils.secrets) ----
    def store_tokens(self, user_id: str, provider: str, data_ref: str) -> None:
        self._insert("connector_tokens", dict(user_id=user_id, provider=provider,
                                              data=data_ref, updated_at=_now()))

    def get_token_ref(self, user_id: str, provider: str) -> Optional[str]:
        rows = self._select("connector_tokens", "user_id=? AND provider=?",
                            [user_id, provider])
        return rows[0]["data"] if rows else None

    def connected_providers(self, user_id: str) -> List[str]:
        return [r["provider"] for r in
                self._select("connector_tokens", "user_id=?", [user_id])]

    # ---- infra graph persistence ----
    def upsert_graph_node(self, node_id: str, kind: str, props: dict) -> None:
        self._insert("graph_nodes", dict(id=node_id, kind=kind,
                                         props=json.dumps(props), updated_at=_now()))

    def upsert_graph_edge(self, src: str, dst: str, kind: str, props: dict) -> None:
        self._insert("graph_edges", dict(src=src, dst=dst, kind=kind,
                                         props=json.dumps(props), updated_at=_now()))

    def graph_nodes(self) -> List[Dict[str, Any]]:
        return self._select("graph_nodes")

    def graph_edges(self) -> List[Dict[str, Any]]:
        return self._select("graph_edges")


class AdminStore:
    """Cross-tenant operations (user/org management, reaper, metrics)."""

    def __init__(self, db: Database):
        self.db = db

    def create_org(self, name: str, org_id: Optional[str] = None) -> str:
        oid = org_id or new_id("org")
        self.db.execute("INSERT OR IGNORE INTO orgs (id, name, created_at) VALUES (?,?,?)",
                        [oid, name, _now()])
        return oid

    def create_user(self, org_id: str, email: str, role: str = "member",
                    user_id: Optional[str] = None) -> str:
        uid = user_id or new_id("usr")
        self.db.execute(
            "INSERT OR IGNORE INTO users (id, org_id, email, role, created_at) "
            "VALUES (?,?,?,?,?)", [uid, org_id, email, role, _now()])
        return uid

    def get_user(self, user_id: str) -> Optional[Dict[str, Any]]:
        rows = self.db.query("SELECT * FROM users WHERE id=?", [user_id])
        return _row(rows[0]) if rows else None

    def user_org(self, user_id: str) -> Optional[str]:
        u = self.get_user(user_id)
        return u["org_id"] if u else None

    def orgs(self) -> List[Dict[str, Any]]:
        return [dict(r) for r in self.db.query("SELECT * FROM orgs")]

    def stale_running_incidents(self, older_than_s: float) -> List[Dict[str, Any]]:
        return [dict(r) for r in self.db.query(
            "SELECT * FROM incidents WHERE aurora_status='running' AND updated_at < ?",
            [_now() - older_than_s])]

    def org_store(self, org_id: str) -> OrgStore:
        return OrgStore(self.db, org_id)
