"""SQLite-backed persistence: tasks, agents, logs, messages, costs, secrets,
profiles.

The reference persists through Postgres/Ecto (reference: lib/quoracle/repo.ex,
tasks/task_manager.ex:320-423, priv/repo/migrations/); the benchmark-relevant
semantics are continuous checkpointing (agent row at spawn, conversation after
every decision, ACE state after condensation and on terminate, every
action/message/cost) and resumability — a lightweight embedded store provides
them without an external service.  WAL mode keeps writes off the agent hot
path's critical section; an in-memory store (path=":memory:") backs tests.
"""

from __future__ import annotations

import json
import os
import sqlite3
import threading
import time
from typing import Any, Dict, List, Optional

_SCHEMA = """
CREATE TABLE IF NOT EXISTS tasks (
  task_id TEXT PRIMARY KEY,
  status TEXT NOT NULL DEFAULT 'running',
  prompt TEXT,
  profile TEXT,
  budget_limit REAL,
  global_context TEXT,
  initial_constraints TEXT,
  grove TEXT,
  inserted_at REAL,
  updated_at REAL
);
CREATE TABLE IF NOT EXISTS agents (
  agent_id TEXT PRIMARY KEY,
  task_id TEXT NOT NULL,
  parent_id TEXT,
  status TEXT NOT NULL DEFAULT 'running',
  config TEXT,
  state TEXT,
  inserted_at REAL,
  updated_at REAL
);
CREATE INDEX IF NOT EXISTS agents_task ON agents(task_id);
CREATE TABLE IF NOT EXISTS logs (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  agent_id TEXT,
  task_id TEXT,
  level TEXT,
  event_type TEXT,
  message TEXT,
  metadata TEXT,
  inserted_at REAL
);
CREATE INDEX IF NOT EXISTS logs_agent ON logs(agent_id);
CREATE TABLE IF NOT EXISTS messages (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  task_id TEXT,
  from_agent TEXT,
  to_agent TEXT,
  content TEXT,
  inserted_at REAL
);
CREATE INDEX IF NOT EXISTS messages_task ON messages(task_id);
CREATE TABLE IF NOT EXISTS agent_costs (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  agent_id TEXT,
  task_id TEXT,
  model TEXT,
  amount REAL,
  category TEXT,
  description TEXT,
  metadata TEXT,
  inserted_at REAL
);
CREATE INDEX IF NOT EXISTS costs_agent ON agent_costs(agent_id);
CREATE TABLE IF NOT EXISTS secrets (
  name TEXT PRIMARY KEY,
  value BLOB,
  description TEXT,
  inserted_at REAL
);
CREATE TABLE IF NOT EXISTS secret_usage (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  name TEXT,
  agent_id TEXT,
  action TEXT,
  inserted_at REAL
);
CREATE TABLE IF NOT EXISTS profiles (
  name TEXT PRIMARY KEY,
  data TEXT,
  inserted_at REAL
);
"""


class Store:
    """Thread-safe embedded store.  All values JSON-encoded where structured."""

    def __init__(self, path: str = ":memory:"):
        self.path = path
        if path != ":memory:":
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self._conn = sqlite3.connect(path, check_same_thread=False)
        self._conn.row_factory = sqlite3.Row
        self._lock = threading.Lock()
        with self._lock:
            if path != ":memory:":
                self._conn.execute("PRAGMA journal_mode=WAL")
            self._conn.execute("PRAGMA synchronous=NORMAL")
            self._conn.executescript(_SCHEMA)
            self._conn.commit()

    def close(self) -> None:
        with self._lock:
            self._conn.close()

    def _exec(self, sql: str, params: tuple = ()) -> sqlite3.Cursor:
        with self._lock:
            cur = self._conn.execute(sql, params)
            self._conn.commit()
            return cur

    def _query(self, sql: str, params: tuple = ()) -> List[sqlite3.Row]:
        with self._lock:
            return self._conn.execute(sql, params).fetchall()

    # -- tasks -----------------------------------------------------------------
    def save_task(self, task: Dict[str, Any]) -> None:
        now = time.time()
        self._exec(
            "INSERT INTO tasks (task_id, status, prompt, profile, budget_limit,"
            " global_context, initial_constraints, grove, inserted_at, updated_at)"
            " VALUES (?,?,?,?,?,?,?,?,?,?)"
            " ON CONFLICT(task_id) DO UPDATE SET status=excluded.status,"
            " prompt=excluded.prompt, profile=excluded.profile,"
            " budget_limit=excluded.budget_limit,"
            " global_context=excluded.global_context,"
            " initial_constraints=excluded.initial_constraints,"
            " grove=excluded.grove, updated_at=excluded.updated_at",
            (task["task_id"], task.get("status", "running"), task.get("prompt"),
             task.get("profile"), task.get("budget_limit"),
             task.get("global_context"),
             json.dumps(task.get("initial_constraints") or []),
             json.dumps(task.get("grove")) if task.get("grove") else None,
             now, now))

    def update_task_status(self, task_id: str, status: str) -> None:
        self._exec("UPDATE tasks SET status=?, updated_at=? WHERE task_id=?",
                   (status, time.time(), task_id))

    def get_task(self, task_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM tasks WHERE task_id=?", (task_id,))
        return self._task_row(rows[0]) if rows else None

    def list_tasks(self, status: Optional[str] = None) -> List[Dict[str, Any]]:
        if status:
            rows = self._query("SELECT * FROM tasks WHERE status=?", (status,))
        else:
            rows = self._query("SELECT * FROM tasks")
        return [self._task_row(r) for r in rows]

    def delete_task(self, task_id: str) -> None:
        self._exec("DELETE FROM agents WHERE task_id=?", (task_id,))
        self._exec("DELETE FROM tasks WHERE task_id=?", (task_id,))

    @staticmethod
    def _task_row(row: sqlite3.Row) -> Dict[str, Any]:
        d = dict(row)
        d["initial_constraints"] = json.loads(d.get("initial_constraints") or "[]")
        d["grove"] = json.loads(d["grove"]) if d.get("grove") else None
        return d

    # -- agents ----------------------------------------------------------------
    def save_agent(self, agent_id: str, task_id: str, parent_id: Optional[str],
                   config: Dict[str, Any], state: Optional[Dict[str, Any]] = None,
                   status: str = "running") -> None:
        now = time.time()
        self._exec(
            "INSERT INTO agents (agent_id, task_id, parent_id, status, config,"
            " state, inserted_at, updated_at) VALUES (?,?,?,?,?,?,?,?)"
            " ON CONFLICT(agent_id) DO UPDATE SET status=excluded.status,"
            " config=excluded.config, state=excluded.state,"
            " updated_at=excluded.updated_at",
            (agent_id, task_id, parent_id, status,
             json.dumps(config, default=str),
             json.dumps(state, default=str) if state is not None else None,
             now, now))

    def update_agent_state(self, agent_id: str, state: Dict[str, Any]) -> None:
        self._exec("UPDATE agents SET state=?, updated_at=? WHERE agent_id=?",
                   (json.dumps(state, default=str), time.time(), agent_id))

    def update_agent_status(self, agent_id: str, status: str) -> None:
        self._exec("UPDATE agents SET status=?, updated_at=? WHERE agent_id=?",
                   (status, time.time(), agent_id))

    def get_agent(self, agent_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM agents WHERE agent_id=?", (agent_id,))
        return self._agent_row(rows[0]) if rows else None

    def agents_for_task(self, task_id: str,
                        status: Optional[str] = None) -> List[Dict[str, Any]]:
        if status:
            rows = self._query(
                "SELECT * FROM agents WHERE task_id=? AND status=?",
                (task_id, status))
        else:
            rows = self._query("SELECT * FROM agents WHERE task_id=?", (task_id,))
        return [self._agent_row(r) for r in rows]

    @staticmethod
    def _agent_row(row: sqlite3.Row) -> Dict[str, Any]:
        d = dict(row)
        d["config"] = json.loads(d["config"]) if d.get("config") else {}
        d["state"] = json.loads(d["state"]) if d.get("state") else None
        return d

    # -- logs / messages ---------------------------------------------------------
    def save_log(self, agent_id: str, task_id: str, level: str, event_type: str,
                 message: str, metadata: Optional[Dict[str, Any]] = None) -> None:
        self._exec(
            "INSERT INTO logs (agent_id, task_id, level, event_type, message,"
            " metadata, inserted_at) VALUES (?,?,?,?,?,?,?)",
            (agent_id, task_id, level, event_type, message,
             json.dumps(metadata or {}, default=str), time.time()))

    def logs_for_agent(self, agent_id: str, limit: int = 100) -> List[Dict[str, Any]]:
        rows = self._query(
            "SELECT * FROM logs WHERE agent_id=? ORDER BY id DESC LIMIT ?",
            (agent_id, limit))
        return [dict(r) for r in rows]

    def save_message(self, task_id: str, from_agent: str, to_agent: str,
                     content: str) -> None:
        self._exec(
            "INSERT INTO messages (task_id, from_agent, to_agent, content,"
            " inserted_at) VALUES (?,?,?,?,?)",
            (task_id, from_agent, to_agent, content, time.time()))

    def messages_for_task(self, task_id: str, limit: int = 200) -> List[Dict[str, Any]]:
        rows = self._query(
            "SELECT * FROM messages WHERE task_id=? ORDER BY id DESC LIMIT ?",
            (task_id, limit))
        return [dict(r) for r in rows]

    # -- costs -----------------------------------------------------------------
    def save_cost(self, agent_id: str, task_id: str, model: Optional[str],
                  amount: float, category: str = "model_query",
                  description: str = "",
                  metadata: Optional[Dict[str, Any]] = None) -> None:
        self._exec(
            "INSERT INTO agent_costs (agent_id, task_id, model, amount, category,"
            " description, metadata, inserted_at) VALUES (?,?,?,?,?,?,?,?)",
            (agent_id, task_id, model, amount, category, description,
             json.dumps(metadata or {}, default=str), time.time()))

    def costs_for_agent(self, agent_id: str) -> List[Dict[str, Any]]:
        rows = self._query("SELECT * FROM agent_costs WHERE agent_id=?", (agent_id,))
        return [dict(r) for r in rows]

    def total_cost(self, agent_ids: List[str]) -> float:
        if not agent_ids:
            return 0.0
        marks = ",".join("?" for _ in agent_ids)
        rows = self._query(
            f"SELECT COALESCE(SUM(amount),0) AS total FROM agent_costs"
            f" WHERE agent_id IN ({marks})", tuple(agent_ids))
        return float(rows[0]["total"])

    def cost_rollup(self, agent_id: str) -> Dict[str, Any]:
        """Recursive descendant cost aggregation via CTE over the agents
        table's parent links (reference: costs/aggregator.ex:122-163)."""
        rows = self._query(
            """
            WITH RECURSIVE subtree(id) AS (
                SELECT agent_id FROM agents WHERE agent_id = ?
                UNION ALL
                SELECT a.agent_id FROM agents a
                JOIN subtree s ON a.parent_id = s.id
            )
            SELECT c.model, c.category,
                   COALESCE(SUM(c.amount), 0) AS amount,
                   COUNT(*) AS entries
            FROM agent_costs c JOIN subtree s ON c.agent_id = s.id
            GROUP BY c.model, c.category
            """, (agent_id,))
        by_model: Dict[str, float] = {}
        total = 0.0
        for r in rows:
            total += r["amount"]
            key = r["model"] or r["category"]
            by_model[key] = by_model.get(key, 0.0) + r["amount"]
        own = self.total_cost([agent_id])
        return {"agent_id": agent_id, "total": total, "own": own,
                "descendants": total - own, "by_model": by_model}

    # -- secrets -----------------------------------------------------------------
    def save_secret(self, name: str, value: bytes, description: str = "") -> None:
        self._exec(
            "INSERT INTO secrets (name, value, description, inserted_at)"
            " VALUES (?,?,?,?) ON CONFLICT(name) DO UPDATE SET"
            " value=excluded.value, description=excluded.description",
            (name, value, description, time.time()))

    def get_secret(self, name: str) -> Optional[bytes]:
        rows = self._query("SELECT value FROM secrets WHERE name=?", (name,))
        return rows[0]["value"] if rows else None

    def list_secret_names(self) -> List[str]:
        return [r["name"] for r in self._query("SELECT name FROM secrets")]

    def record_secret_usage(self, name: str, agent_id: str, action: str) -> None:
        self._exec(
            "INSERT INTO secret_usage (name, agent_id, action, inserted_at)"
            " VALUES (?,?,?,?)", (name, agent_id, action, time.time()))

    def secret_usage(self, name: str) -> List[Dict[str, Any]]:
        rows = self._query("SELECT * FROM secret_usage WHERE name=?", (name,))
        return [dict(r) for r in rows]

    # -- profiles -----------------------------------------------------------------
    def save_profile(self, profile: Dict[str, Any]) -> None:
        self._exec(
            "INSERT INTO profiles (name, data, inserted_at) VALUES (?,?,?)"
            " ON CONFLICT(name) DO UPDATE SET data=excluded.data",
            (profile["name"], json.dumps(profile, default=str), time.time()))

    def list_profiles(self) -> List[Dict[str, Any]]:
        return [json.loads(r["data"]) for r in self._query("SELECT data FROM profiles")]
