"""Serve controller state DB (reference: sky/serve/serve_state.py)."""
from __future__ import annotations

import contextlib
import json
import sqlite3
import time
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state

# Service statuses (reference: serve_state.ServiceStatus)
CONTROLLER_INIT = "CONTROLLER_INIT"
REPLICA_INIT = "REPLICA_INIT"
READY = "READY"
SHUTTING_DOWN = "SHUTTING_DOWN"
FAILED = "FAILED"
SHUTDOWN = "SHUTDOWN"

# Replica statuses (reference: serve_state.ReplicaStatus)
R_PROVISIONING = "PROVISIONING"
R_STARTING = "STARTING"
R_READY = "READY"
R_NOT_READY = "NOT_READY"
R_FAILED = "FAILED"
R_SHUTTING_DOWN = "SHUTTING_DOWN"
R_SHUTDOWN = "SHUTDOWN"

_SCHEMA = """
CREATE TABLE IF NOT EXISTS services (
    name TEXT PRIMARY KEY,
    status TEXT NOT NULL,
    task TEXT NOT NULL,
    spec TEXT NOT NULL,
    controller_pid INTEGER,
    lb_port INTEGER,
    version INTEGER DEFAULT 1,
    created_at REAL
);
CREATE TABLE IF NOT EXISTS replicas (
    service TEXT NOT NULL,
    replica_id INTEGER NOT NULL,
    status TEXT NOT NULL,
    cluster_name TEXT,
    endpoint TEXT,
    version INTEGER DEFAULT 1,
    launched_at REAL,
    PRIMARY KEY (service, replica_id)
);
"""


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(global_state.root_dir() / "serve.db", timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        cols = [r[1] for r in conn.execute("PRAGMA table_info(services)")]
        if "tls" not in cols:
            conn.execute("ALTER TABLE services ADD COLUMN tls INTEGER "
                         "DEFAULT 0")
        with conn:
            yield conn
    finally:
        conn.close()


def add_service(name: str, task: Dict[str, Any], spec: Dict[str, Any],
                lb_port: int) -> None:
    with _conn() as c:
        c.execute(
            "INSERT OR REPLACE INTO services "
            "(name,status,task,spec,lb_port,version,created_at) "
            "VALUES (?,?,?,?,?,1,?)",
            (name, CONTROLLER_INIT, json.dumps(task), json.dumps(spec),
             lb_port, time.time()))


def bump_version(name: str, task: Dict[str, Any],
                 spec: Dict[str, Any]) -> int:
    """Rolling update: record the new task/spec and bump the service
    version (reference: serve replica_managers version tracking)."""
    with _conn() as c:
        c.execute("UPDATE services SET task=?, spec=?, version=version+1 "
                  "WHERE name=?", (json.dumps(task), json.dumps(spec), name))
        row = c.execute("SELECT version FROM services WHERE name=?",
                        (name,)).fetchone()
    return row[0] if row else 0


def service_dir(name: str):
    from pathlib import Path
    d = Path(global_state.root_dir()) / "serve" / name
    d.mkdir(parents=True, exist_ok=True)
    return d


def update_service(name: str, **fields) -> None:
    if not fields:
        return
    cols = ", ".join(f"{k}=?" for k in fields)
    with _conn() as c:
        c.execute(f"UPDATE services SET {cols} WHERE name=?",
                  (*fields.values(), name))


def get_service(name: str) -> Optional[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM services LIMIT 0").description]
        row = c.execute("SELECT * FROM services WHERE name=?",
                        (name,)).fetchone()
    if row is None:
        return None
    d = dict(zip(cols, row))
    d["task"] = json.loads(d["task"])
    d["spec"] = json.loads(d["spec"])
    return d


def list_services() -> List[Dict[str, Any]]:
    with _conn() as c:
        rows = c.execute("SELECT name FROM services").fetchall()
    return [get_service(r[0]) for r in rows]


def remove_service(name: str) -> None:
    with _conn() as c:
        c.execute("DELETE FROM services WHERE name=?", (name,))
        c.execute("DELETE FROM replicas WHERE service=?", (name,))


def upsert_replica(service: str, replica_id: int, **fields) -> None:
    with _conn() as c:
        c.execute(
            "INSERT INTO replicas (service,replica_id,status) "
            "VALUES (?,?,?) ON CONFLICT(service,replica_id) DO NOTHING",
            (service, replica_id, fields.get("status", R_PROVISIONING)))
        if fields:
            cols = ", ".join(f"{k}=?" for k in fields)
            c.execute(
                f"UPDATE replicas SET {cols} WHERE service=? AND "
                "replica_id=?", (*fields.values(), service, replica_id))


def remove_replica(service: str, replica_id: int) -> None:
    with _conn() as c:
        c.execute("DELETE FROM replicas WHERE service=? AND replica_id=?",
                  (service, replica_id))


def list_replicas(service: str) -> List[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM replicas LIMIT 0").description]
        rows = c.execute("SELECT * FROM replicas WHERE service=? "
                         "ORDER BY replica_id", (service,)).fetchall()
    return [dict(zip(cols, r)) for r in rows]


def reconcile() -> None:
    import os
    for s in list_services():
        if s["status"] in (SHUTDOWN, FAILED):
            continue
        pid = s.get("controller_pid")
        if pid:
            try:
                os.kill(pid, 0)
            except ProcessLookupError:
                update_service(s["name"], status=FAILED)
            except PermissionError:
                pass
