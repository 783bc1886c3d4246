"""Managed-jobs controller state DB.

Reference: sky/jobs/state.py (ManagedJobStatus :507 — PENDING/STARTING/
RUNNING/RECOVERING/SUCCEEDED/FAILED/CANCELLED..., recovery bookkeeping).
"""
from __future__ import annotations

import contextlib
import json
import sqlite3
import time
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state

PENDING = "PENDING"
STARTING = "STARTING"
RUNNING = "RUNNING"
RECOVERING = "RECOVERING"
SUCCEEDED = "SUCCEEDED"
FAILED = "FAILED"
FAILED_SETUP = "FAILED_SETUP"
FAILED_CONTROLLER = "FAILED_CONTROLLER"
CANCELLED = "CANCELLED"

TERMINAL = {SUCCEEDED, FAILED, FAILED_SETUP, FAILED_CONTROLLER, CANCELLED}

_SCHEMA = """
CREATE TABLE IF NOT EXISTS managed_jobs (
    job_id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT,
    status TEXT NOT NULL,
    task TEXT NOT NULL,
    cluster_name TEXT,
    submitted_at REAL,
    started_at REAL,
    ended_at REAL,
    recovery_count INTEGER DEFAULT 0,
    controller_pid INTEGER,
    failure_reason TEXT
);
"""


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(global_state.root_dir() / "managed_jobs.db",
                           timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        with conn:
            yield conn
    finally:
        conn.close()


def create(name: Optional[str], task: Dict[str, Any]) -> int:
    with _conn() as c:
        cur = c.execute(
            "INSERT INTO managed_jobs (name,status,task,submitted_at) "
            "VALUES (?,?,?,?)", (name, PENDING, json.dumps(task),
                                 time.time()))
        return cur.lastrowid


def get(job_id: int) -> Optional[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM managed_jobs LIMIT 0").description]
        row = c.execute("SELECT * FROM managed_jobs WHERE job_id=?",
                        (job_id,)).fetchone()
    if row is None:
        return None
    d = dict(zip(cols, row))
    d["task"] = json.loads(d["task"])
    return d


def list_jobs() -> List[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM managed_jobs LIMIT 0").description]
        rows = c.execute(
            "SELECT * FROM managed_jobs ORDER BY job_id DESC").fetchall()
    out = []
    for r in rows:
        d = dict(zip(cols, r))
        d["task"] = json.loads(d["task"])
        out.append(d)
    return out


def update(job_id: int, **fields) -> None:
    if not fields:
        return
    cols = ", ".join(f"{k}=?" for k in fields)
    with _conn() as c:
        c.execute(f"UPDATE managed_jobs SET {cols} WHERE job_id=?",
                  (*fields.values(), job_id))


def set_status(job_id: int, status: str, reason: str = None) -> None:
    fields: Dict[str, Any] = {"status": status}
    if status == RUNNING:
        fields["started_at"] = time.time()
    if status in TERMINAL:
        fields["ended_at"] = time.time()
    if reason:
        fields["failure_reason"] = reason
    update(job_id, **fields)


def bump_recovery(job_id: int) -> int:
    with _conn() as c:
        c.execute("UPDATE managed_jobs SET recovery_count=recovery_count+1 "
                  "WHERE job_id=?", (job_id,))
        row = c.execute("SELECT recovery_count FROM managed_jobs "
                        "WHERE job_id=?", (job_id,)).fetchone()
    return row[0] if row else 0


def reconcile() -> None:
    """Mark jobs whose controller died as FAILED_CONTROLLER
    (reference: server/daemons.py managed-jobs refresh)."""
    import os
    for j in list_jobs():
        if j["status"] in TERMINAL or j["status"] == PENDING:
            continue
        pid = j.get("controller_pid")
        if pid:
            try:
                os.kill(pid, 0)
            except ProcessLookupError:
                set_status(j["job_id"], FAILED_CONTROLLER,
                           "controller process died")
            except PermissionError:
                pass
