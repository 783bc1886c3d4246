"""Persistent async-request table (reference: sky/server/requests/
requests.py — id, name, status, pickled return value, logs path)."""
from __future__ import annotations

import json
import sqlite3
import time
import uuid
from pathlib import Path
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state

PENDING = "PENDING"
RUNNING = "RUNNING"
SUCCEEDED = "SUCCEEDED"
FAILED = "FAILED"
CANCELLED = "CANCELLED"
TERMINAL = {SUCCEEDED, FAILED, CANCELLED}

_SCHEMA = """
CREATE TABLE IF NOT EXISTS requests (
    request_id TEXT PRIMARY KEY,
    name TEXT NOT NULL,
    queue TEXT NOT NULL,
    status TEXT NOT NULL,
    created_at REAL,
    started_at REAL,
    finished_at REAL,
    body TEXT,
    result TEXT,
    error TEXT,
    worker_pid INTEGER,
    log_path TEXT,
    user TEXT,
    workspace TEXT
);
"""


def api_dir() -> Path:
    d = global_state.root_dir() / "api"
    (d / "logs").mkdir(parents=True, exist_ok=True)
    return d


import contextlib


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(api_dir() / "requests.db", timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        try:
            # WAL is a persistent property of the db file; switching it
            # needs an exclusive lock, so under heavy concurrent opens
            # (or teardown races in tests) this can raise "database is
            # locked" — it is an optimization, never correctness.
            conn.execute("PRAGMA journal_mode=WAL")
        except sqlite3.OperationalError:
            pass
        conn.executescript(_SCHEMA)
        cols = [r[1] for r in conn.execute("PRAGMA table_info(requests)")]
        if "user" not in cols:  # pre-RBAC databases
            conn.execute("ALTER TABLE requests ADD COLUMN user TEXT")
        if "workspace" not in cols:
            conn.execute("ALTER TABLE requests ADD COLUMN workspace TEXT")
        with conn:
            yield conn
    finally:
        conn.close()


def create(name: str, body: Dict[str, Any], queue: str,
           user: Optional[str] = None,
           workspace: Optional[str] = None) -> str:
    rid = uuid.uuid4().hex[:16]
    log_path = str(api_dir() / "logs" / f"{rid}.log")
    with _conn() as c:
        c.execute(
            "INSERT INTO requests (request_id,name,queue,status,created_at,"
            "body,log_path,user,workspace) VALUES (?,?,?,?,?,?,?,?,?)",
            (rid, name, queue, PENDING, time.time(), json.dumps(body),
             log_path, user, workspace))
    return rid


def claim_next(queue: str, worker_pid: int) -> Optional[Dict[str, Any]]:
    with _conn() as c:
        c.execute("BEGIN IMMEDIATE")
        row = c.execute(
            "SELECT request_id FROM requests WHERE status=? AND queue=? "
            "ORDER BY created_at LIMIT 1", (PENDING, queue)).fetchone()
        if row is None:
            c.execute("COMMIT")
            return None
        rid = row[0]
        c.execute(
            "UPDATE requests SET status=?, started_at=?, worker_pid=? "
            "WHERE request_id=?", (RUNNING, time.time(), worker_pid, rid))
        c.execute("COMMIT")
    return get(rid)


def get(rid: str) -> Optional[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM requests LIMIT 0").description]
        row = c.execute("SELECT * FROM requests WHERE request_id=?",
                        (rid,)).fetchone()
    if row is None:
        return None
    d = dict(zip(cols, row))
    d["body"] = json.loads(d["body"] or "{}")
    d["result"] = json.loads(d["result"]) if d["result"] else None
    return d


def finish(rid: str, status: str, result: Any = None,
           error: Optional[str] = None) -> None:
    with _conn() as c:
        # never resurrect a CANCELLED row: a runner that slipped through
        # the cancel window may still call finish() when it completes
        c.execute(
            "UPDATE requests SET status=?, finished_at=?, result=?, error=? "
            "WHERE request_id=? AND status != ?",
            (status, time.time(), json.dumps(result), error, rid,
             CANCELLED))


def set_pid(rid: str, pid: int) -> None:
    with _conn() as c:
        c.execute("UPDATE requests SET worker_pid=? WHERE request_id=?",
                  (pid, rid))


def list_requests(limit: int = 100) -> List[Dict[str, Any]]:
    with _conn() as c:
        cols = [d[0] for d in
                c.execute("SELECT * FROM requests LIMIT 0").description]
        rows = c.execute(
            "SELECT * FROM requests ORDER BY created_at DESC LIMIT ?",
            (limit,)).fetchall()
    out = []
    for r in rows:
        d = dict(zip(cols, r))
        d["body"] = json.loads(d["body"] or "{}")
        d["result"] = json.loads(d["result"]) if d["result"] else None
        out.append(d)
    return out


def mark_cancelled(rid: str):
    """Mark cancelled.  Returns (marked, worker_pid): worker_pid is the
    dedicated runner's pid or None (inline request / not yet spawned)."""
    with _conn() as c:
        row = c.execute(
            "SELECT status, worker_pid FROM requests WHERE request_id=?",
            (rid,)).fetchone()
        if row is None or row[0] in TERMINAL:
            return False, None
        c.execute(
            "UPDATE requests SET status=?, finished_at=? WHERE request_id=?",
            (CANCELLED, time.time(), rid))
    return True, row[1]


def gc_requests(max_age_days: float = None, keep_latest: int = None
                ) -> int:
    """Retention for terminal request rows + their log files
    (reference: sky server requests GC — the table is otherwise
    unbounded on a long-lived server).  Keeps every non-terminal row,
    the newest `keep_latest` terminal rows, and anything younger than
    `max_age_days`.  Returns rows deleted."""
    import os as _os
    if max_age_days is None:
        max_age_days = float(_os.environ.get(
            "SKY_AMD_REQUEST_RETENTION_DAYS", "7"))
    if keep_latest is None:
        keep_latest = int(_os.environ.get(
            "SKY_AMD_REQUEST_RETENTION_COUNT", "2000"))
    cutoff = time.time() - max_age_days * 86400
    removed = 0
    with _conn() as c:
        rows = c.execute(
            "SELECT request_id, log_path FROM requests WHERE status IN "
            "('SUCCEEDED','FAILED','CANCELLED') AND created_at < ? "
            "AND request_id NOT IN (SELECT request_id FROM requests "
            "WHERE status IN ('SUCCEEDED','FAILED','CANCELLED') "
            "ORDER BY created_at DESC LIMIT ?)",
            (cutoff, keep_latest)).fetchall()
        for rid, log_path in rows:
            c.execute("DELETE FROM requests WHERE request_id=?", (rid,))
            removed += 1
            if log_path:
                try:
                    _os.unlink(log_path)
                except OSError:
                    pass
    return removed
