"""Global user state — SQLite-backed cluster/request/event tables.

Reference: sky/global_user_state.py (tables clusters, cluster_history,
cluster_events; handle pickled into the cluster row :1456).  We store the
handle as JSON instead of pickle (no cross-version pickle compat burden)
and keep the event-history table that powers job-start-latency
measurement (reference: global_user_state.py:1001 add_cluster_event).
"""
from __future__ import annotations

import json
import os
import sqlite3
import threading
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

_DB_LOCK = threading.Lock()


def root_dir() -> Path:
    d = Path(os.environ.get("SKY_AMD_HOME", "~/.sky_amd")).expanduser()
    d.mkdir(parents=True, exist_ok=True)
    return d


def _db_path() -> Path:
    return root_dir() / "state.db"


_SCHEMA = """
CREATE TABLE IF NOT EXISTS clusters (
    name TEXT PRIMARY KEY,
    status TEXT NOT NULL,
    handle TEXT NOT NULL,
    resources TEXT NOT NULL,
    launched_at REAL,
    last_use TEXT,
    user TEXT,
    autostop_idle_minutes INTEGER DEFAULT -1,
    autostop_down INTEGER DEFAULT 0,
    to_down INTEGER DEFAULT 0,
    workspace TEXT DEFAULT 'default'
);
CREATE TABLE IF NOT EXISTS users (
    name TEXT PRIMARY KEY,
    created_at REAL
);
CREATE TABLE IF NOT EXISTS cluster_events (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    cluster TEXT NOT NULL,
    ts REAL NOT NULL,
    event TEXT NOT NULL,
    detail TEXT
);
CREATE TABLE IF NOT EXISTS cluster_history (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT NOT NULL,
    launched_at REAL,
    torn_down_at REAL,
    resources TEXT
);
CREATE TABLE IF NOT EXISTS storage (
    name TEXT PRIMARY KEY,
    source TEXT,
    store_type TEXT,
    created_at REAL
);
CREATE TABLE IF NOT EXISTS config_kv (
    key TEXT PRIMARY KEY,
    value TEXT
);
"""

# Cluster status values (reference: sky/utils/status_lib.ClusterStatus).
INIT = "INIT"
UP = "UP"
STOPPED = "STOPPED"


import contextlib


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(_db_path(), timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        cols = [r[1] for r in conn.execute("PRAGMA table_info(clusters)")]
        if "workspace" not in cols:  # pre-workspace databases
            conn.execute("ALTER TABLE clusters ADD COLUMN workspace TEXT "
                         "DEFAULT 'default'")
        with conn:
            yield conn
    finally:
        conn.close()


_request_user = threading.local()


def set_request_user(name):
    """Thread-scoped identity for SHORT in-process request handlers
    (LONG handlers run in their own process and use the env var)."""
    _request_user.name = name


def set_request_workspace(ws):
    """Thread-scoped workspace for SHORT in-process request handlers
    (LONG handlers run in their own process and use the env var)."""
    _request_user.workspace = ws


def current_workspace() -> str:
    """Active workspace (reference: sky workspaces — named scopes that
    partition clusters/jobs): request header (thread-local/env, set by
    the API executor), else SKY_AMD_WORKSPACE env, else config
    `workspace:`, else "default"."""
    ws = (getattr(_request_user, "workspace", None)
          or os.environ.get("SKY_AMD_WORKSPACE"))
    if ws:
        return ws
    try:
        from skypilot_amd import config as sky_config
        return sky_config.get_nested(["workspace"], "default") or "default"
    except Exception:  # noqa: BLE001
        return "default"


def current_user() -> str:
    """reference: sky/models.py User — request identity (thread-local,
    set by the API executor), else env override, else OS user."""
    return (getattr(_request_user, "name", None)
            or os.environ.get("SKY_AMD_USER")
            or os.environ.get("USER", "root"))


def add_or_update_cluster(name: str, status: str, handle: Dict[str, Any],
                          resources: Dict[str, Any],
                          launched_at: Optional[float] = None) -> None:
    user = current_user()
    with _DB_LOCK, _conn() as c:
        existing = c.execute("SELECT launched_at FROM clusters WHERE name=?",
                             (name,)).fetchone()
        if launched_at is None:
            launched_at = existing[0] if existing else time.time()
        c.execute("INSERT OR IGNORE INTO users (name, created_at) "
                  "VALUES (?,?)", (user, time.time()))
        c.execute(
            "INSERT INTO clusters "
            "(name,status,handle,resources,launched_at,user,workspace)"
            " VALUES (?,?,?,?,?,?,?) ON CONFLICT(name) DO UPDATE SET "
            "status=excluded.status, handle=excluded.handle, "
            "resources=excluded.resources, launched_at=excluded.launched_at",
            (name, status, json.dumps(handle), json.dumps(resources),
             launched_at, user, current_workspace()))


def set_cluster_status(name: str, status: str) -> None:
    with _DB_LOCK, _conn() as c:
        c.execute("UPDATE clusters SET status=? WHERE name=?", (status, name))


def set_cluster_autostop(name: str, idle_minutes: int, down: bool) -> None:
    with _DB_LOCK, _conn() as c:
        c.execute(
            "UPDATE clusters SET autostop_idle_minutes=?, autostop_down=? "
            "WHERE name=?", (idle_minutes, int(down), name))


def get_cluster(name: str) -> Optional[Dict[str, Any]]:
    with _DB_LOCK, _conn() as c:
        row = c.execute(
            "SELECT name,status,handle,resources,launched_at,"
            "autostop_idle_minutes,autostop_down,user,workspace "
            "FROM clusters WHERE name=?",
            (name,)).fetchone()
    if row is None:
        return None
    return _row_to_cluster(row)


def _row_to_cluster(row) -> Dict[str, Any]:
    return {
        "name": row[0],
        "status": row[1],
        "handle": json.loads(row[2]),
        "resources": json.loads(row[3]),
        "launched_at": row[4],
        "autostop_idle_minutes": row[5],
        "autostop_down": bool(row[6]),
        "user": row[7],
        "workspace": row[8] if len(row) > 8 else "default",
    }


def list_clusters(all_workspaces: bool = False) -> List[Dict[str, Any]]:
    """Clusters in the active workspace (all with all_workspaces)."""
    q = ("SELECT name,status,handle,resources,launched_at,"
         "autostop_idle_minutes,autostop_down,user,workspace FROM "
         "clusters ")
    with _DB_LOCK, _conn() as c:
        if all_workspaces:
            rows = c.execute(q + "ORDER BY launched_at DESC").fetchall()
        else:
            rows = c.execute(
                q + "WHERE workspace=? ORDER BY launched_at DESC",
                (current_workspace(),)).fetchall()
    return [_row_to_cluster(r) for r in rows]


def remove_cluster(name: str) -> None:
    with _DB_LOCK, _conn() as c:
        row = c.execute(
            "SELECT launched_at, resources FROM clusters WHERE name=?",
            (name,)).fetchone()
        if row:
            c.execute(
                "INSERT INTO cluster_history "
                "(name,launched_at,torn_down_at,resources) VALUES (?,?,?,?)",
                (name, row[0], time.time(), row[1]))
        c.execute("DELETE FROM clusters WHERE name=?", (name,))


def list_cluster_history(limit: int = 50):
    with _conn() as c:
        rows = c.execute(
            "SELECT name, launched_at, torn_down_at, resources FROM "
            "cluster_history ORDER BY torn_down_at DESC LIMIT ?",
            (limit,)).fetchall()
    return [{"name": r[0], "launched_at": r[1], "torn_down_at": r[2],
             "resources": json.loads(r[3]) if r[3] else {}}
            for r in rows]


def add_cluster_event(cluster: str, event: str, detail: str = "") -> None:
    with _DB_LOCK, _conn() as c:
        c.execute(
            "INSERT INTO cluster_events (cluster,ts,event,detail) "
            "VALUES (?,?,?,?)", (cluster, time.time(), event, detail))


def get_cluster_events(cluster: str) -> List[Dict[str, Any]]:
    with _DB_LOCK, _conn() as c:
        rows = c.execute(
            "SELECT ts,event,detail FROM cluster_events WHERE cluster=? "
            "ORDER BY ts", (cluster,)).fetchall()
    return [{"ts": r[0], "event": r[1], "detail": r[2]} for r in rows]


def list_users() -> List[Dict[str, Any]]:
    with _DB_LOCK, _conn() as c:
        rows = c.execute("SELECT name, created_at FROM users").fetchall()
    return [{"name": r[0], "created_at": r[1]} for r in rows]


def set_config(key: str, value: Any) -> None:
    with _DB_LOCK, _conn() as c:
        c.execute(
            "INSERT INTO config_kv (key,value) VALUES (?,?) "
            "ON CONFLICT(key) DO UPDATE SET value=excluded.value",
            (key, json.dumps(value)))


def get_config(key: str, default: Any = None) -> Any:
    with _DB_LOCK, _conn() as c:
        row = c.execute("SELECT value FROM config_kv WHERE key=?",
                        (key,)).fetchone()
    return json.loads(row[0]) if row else default
