"""Worker pools — warm clusters that managed jobs schedule onto.

Reference: `sky jobs pool apply` (pool = serve-style
replica set of idle clusters; managed jobs scheduled onto them via the
QueueLengthAutoscaler).  Here: a pool is N warm clusters provisioned
from a task template's resources/setup; `sky jobs launch --pool <name>`
makes the job controller acquire a free worker (skipping provisioning,
so job-start latency is one `exec`), and release it on completion.
"""
from __future__ import annotations

import contextlib
import json
import sqlite3
import time
from typing import Any, Dict, List, Optional

from skypilot_amd import execution, global_state
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.task import Task

_SCHEMA = """
CREATE TABLE IF NOT EXISTS pools (
    name TEXT PRIMARY KEY,
    template TEXT NOT NULL,
    num_workers INTEGER NOT NULL,
    created_at REAL
);
CREATE TABLE IF NOT EXISTS pool_workers (
    pool TEXT NOT NULL,
    worker_id INTEGER NOT NULL,
    cluster_name TEXT NOT NULL,
    status TEXT NOT NULL,           -- READY | BUSY | FAILED
    assigned_job INTEGER,
    PRIMARY KEY (pool, worker_id)
);
"""


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(global_state.root_dir() / "pools.db", timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        with conn:
            yield conn
    finally:
        conn.close()


def apply(name: str, template: Dict[str, Any], num_workers: int
          ) -> Dict[str, Any]:
    """Create (or resize) a pool: provision warm clusters running only
    the template's setup."""
    tmpl = dict(template)
    tmpl.pop("run", None)  # workers are warm, not running anything
    Task.from_yaml_config(dict(tmpl))
    with _conn() as c:
        c.execute(
            "INSERT INTO pools (name,template,num_workers,created_at) "
            "VALUES (?,?,?,?) ON CONFLICT(name) DO UPDATE SET "
            "template=excluded.template, num_workers=excluded.num_workers",
            (name, json.dumps(tmpl), num_workers, time.time()))
    workers = []
    for i in range(num_workers):
        cluster = f"sky-pool-{name}-{i}"
        task = Task.from_yaml_config(dict(tmpl))
        execution.launch(task, cluster, detach_run=True)
        with _conn() as c:
            c.execute(
                "INSERT INTO pool_workers (pool,worker_id,cluster_name,"
                "status) VALUES (?,?,?,?) ON CONFLICT(pool,worker_id) "
                "DO UPDATE SET status='READY', assigned_job=NULL",
                (name, i, cluster, "READY"))
        workers.append(cluster)
    return {"pool": name, "workers": workers}


def status(name: Optional[str] = None) -> List[Dict[str, Any]]:
    with _conn() as c:
        pools = c.execute("SELECT name,num_workers FROM pools").fetchall()
        out = []
        for pname, n in pools:
            if name and pname != name:
                continue
            rows = c.execute(
                "SELECT worker_id,cluster_name,status,assigned_job FROM "
                "pool_workers WHERE pool=?", (pname,)).fetchall()
            out.append({
                "name": pname, "num_workers": n,
                "workers": [{"worker_id": r[0], "cluster_name": r[1],
                             "status": r[2], "assigned_job": r[3]}
                            for r in rows],
            })
    return out


def down(name: str) -> int:
    backend = PoolBackend()
    n = 0
    for p in status(name):
        for w in p["workers"]:
            rec = global_state.get_cluster(w["cluster_name"])
            if rec:
                try:
                    backend.teardown(rec["handle"], terminate=True)
                    n += 1
                except Exception:  # noqa: BLE001
                    pass
    with _conn() as c:
        c.execute("DELETE FROM pools WHERE name=?", (name,))
        c.execute("DELETE FROM pool_workers WHERE pool=?", (name,))
    return n


def acquire(pool: str, job_id: int) -> Optional[str]:
    """Atomically claim a READY worker for a managed job."""
    with _conn() as c:
        c.execute("BEGIN IMMEDIATE")
        row = c.execute(
            "SELECT worker_id, cluster_name FROM pool_workers WHERE pool=? "
            "AND status='READY' LIMIT 1", (pool,)).fetchone()
        if row is None:
            c.execute("COMMIT")
            return None
        c.execute(
            "UPDATE pool_workers SET status='BUSY', assigned_job=? "
            "WHERE pool=? AND worker_id=?", (job_id, pool, row[0]))
        c.execute("COMMIT")
    return row[1]


def release(pool: str, job_id: int) -> None:
    with _conn() as c:
        c.execute(
            "UPDATE pool_workers SET status='READY', assigned_job=NULL "
            "WHERE pool=? AND assigned_job=?", (pool, job_id))


def exists(pool: str) -> bool:
    with _conn() as c:
        return c.execute("SELECT 1 FROM pools WHERE name=?",
                         (pool,)).fetchone() is not None
