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
    min_workers INTEGER,
    max_workers INTEGER,
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
        cols = [r[1] for r in conn.execute("PRAGMA table_info(pools)")]
        if "min_workers" not in cols:  # pre-autoscaler databases
            conn.execute("ALTER TABLE pools ADD COLUMN min_workers INTEGER")
            conn.execute("ALTER TABLE pools ADD COLUMN max_workers INTEGER")
        with conn:
            yield conn
    finally:
        conn.close()


def apply(name: str, template: Dict[str, Any], num_workers: int,
          min_workers: Optional[int] = None,
          max_workers: Optional[int] = None) -> Dict[str, Any]:
    """Create (or resize) a pool: provision warm clusters running only
    the template's setup.  With min/max_workers the pool autoscales on
    managed-job queue length (reference: serve/autoscalers.py:1094
    QueueLengthAutoscaler); num_workers is then the initial size."""
    tmpl = dict(template)
    tmpl.pop("run", None)  # workers are warm, not running anything
    Task.from_yaml_config(dict(tmpl))
    if min_workers is not None:
        num_workers = max(min_workers, min(num_workers,
                                           max_workers or num_workers))
    with _conn() as c:
        c.execute(
            "INSERT INTO pools (name,template,num_workers,min_workers,"
            "max_workers,created_at) "
            "VALUES (?,?,?,?,?,?) ON CONFLICT(name) DO UPDATE SET "
            "template=excluded.template, num_workers=excluded.num_workers,"
            "min_workers=excluded.min_workers, "
            "max_workers=excluded.max_workers",
            (name, json.dumps(tmpl), num_workers, min_workers, max_workers,
             time.time()))
    workers = [_add_worker(name, tmpl, i) for i in range(num_workers)]
    return {"pool": name, "workers": workers}


def _add_worker(name: str, tmpl: Dict[str, Any], worker_id: int) -> str:
    cluster = f"sky-pool-{name}-{worker_id}"
    task = Task.from_yaml_config(dict(tmpl))
    execution.launch(task, cluster, detach_run=True)
    with _conn() as c:
        c.execute(
            "INSERT INTO pool_workers (pool,worker_id,cluster_name,"
            "status) VALUES (?,?,?,?) ON CONFLICT(pool,worker_id) "
            "DO UPDATE SET status='READY', assigned_job=NULL",
            (name, worker_id, cluster, "READY"))
    return cluster


def _queue_length(name: str) -> int:
    """Managed jobs queued for this pool: STARTING, targeting this
    pool, and not yet holding a worker."""
    from skypilot_amd.jobs import state as jobs_state
    with _conn() as c:
        assigned = {r[0] for r in c.execute(
            "SELECT assigned_job FROM pool_workers WHERE pool=? AND "
            "assigned_job IS NOT NULL", (name,))}
    n = 0
    for j in jobs_state.list_jobs():
        if j["status"] != jobs_state.STARTING:
            continue
        if j["job_id"] in assigned:
            continue
        task = j.get("task") or {}
        if task.get("pool") == name:
            n += 1
    return n


def autoscale(name: Optional[str] = None) -> Dict[str, Any]:
    """Queue-length autoscaling pass: target = clamp(busy + queued,
    min_workers, max_workers).  Scale-down only removes idle READY
    workers and only when nothing is queued."""
    backend = PoolBackend()
    actions: Dict[str, Any] = {}
    with _conn() as c:
        pool_rows = c.execute(
            "SELECT name, template, min_workers, max_workers FROM pools "
            "WHERE min_workers IS NOT NULL").fetchall()
    for pname, tmpl_json, mn, mx in pool_rows:
        if name and pname != name:
            continue
        tmpl = json.loads(tmpl_json)
        with _conn() as c:
            workers = c.execute(
                "SELECT worker_id, cluster_name, status FROM pool_workers "
                "WHERE pool=?", (pname,)).fetchall()
        busy = sum(1 for w in workers if w[2] == "BUSY")
        live = [w for w in workers if w[2] in ("READY", "BUSY")]
        queued = _queue_length(pname)
        target = max(mn, min(busy + queued, mx if mx else busy + queued))
        act = {"queued": queued, "busy": busy, "live": len(live),
               "target": target, "added": 0, "removed": 0}
        if target > len(live):
            used = {w[0] for w in workers}
            wid = 0
            for _ in range(target - len(live)):
                while wid in used:
                    wid += 1
                used.add(wid)
                _add_worker(pname, tmpl, wid)
                act["added"] += 1
        elif target < len(live) and queued == 0:
            idle = [w for w in workers if w[2] == "READY"]
            for w in idle[: len(live) - target]:
                rec = global_state.get_cluster(w[1])
                if rec:
                    try:
                        backend.teardown(rec["handle"], terminate=True)
                    except Exception:  # noqa: BLE001
                        pass
                with _conn() as c:
                    c.execute(
                        "DELETE FROM pool_workers WHERE pool=? AND "
                        "worker_id=?", (pname, w[0]))
                act["removed"] += 1
        actions[pname] = act
    return actions


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
