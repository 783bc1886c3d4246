"""Job groups — several tasks launched as ONE unit on shared infra.

Reference: sky/jobs/job_group_networking.py:5-20 (JobGroup: co-located
tasks with cross-task addressing via env vars + /etc/hosts or K8s DNS)
and optimizer.py:optimize_job_group:2035.  On the pool: one cluster is
provisioned with the SUM of the member tasks' GPU demands, every member
is submitted as a concurrent job on it (the agent's GPU-aware scheduler
hands each a disjoint GPU slice), and members address each other
through the injected SKYPILOT_JOBGROUP_* env (same host -> 127.0.0.1).
"""
from __future__ import annotations

import contextlib
import json
import sqlite3
import time
from typing import Any, Dict, List

from skypilot_amd import execution, global_state
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.exceptions import TaskValidationError
from skypilot_amd.task import Task

_SCHEMA = """
CREATE TABLE IF NOT EXISTS job_groups (
    name TEXT PRIMARY KEY,
    cluster TEXT NOT NULL,
    members TEXT NOT NULL,          -- json [{task, job_id}]
    created_at REAL
);
"""


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(global_state.root_dir() / "job_groups.db",
                           timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        with conn:
            yield conn
    finally:
        conn.close()


def launch(name: str, task_cfgs: List[Dict[str, Any]]) -> Dict[str, Any]:
    if not task_cfgs:
        raise TaskValidationError("job group needs at least one task")
    tasks = [Task.from_yaml_config(dict(c)) for c in task_cfgs]
    names = [t.name or f"task{i}" for i, t in enumerate(tasks)]
    if len(set(names)) != len(names):
        raise TaskValidationError("job-group task names must be unique")
    total_gpus = sum(t.resources.accelerator_count * t.num_nodes
                     for t in tasks)
    # One cluster sized so every member can run CONCURRENTLY.
    acc = next((t.resources.accelerators for t in tasks
                if t.resources.accelerators), None)
    shell_cfg: Dict[str, Any] = {"name": f"group-{name}"}
    if total_gpus:
        shell_cfg["resources"] = {"accelerators": f"{acc}:{total_gpus}"}
    shell = Task.from_yaml_config(shell_cfg)
    cluster = f"sky-group-{name}"
    _, handle = execution.launch(shell, cluster, detach_run=True)

    backend = PoolBackend()
    group_env = {
        "SKYPILOT_JOBGROUP_NAME": name,
        "SKYPILOT_JOBGROUP_TASKS": ",".join(names),
        # (comma-separated: newline values don't survive `env | grep`)
        # same-infra co-location: members reach each other on localhost
        "SKYPILOT_JOBGROUP_HOST": "127.0.0.1",
    }
    members = []
    for t, tname in zip(tasks, names):
        t.envs = {**(t.envs or {}), **group_env,
                  "SKYPILOT_JOBGROUP_TASK": tname}
        job_id = backend.execute(handle, t, detach_run=True)
        members.append({"task": tname, "job_id": job_id})
    with _conn() as c:
        c.execute(
            "INSERT OR REPLACE INTO job_groups "
            "(name,cluster,members,created_at) VALUES (?,?,?,?)",
            (name, cluster, json.dumps(members), time.time()))
    return {"group": name, "cluster": cluster, "members": members}


def status(name: str) -> Dict[str, Any]:
    with _conn() as c:
        row = c.execute(
            "SELECT cluster, members FROM job_groups WHERE name=?",
            (name,)).fetchone()
    if row is None:
        raise TaskValidationError(f"no such job group {name!r}")
    cluster, members = row[0], json.loads(row[1])
    rec = global_state.get_cluster(cluster)
    out = []
    if rec:
        agent = PoolBackend()._agent(rec["handle"])
        for m in members:
            j = agent.get_job(m["job_id"])
            out.append({**m, "status": j["status"] if j else "UNKNOWN"})
    else:
        out = [{**m, "status": "CLUSTER_GONE"} for m in members]
    return {"group": name, "cluster": cluster, "members": out}


def down(name: str) -> Dict[str, Any]:
    st = status(name)
    rec = global_state.get_cluster(st["cluster"])
    if rec:
        PoolBackend().teardown(rec["handle"], terminate=True)
    with _conn() as c:
        c.execute("DELETE FROM job_groups WHERE name=?", (name,))
    return {"group": name, "torn_down": rec is not None}
