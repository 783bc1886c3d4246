"""On-cluster job table + scheduler primitives (skylet equivalent).

Reference: sky/skylet/job_lib.py — JobStatus lifecycle (:156), FIFO
scheduler (:278), driver-PID liveness (:833), is_cluster_idle (:1017).
Here the job table is a per-cluster SQLite DB under the cluster runtime
dir; the scheduler is GPU-aware (allocates specific MI355X indices to
jobs) instead of Ray placement groups.
"""
from __future__ import annotations

import json
import os
import signal
import sqlite3
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

# Job lifecycle (reference: job_lib.py:156-196).
INIT = "INIT"
PENDING = "PENDING"
SETTING_UP = "SETTING_UP"
RUNNING = "RUNNING"
SUCCEEDED = "SUCCEEDED"
FAILED = "FAILED"
FAILED_SETUP = "FAILED_SETUP"
FAILED_DRIVER = "FAILED_DRIVER"
CANCELLED = "CANCELLED"

TERMINAL = {SUCCEEDED, FAILED, FAILED_SETUP, FAILED_DRIVER, CANCELLED}
NONTERMINAL = {INIT, PENDING, SETTING_UP, RUNNING}

_SCHEMA = """
CREATE TABLE IF NOT EXISTS jobs (
    job_id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT,
    status TEXT NOT NULL,
    submitted_at REAL,
    started_at REAL,
    ended_at REAL,
    spec TEXT NOT NULL,
    driver_pid INTEGER,
    gpus TEXT DEFAULT '[]',
    exit_code INTEGER,
    log_dir TEXT
);
"""


class JobTable:
    def __init__(self, cluster_dir: str):
        self.cluster_dir = Path(cluster_dir)
        self.cluster_dir.mkdir(parents=True, exist_ok=True)
        self.db_path = self.cluster_dir / "jobs.db"

    def _conn(self):
        import contextlib

        @contextlib.contextmanager
        def cm():
            conn = sqlite3.connect(self.db_path, timeout=30)
            try:
                conn.execute("PRAGMA busy_timeout=30000")
                conn.execute("PRAGMA journal_mode=WAL")
                conn.executescript(_SCHEMA)
                with conn:
                    yield conn
            finally:
                conn.close()
        return cm()

    def add_job(self, name: Optional[str], spec: Dict[str, Any]) -> int:
        with self._conn() as c:
            cur = c.execute(
                "INSERT INTO jobs (name,status,submitted_at,spec) "
                "VALUES (?,?,?,?)",
                (name, PENDING, time.time(), json.dumps(spec)))
            return cur.lastrowid

    def get(self, job_id: int) -> Optional[Dict[str, Any]]:
        with self._conn() as c:
            row = c.execute("SELECT * FROM jobs WHERE job_id=?",
                            (job_id,)).fetchone()
            cols = [d[0] for d in c.execute(
                "SELECT * FROM jobs LIMIT 0").description]
        if row is None:
            return None
        d = dict(zip(cols, row))
        d["spec"] = json.loads(d["spec"])
        d["gpus"] = json.loads(d["gpus"] or "[]")
        return d

    def list(self, limit: int = 1000) -> List[Dict[str, Any]]:
        with self._conn() as c:
            cols = [d[0] for d in c.execute(
                "SELECT * FROM jobs LIMIT 0").description]
            rows = c.execute(
                "SELECT * FROM jobs ORDER BY job_id DESC LIMIT ?",
                (limit,)).fetchall()
        out = []
        for row in rows:
            d = dict(zip(cols, row))
            d["spec"] = json.loads(d["spec"])
            d["gpus"] = json.loads(d["gpus"] or "[]")
            out.append(d)
        return out

    def set_status(self, job_id: int, status: str,
                   exit_code: Optional[int] = None):
        now = time.time()
        with self._conn() as c:
            if status == RUNNING:
                c.execute(
                    "UPDATE jobs SET status=?, started_at=? WHERE job_id=?",
                    (status, now, job_id))
            elif status in TERMINAL:
                c.execute(
                    "UPDATE jobs SET status=?, ended_at=?, exit_code=? "
                    "WHERE job_id=?", (status, now, exit_code, job_id))
            else:
                c.execute("UPDATE jobs SET status=? WHERE job_id=?",
                          (status, job_id))

    def set_driver(self, job_id: int, pid: int, gpus: List[int],
                   log_dir: str):
        with self._conn() as c:
            c.execute(
                "UPDATE jobs SET driver_pid=?, gpus=?, log_dir=? "
                "WHERE job_id=?",
                (pid, json.dumps(gpus), log_dir, job_id))

    def pending_jobs(self) -> List[Dict[str, Any]]:
        """FIFO within priority class (reference: resources.priority)."""
        out = [j for j in reversed(self.list()) if j["status"] == PENDING]
        out.sort(key=lambda j: (-(j["spec"].get("priority") or 0),
                                j["job_id"]))
        return out

    def active_jobs(self) -> List[Dict[str, Any]]:
        return [j for j in self.list() if j["status"] in
                (SETTING_UP, RUNNING)]

    def allocated_gpus(self) -> List[int]:
        out: List[int] = []
        for j in self.active_jobs():
            out.extend(j["gpus"])
        return out

    def is_idle(self) -> bool:
        """reference: job_lib.py:1017 is_cluster_idle."""
        return not any(j["status"] in NONTERMINAL for j in self.list())

    def cancel(self, job_id: int) -> bool:
        j = self.get(job_id)
        if j is None or j["status"] in TERMINAL:
            return False
        # Mark first so the driver's final-status write sees CANCELLED
        # (kill-then-mark would race the driver's own FAILED write).
        self.set_status(job_id, CANCELLED)
        pid = j.get("driver_pid")
        if pid and _proc_matches(pid, b"skypilot_amd.agent.driver"):
            try:
                os.killpg(pid, signal.SIGTERM)
            except (ProcessLookupError, PermissionError):
                try:
                    os.kill(pid, signal.SIGTERM)
                except ProcessLookupError:
                    pass
        return True

    def reconcile(self):
        """Mark jobs whose driver died without reporting as FAILED_DRIVER
        and reap their orphaned node processes (reference:
        job_lib.py:833/:850 + skylet/subprocess_daemon.py)."""
        for j in self.active_jobs():
            pid = j.get("driver_pid")
            if pid and not _pid_alive(pid):
                self.set_status(j["job_id"], FAILED_DRIVER, exit_code=-1)
                tag = (f"SKYPILOT_INTERNAL_JOB_ID={j['job_id']}"
                       .encode())
                for npid in j["spec"].get("node_pids", []):
                    # PID-recycling guard: only kill a recorded pgid if
                    # the live process still carries THIS job's env tag
                    # (a long-dead pid number can be reused by an
                    # unrelated process — killing it blind took out
                    # innocent processes in CI).
                    if not _proc_matches(npid, tag):
                        continue
                    try:
                        os.killpg(npid, signal.SIGTERM)
                    except (ProcessLookupError, PermissionError):
                        pass


def _proc_matches(pid: int, needle: bytes) -> bool:
    """True iff /proc/<pid>'s cmdline or environ contains `needle` —
    guards every kill-by-recorded-pid against PID recycling."""
    for f in ("cmdline", "environ"):
        try:
            with open(f"/proc/{pid}/{f}", "rb") as fh:
                if needle in fh.read():
                    return True
        except OSError:
            pass
    return False


def _pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False
    except PermissionError:
        return True
