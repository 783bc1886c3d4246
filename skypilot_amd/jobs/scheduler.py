"""Managed-jobs controller scheduler — bounds concurrent controllers.

Reference: sky/jobs/scheduler.py (maybe_start_controllers:232,
submit_jobs:329) limits parallel controller processes by system
resources so 100 `sky jobs launch` calls don't fork 100 controllers.
Here the cap defaults to half the CPUs (min 2, max 16) and is
overridable with SKY_AMD_MAX_CONTROLLERS.  PENDING jobs without a
controller wait in the state DB; a finishing controller (and the server
background daemon) call maybe_start_controllers() to drain the queue.
"""
from __future__ import annotations

import os
import subprocess
import sys
from typing import Optional

from skypilot_amd.jobs import state

_LOCK_NAME = "jobs-scheduler.lock"


def max_controllers() -> int:
    env = os.environ.get("SKY_AMD_MAX_CONTROLLERS")
    if env:
        return max(1, int(env))
    ncpu = os.cpu_count() or 4
    return max(2, min(16, ncpu // 2))


def _pid_alive(pid: Optional[int]) -> bool:
    if not pid:
        return False
    try:
        os.kill(pid, 0)
        with open(f"/proc/{pid}/stat") as f:
            return f.read().rsplit(")", 1)[1].split()[0] != "Z"
    except (OSError, IndexError):
        return False


def _spawn_controller(job_id: int) -> int:
    pkg_root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    env = dict(os.environ)
    env["PYTHONPATH"] = pkg_root + (
        ":" + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
    log = open(state.global_state.root_dir() /
               f"jobs-controller-{job_id}.log", "ab")
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.jobs.controller",
         str(job_id)],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env=env)
    log.close()
    state.update(job_id, controller_pid=proc.pid)
    return proc.pid


def maybe_start_controllers() -> int:
    """Start controllers for waiting jobs up to the cap; returns how
    many were started.  Serialized by a file lock (multiple request
    processes may race here)."""
    import fcntl
    lock_path = state.global_state.root_dir() / _LOCK_NAME
    with open(lock_path, "w") as lk:
        fcntl.flock(lk, fcntl.LOCK_EX)
        jobs = state.list_jobs()
        running = sum(
            1 for j in jobs
            if j["status"] not in state.TERMINAL
            and _pid_alive(j.get("controller_pid")))
        cap = max_controllers()
        started = 0
        for j in jobs:
            if running + started >= cap:
                break
            if (j["status"] == state.PENDING
                    and not j.get("controller_pid")):
                _spawn_controller(j["job_id"])
                started += 1
        return started
