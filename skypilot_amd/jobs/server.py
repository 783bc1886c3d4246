"""Managed-jobs server entrypoints (reference: sky/jobs/server/core.py
launch:645, queue, cancel).  Controllers run in consolidation mode —
spawned on the API-server host (reference: utils/controller_utils.py:1422)
— and recurse into execution.launch for the user's cluster."""
from __future__ import annotations

import os
import signal
import subprocess
import sys
from typing import Any, Dict, List, Optional

from skypilot_amd.jobs import state


def launch(task: Dict[str, Any], name: Optional[str] = None
           ) -> Dict[str, Any]:
    from skypilot_amd.task import Task
    Task.from_yaml_config(dict(task))  # validate before persisting
    job_id = state.create(name or task.get("name"), task)
    pkg_root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    env = dict(os.environ)
    env["PYTHONPATH"] = pkg_root + (
        ":" + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
    log = open(state.global_state.root_dir() / f"jobs-controller-{job_id}.log",
               "ab")
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.jobs.controller", str(job_id)],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env=env)
    log.close()
    state.update(job_id, controller_pid=proc.pid)
    return {"job_id": job_id, "controller_pid": proc.pid}


def queue() -> List[Dict[str, Any]]:
    state.reconcile()
    jobs = state.list_jobs()
    for j in jobs:
        j.pop("task", None)
    return jobs


def cancel(job_ids: Optional[List[int]] = None,
           all_jobs: bool = False) -> int:
    jobs = state.list_jobs()
    if not all_jobs and job_ids is not None:
        jobs = [j for j in jobs if j["job_id"] in set(job_ids)]
    n = 0
    for j in jobs:
        if j["status"] in state.TERMINAL:
            continue
        state.set_status(j["job_id"], state.CANCELLED)
        n += 1
        # The controller's monitor loop sees CANCELLED within one poll
        # interval, cancels the cluster job and tears the cluster down.
    return n
