"""Managed-jobs server entrypoints (reference: sky/jobs/server/core.py
launch:645, queue, cancel).  Controllers run in consolidation mode —
spawned on the API-server host (reference: utils/controller_utils.py:1422)
— and recurse into execution.launch for the user's cluster."""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from skypilot_amd.jobs import state


def launch(task: Dict[str, Any], name: Optional[str] = None
           ) -> Dict[str, Any]:
    from skypilot_amd.task import Task
    if task.get("pool"):
        from skypilot_amd.jobs import pools
        if not pools.exists(task["pool"]):
            raise ValueError(f"pool {task['pool']!r} does not exist; "
                             "create it with `sky jobs pool apply`")
    if "tasks" in task:  # pipeline: validate each stage
        for t in task["tasks"]:
            Task.from_yaml_config(dict(t))
    else:
        Task.from_yaml_config(dict(task))  # validate before persisting
    job_id = state.create(name or task.get("name"), task)
    # The controller scheduler bounds concurrent controller processes
    # (reference: sky/jobs/scheduler.py:232); beyond the cap, jobs wait
    # as PENDING and start when a running controller finishes.
    from skypilot_amd.jobs import scheduler
    scheduler.maybe_start_controllers()
    pid = (state.get(job_id) or {}).get("controller_pid")
    return {"job_id": job_id, "controller_pid": pid}


def queue() -> List[Dict[str, Any]]:
    state.reconcile()
    jobs = state.list_jobs()
    for j in jobs:
        j.pop("task", None)
    return jobs


def controller_log_path(job_id: int):
    return state.global_state.root_dir() / f"jobs-controller-{job_id}.log"


def logs(job_id: int, tail_lines: int = 200) -> Dict[str, Any]:
    """Controller log + the current cluster job's log tail."""
    out: Dict[str, Any] = {"job_id": job_id}
    p = controller_log_path(job_id)
    if p.exists():
        lines = p.read_text(errors="replace").splitlines()[-tail_lines:]
        out["controller_log"] = "\n".join(lines)
    j = state.get(job_id)
    if j and j.get("cluster_name"):
        from skypilot_amd import global_state
        rec = global_state.get_cluster(j["cluster_name"])
        if rec:
            try:
                from skypilot_amd.backends.pool_backend import PoolBackend
                agent = PoolBackend()._agent(rec["handle"])
                jobs_on_cluster = agent.get_job_queue()
                if jobs_on_cluster:
                    jid = jobs_on_cluster[0]["job_id"]
                    chunks = []
                    for c in agent.tail_logs(jid, follow=False):
                        chunks.append(c)
                        if sum(len(x) for x in chunks) > 1 << 20:
                            break
                    out["task_log"] = b"".join(chunks).decode(
                        errors="replace")[-65536:]
            except Exception as e:  # noqa: BLE001
                out["task_log_error"] = str(e)
    return out


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
