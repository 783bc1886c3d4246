"""Execution — the launch/exec stage machine.

Reference: sky/execution.py:48-59 (Stage enum CLONE_DISK→OPTIMIZE→
PROVISION→SYNC_WORKDIR→SYNC_FILE_MOUNTS→SETUP→PRE_EXEC→EXEC→DOWN),
:343 _execute_dag, :688 launch, :925 exec (runs only SYNC_WORKDIR+EXEC).
"""
from __future__ import annotations

import enum
import time
import uuid
from typing import Any, Dict, Optional, Tuple

from skypilot_amd import global_state
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.dag import to_dag
from skypilot_amd.exceptions import (ClusterDoesNotExist,
                                     ClusterNotUpError,
                                     ResourcesUnavailableError)
from skypilot_amd.optimizer import Optimizer
from skypilot_amd.utils.timeline import event as timeline_event


class Stage(enum.Enum):
    OPTIMIZE = enum.auto()
    PROVISION = enum.auto()
    SYNC_WORKDIR = enum.auto()
    SYNC_FILE_MOUNTS = enum.auto()
    SETUP = enum.auto()
    PRE_EXEC = enum.auto()
    EXEC = enum.auto()
    DOWN = enum.auto()

ALL_STAGES = list(Stage)


def _generate_cluster_name() -> str:
    return f"sky-{uuid.uuid4().hex[:8]}"


@timeline_event("execution.execute")
def _execute(task_or_dag, cluster_name: Optional[str], stages,
             detach_run: bool = True, down: bool = False,
             idle_minutes_to_autostop: Optional[int] = None,
             managed_job_id: Optional[int] = None,
             retry_until_up: bool = False
             ) -> Tuple[Optional[int], Dict[str, Any]]:
    dag = to_dag(task_or_dag)
    assert len(dag.tasks) == 1, "chained DAGs execute task-by-task"
    task = dag.tasks[0]
    cluster_name = cluster_name or _generate_cluster_name()
    backend = PoolBackend()

    if Stage.OPTIMIZE in stages:
        Optimizer.optimize(dag)

    handle: Optional[Dict[str, Any]] = None
    if Stage.PROVISION in stages:
        # Candidate failover (reference: RetryingVmProvisioner over
        # any_of/ordered resources): try each candidate in order until
        # one provisions.
        cands = task.resources.candidates or (task.resources,)
        last_err: Optional[Exception] = None
        for cand in cands:
            task.resources = cand
            try:
                handle = backend.provision(task, cluster_name)
                last_err = None
                break
            except ResourcesUnavailableError as e:
                last_err = e
                global_state.add_cluster_event(
                    cluster_name, "PROVISION_FAILOVER",
                    f"{cand.accelerators or 'cpu'}:"
                    f"{cand.accelerator_count}: {e}")
        if last_err is not None:
            if retry_until_up:
                # reference: `sky launch --retry-until-up` — loop the
                # whole candidate list with backoff until capacity
                # appears.
                delay = 5.0
                while last_err is not None:
                    time.sleep(delay)
                    delay = min(delay * 2, 60.0)
                    for cand in cands:
                        task.resources = cand
                        try:
                            handle = backend.provision(task, cluster_name)
                            last_err = None
                            break
                        except ResourcesUnavailableError as e:
                            last_err = e
            if last_err is not None:
                raise last_err
    else:
        record = global_state.get_cluster(cluster_name)
        if record is None:
            raise ClusterDoesNotExist(f"cluster {cluster_name!r} not found")
        if record["status"] != global_state.UP:
            raise ClusterNotUpError(
                f"cluster {cluster_name!r} is {record['status']}")
        handle = record["handle"]

    if Stage.SYNC_WORKDIR in stages and task.workdir:
        backend.sync_workdir(handle, task.workdir)
    if Stage.SYNC_FILE_MOUNTS in stages and task.file_mounts:
        backend.sync_file_mounts(handle, task.file_mounts)
    if Stage.SETUP in stages:
        backend.setup(handle, task)

    if Stage.PRE_EXEC in stages:
        autostop = idle_minutes_to_autostop
        if autostop is None and task.resources.autostop:
            autostop = task.resources.autostop.idle_minutes
            down = down or task.resources.autostop.down
        if autostop is not None:
            backend.set_autostop(handle, autostop, down)

    job_id = None
    if Stage.EXEC in stages and (task.run or task.setup):
        job_id = backend.execute(handle, task, detach_run,
                                 managed_job_id=managed_job_id)
        if not detach_run:
            backend.wait_job(handle, job_id)

    if Stage.DOWN in stages and down and job_id is not None:
        backend.wait_job(handle, job_id)
        backend.teardown(handle, terminate=True)
    return job_id, handle


def launch(task, cluster_name: Optional[str] = None, *,
           detach_run: bool = True, down: bool = False,
           idle_minutes_to_autostop: Optional[int] = None,
           managed_job_id: Optional[int] = None,
           retry_until_up: bool = False, dryrun: bool = False):
    """reference: sky/execution.py:688 (launch; --dryrun stops after
    OPTIMIZE and returns the placement plan)."""
    if dryrun:
        dag = to_dag(task)
        Optimizer.optimize(dag)
        t = dag.tasks[0]
        return None, {
            "dryrun": True,
            "cluster_name": cluster_name,
            "resources": t.resources.to_yaml_config(),
            "estimated_hourly_cost": getattr(
                t, "estimated_hourly_cost", None),
            "candidates": [c.to_yaml_config()
                           for c in (t.resources.candidates
                                     or (t.resources,))],
        }
    return _execute(task, cluster_name, ALL_STAGES, detach_run, down,
                    idle_minutes_to_autostop, managed_job_id,
                    retry_until_up)


def exec_(task, cluster_name: str, *, detach_run: bool = True):
    """reference: sky/execution.py:925 (exec: SYNC_WORKDIR + EXEC only)."""
    return _execute(task, cluster_name,
                    [Stage.SYNC_WORKDIR, Stage.EXEC], detach_run)
