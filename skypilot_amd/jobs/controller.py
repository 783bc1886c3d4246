"""Managed-job controller — one process per managed job.

Reference: sky/jobs/controller.py (JobController:193, _run_one_task:557,
monitor loop :998-1120 detecting preemption and entering RECOVERING).
The controller re-enters the ordinary execution path (`execution.launch`)
to run the user's cluster — the signature controller-as-task recursion
of the reference (SURVEY.md §1 layer 10).  Consolidation mode: the
controller runs on the API-server host (reference:
utils/controller_utils.py:1422), the right default for a one-node pool.

Usage: python -m skypilot_amd.jobs.controller <managed_job_id>
"""
from __future__ import annotations

import os
import sys
import time
import traceback

from skypilot_amd import execution, global_state
from skypilot_amd.agent import job_lib
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.jobs import recovery, state
from skypilot_amd.task import Task

STATUS_CHECK_GAP_SECONDS = float(
    os.environ.get("SKY_AMD_JOBS_POLL_SECONDS", "2.0"))


class JobController:
    def __init__(self, managed_job_id: int):
        self.job_id = managed_job_id
        record = state.get(managed_job_id)
        if record is None:
            raise RuntimeError(f"managed job {managed_job_id} not found")
        self.record = record
        cfg = record["task"]
        self.pool = cfg.get("pool") if isinstance(cfg, dict) else None
        self.worker_cluster = None
        # Pipelines (reference: jobs support chained DAGs): a task config
        # may carry `tasks: [...]` — run sequentially, each with recovery.
        if isinstance(cfg, dict) and "tasks" in cfg:
            self.pipeline = [Task.from_yaml_config(t) for t in cfg["tasks"]]
        else:
            self.pipeline = [Task.from_yaml_config(cfg)]
        self.task = self.pipeline[0]
        jr = self.task.resources.job_recovery
        self.strategy = recovery.make(
            jr.strategy if jr else None,
            jr.max_restarts_on_errors if jr else 0)
        self.cluster_name = f"sky-jobs-{managed_job_id}"
        self.backend = PoolBackend()

    # ------------------------------------------------------------------
    def run(self) -> None:
        state.update(self.job_id, controller_pid=os.getpid(),
                     cluster_name=self.cluster_name)
        try:
            final = state.SUCCEEDED
            for i, task in enumerate(self.pipeline):
                self.task = task
                jr = task.resources.job_recovery
                self.strategy = recovery.make(
                    jr.strategy if jr else None,
                    jr.max_restarts_on_errors if jr else 0)
                final = self._run_with_recovery()
                if final != state.SUCCEEDED:
                    break
            state.set_status(self.job_id, final)
        except BaseException as e:  # noqa: BLE001
            traceback.print_exc()
            state.set_status(self.job_id, state.FAILED_CONTROLLER, str(e))
        finally:
            self._teardown_cluster()

    def _launch_cluster(self):
        state.set_status(self.job_id, state.STARTING)
        if self.pool:
            # Pool mode (reference: `sky jobs pool`): claim a warm worker
            # and just exec — no provisioning on the job path.
            from skypilot_amd.jobs import pools
            nudged = False
            while True:
                cluster = pools.acquire(self.pool, self.job_id)
                if cluster is not None:
                    break
                if not nudged:
                    # queue-length autoscaler (serve/autoscalers.py:1094)
                    try:
                        pools.autoscale(self.pool)
                    except Exception:  # noqa: BLE001
                        pass
                    nudged = True
                me = state.get(self.job_id)
                if me and me["status"] == state.CANCELLED:
                    raise RuntimeError("cancelled while queued for pool")
                time.sleep(1.0)
            self.worker_cluster = cluster
            state.update(self.job_id, cluster_name=cluster)
            job_id, handle = execution.exec_(self.task, cluster,
                                             detach_run=True)
            return job_id, handle
        job_id, handle = execution.launch(
            self.task, self.cluster_name, detach_run=True,
            managed_job_id=self.job_id)
        return job_id, handle

    def _teardown_cluster(self):
        if self.pool:
            from skypilot_amd.jobs import pools
            pools.release(self.pool, self.job_id)
            self.worker_cluster = None
            return
        record = global_state.get_cluster(self.cluster_name)
        if record is not None:
            try:
                self.backend.teardown(record["handle"], terminate=True)
            except Exception:  # noqa: BLE001
                pass

    def _run_with_recovery(self) -> str:
        while True:
            try:
                cluster_job_id, handle = self._launch_cluster()
            except Exception as e:  # noqa: BLE001
                traceback.print_exc()
                return state.FAILED_SETUP if "setup" in str(e).lower() \
                    else state.FAILED
            state.set_status(self.job_id, state.RUNNING)
            outcome = self._monitor(cluster_job_id, handle)
            if outcome == "succeeded":
                return state.SUCCEEDED
            if outcome == "cancelled":
                return state.CANCELLED
            if outcome == "failed_setup":
                return state.FAILED_SETUP
            if outcome == "failed_user":
                # recover_on_exit_codes: these codes always recover and
                # do not consume the restart budget (reference:
                # job_recovery.recover_on_exit_codes).
                jr = self.task.resources.job_recovery
                codes = jr.recover_on_exit_codes if jr else ()
                if self._last_exit_code not in codes:
                    # Other user-code failure: bounded restarts
                    # (resources.job_recovery.max_restarts_on_errors).
                    if not self.strategy.should_restart_on_failure():
                        return state.FAILED
            # preemption / infra failure / bounded user-failure restart:
            state.set_status(self.job_id, state.RECOVERING)
            state.bump_recovery(self.job_id)
            if not self.strategy.keep_placement_first():
                self._teardown_cluster()
            self.strategy.wait_before_retry()

    _last_exit_code = None

    def _monitor(self, cluster_job_id: int, handle) -> str:
        """Poll the cluster job; classify its end state
        (reference: controller.py:998-1120)."""
        while True:
            time.sleep(STATUS_CHECK_GAP_SECONDS)
            me = state.get(self.job_id)
            if me and me["status"] == state.CANCELLED:
                try:
                    self.backend.cancel_jobs(handle, [cluster_job_id])
                except Exception:  # noqa: BLE001
                    pass
                return "cancelled"
            try:
                agent = self.backend._agent(handle)
                job = agent.get_job(cluster_job_id)
            except Exception:  # noqa: BLE001 — cluster gone: preemption
                return "preempted"
            if job is None:
                return "preempted"
            st = job["status"]
            if st == job_lib.SUCCEEDED:
                return "succeeded"
            if st == job_lib.FAILED_SETUP:
                return "failed_setup"
            if st == job_lib.CANCELLED:
                # Cancelled underneath us (not by sky jobs cancel):
                # treat as preemption and recover.
                return "preempted"
            if st in (job_lib.FAILED, job_lib.FAILED_DRIVER):
                self._last_exit_code = job.get("exit_code")
                return "failed_user" if st == job_lib.FAILED else "preempted"


def main():
    job_id = int(sys.argv[1])
    try:
        JobController(job_id).run()
    finally:
        # Free our controller slot: start the next PENDING job's
        # controller (reference: jobs/scheduler.py transitions).
        try:
            from skypilot_amd.jobs import scheduler
            scheduler.maybe_start_controllers()
        except Exception:  # noqa: BLE001
            pass


if __name__ == "__main__":
    main()
