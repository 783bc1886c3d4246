"""PoolBackend — the one real backend: MI355X pool lifecycle + jobs.

Reference: sky/backends/cloud_vm_ray_backend.py (CloudVmRayBackend,
:3285 _provision, :4173 _exec_code_on_head, :7003/_7044 per-node exec,
:5410 teardown, :5873 set_autostop).  No Ray: job submission goes to the
node agent (skylet equivalent) which gang-launches rank processes
directly (Ray existed only to start N bash processes
with ranks)."""
from __future__ import annotations

import os
import time
import uuid
from pathlib import Path
from typing import Any, Dict, Optional

from skypilot_amd import global_state, provision
from skypilot_amd.agent.client import AgentClient
from skypilot_amd.backends.backend import Backend
from skypilot_amd.data import storage as storage_lib
from skypilot_amd.exceptions import ClusterNotUpError
from skypilot_amd.task import Task
from skypilot_amd.utils.command_runner import LocalProcessCommandRunner


class PoolBackend(Backend):
    def _agent(self, handle: Dict[str, Any]) -> AgentClient:
        port = handle.get("agent_port")
        if not port:
            raise ClusterNotUpError("cluster has no agent")
        return AgentClient(port, token=handle.get("agent_token"))

    # ---- provisioning -----------------------------------------------------
    def provision(self, task: Task, cluster_name: str,
                  retry_until_up: bool = False) -> Dict[str, Any]:
        res = task.resources
        cloud = res.infra or provision.DEFAULT_CLOUD
        existing = global_state.get_cluster(cluster_name)
        existing_handle = existing["handle"] if existing else None
        global_state.add_cluster_event(cluster_name, "PROVISION_START")
        t0 = time.time()
        handle = provision.run_instances(
            cloud, cluster_name, task.num_nodes, res.accelerators,
            res.accelerator_count, existing_handle,
            use_spot=res.use_spot)
        global_state.add_or_update_cluster(
            cluster_name, global_state.UP, handle, res.to_yaml_config())
        global_state.add_cluster_event(
            cluster_name, "PROVISION_DONE",
            f"{time.time() - t0:.2f}s")
        return handle

    # ---- file sync --------------------------------------------------------
    def sync_workdir(self, handle: Dict[str, Any], workdir) -> None:
        dst = Path(handle["cluster_dir"]) / "workdir"
        if isinstance(workdir, dict):
            # git-source workdir (reference: schemas.py {url, ref}):
            # clone/fetch into the cluster workdir and check out `ref`.
            import subprocess
            url = workdir["url"]
            ref = workdir.get("ref")
            if (dst / ".git").exists():
                subprocess.run(["git", "-C", str(dst), "fetch", "origin"],
                               check=True, capture_output=True)
            else:
                dst.mkdir(parents=True, exist_ok=True)
                subprocess.run(["git", "clone", url, str(dst)],
                               check=True, capture_output=True)
            if ref:
                subprocess.run(["git", "-C", str(dst), "checkout", ref],
                               check=True, capture_output=True)
            return
        runner = LocalProcessCommandRunner()
        src = os.path.expanduser(workdir)
        if not src.endswith("/"):
            src += "/"
        runner.rsync(src, str(dst))

    def sync_file_mounts(self, handle: Dict[str, Any],
                         file_mounts: Dict[str, Any]) -> None:
        if not file_mounts:
            return
        storage_lib.execute_file_mounts(handle, file_mounts)

    # ---- setup / exec -----------------------------------------------------
    def setup(self, handle: Dict[str, Any], task: Task) -> None:
        # Setup runs as part of the job driver (SETTING_UP state); nothing
        # to do eagerly here for the local pool.
        pass

    def execute(self, handle: Dict[str, Any], task: Task,
                detach_run: bool = False,
                managed_job_id: Optional[int] = None) -> int:
        agent = self._agent(handle)
        run_cmd = task.run if isinstance(task.run, str) else None
        task_id = (f"sky-{time.strftime('%Y-%m-%d-%H-%M-%S')}-"
                   f"{uuid.uuid4().hex[:6]}")
        spec = {
            "run": run_cmd,
            "setup": task.setup,
            "envs": task.envs,
            "secrets": task.secrets,
            "num_nodes": task.num_nodes,
            "gpus_per_node": task.resources.accelerator_count,
            "workdir": str(Path(handle["cluster_dir"]) / "workdir"),
            "node_ips": handle.get("node_ips"),
            "peer_agents": handle.get("peer_agents"),
            "agent_token": handle.get("agent_token"),
            "master_addr": handle.get("master_addr"),
            "task_id": task_id,
            "managed_job_id": managed_job_id,
            "event_callback": task.event_callback,
            "priority": task.resources.priority or 0,
        }
        job_id = agent.queue_job(spec, name=task.name)
        cluster = _cluster_of(handle)
        if cluster:
            global_state.add_cluster_event(
                cluster, "JOB_SUBMIT", f"job_id={job_id}")
        return job_id

    # ---- lifecycle --------------------------------------------------------
    def teardown(self, handle: Dict[str, Any], terminate: bool = True
                 ) -> None:
        cloud = handle.get("cloud", provision.DEFAULT_CLOUD)
        name = _cluster_of(handle)
        if terminate:
            provision.terminate_instances(cloud, name, handle)
            if name:
                global_state.remove_cluster(name)
        else:
            provision.stop_instances(cloud, name, handle)
            if name:
                global_state.set_cluster_status(name, global_state.STOPPED)
        if name:
            global_state.add_cluster_event(
                name, "TERMINATE" if terminate else "STOP")

    def tail_logs(self, handle: Dict[str, Any], job_id: Optional[int],
                  follow: bool = True):
        agent = self._agent(handle)
        if job_id is None:
            jobs = agent.get_job_queue()
            if not jobs:
                return iter(())
            job_id = jobs[0]["job_id"]
        return agent.tail_logs(job_id, follow=follow)

    def cancel_jobs(self, handle: Dict[str, Any],
                    job_ids: Optional[list] = None) -> int:
        agent = self._agent(handle)
        if job_ids is None:
            return agent.cancel_all()
        n = 0
        for jid in job_ids:
            n += bool(agent.cancel_job(int(jid)))
        return n

    def set_autostop(self, handle: Dict[str, Any], idle_minutes: int,
                     down: bool = False) -> None:
        self._agent(handle).set_autostop(idle_minutes, down)
        name = _cluster_of(handle)
        if name:
            global_state.set_cluster_autostop(name, idle_minutes, down)

    # ---- queries ----------------------------------------------------------
    def job_queue(self, handle: Dict[str, Any]):
        return self._agent(handle).get_job_queue()

    def wait_job(self, handle: Dict[str, Any], job_id: int,
                 timeout: float = 3600):
        return self._agent(handle).wait_job(job_id, timeout=timeout)


def _cluster_of(handle: Dict[str, Any]) -> Optional[str]:
    cdir = handle.get("cluster_dir")
    return Path(cdir).name if cdir else None
