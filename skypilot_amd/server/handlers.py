"""Request handlers — the server-side bodies of each API call.

Reference: sky/server/server.py routes (:2034-2555) delegating into
sky/core.py + sky/execution.py.  Each handler runs inside a per-request
process (see executor.py); returns must be JSON-serializable.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from skypilot_amd import core, execution
from skypilot_amd.server.executor import LONG, SHORT, register
from skypilot_amd.task import Task


@register("launch", LONG)
def launch(task: Dict[str, Any], cluster_name: Optional[str] = None,
           down: bool = False, idle_minutes_to_autostop: Optional[int] = None,
           detach_run: bool = True, retry_until_up: bool = False,
           dryrun: bool = False) -> Dict[str, Any]:
    from skypilot_amd import admin_policy
    task = admin_policy.apply(task, cluster_name, "launch")
    t = Task.from_yaml_config(task)
    job_id, handle = execution.launch(
        t, cluster_name, detach_run=detach_run, down=down,
        idle_minutes_to_autostop=idle_minutes_to_autostop,
        retry_until_up=retry_until_up, dryrun=dryrun)
    return {"job_id": job_id, "cluster_name": cluster_name,
            "handle": handle}


@register("exec", LONG)
def exec_(task: Dict[str, Any], cluster_name: str,
          detach_run: bool = True) -> Dict[str, Any]:
    t = Task.from_yaml_config(task)
    job_id, handle = execution.exec_(t, cluster_name, detach_run=detach_run)
    return {"job_id": job_id, "cluster_name": cluster_name}


@register("status", SHORT)
def status(cluster_names: Optional[List[str]] = None,
           all_workspaces: bool = False,
           refresh: bool = False) -> List[Dict[str, Any]]:
    return core.status(cluster_names, refresh,
                       all_workspaces=all_workspaces)


@register("start", LONG)
def start(cluster_name: str) -> Dict[str, Any]:
    return core.start(cluster_name)


@register("stop", LONG)
def stop(cluster_name: str) -> None:
    core.stop(cluster_name)


@register("down", LONG)
def down(cluster_name: str) -> None:
    core.down(cluster_name)


@register("autostop", SHORT)
def autostop(cluster_name: str, idle_minutes: int, down: bool = False
             ) -> None:
    core.autostop(cluster_name, idle_minutes, down)


@register("queue", SHORT)
def queue(cluster_name: str) -> List[Dict[str, Any]]:
    return core.queue(cluster_name)


@register("cancel", SHORT)
def cancel(cluster_name: str, job_ids: Optional[List[int]] = None,
           all_jobs: bool = False) -> int:
    return core.cancel(cluster_name, job_ids, all_jobs)


@register("job_status", SHORT)
def job_status(cluster_name: str, job_id: int):
    return core.job_status(cluster_name, job_id)


@register("cost_report", SHORT)
def cost_report():
    """reference: sky cost-report (GPU-hours on a local pool)."""
    return core.cost_report()


@register("check", SHORT)
def check() -> Dict[str, Any]:
    return core.check()


@register("show_gpus", SHORT)
def show_gpus() -> List[Dict[str, Any]]:
    return core.show_gpus()


@register("cluster_events", SHORT)
def cluster_events(cluster_name: str) -> List[Dict[str, Any]]:
    return core.cluster_events(cluster_name)


@register("storage_list", SHORT)
def storage_list():
    from skypilot_amd.data import storage as st
    return st.list_storage()


@register("storage_sync", SHORT)
def storage_sync(name: str) -> Dict[str, Any]:
    """Push a store to its S3-compatible remote (rclone)."""
    from skypilot_amd.data import storage as storage_mod
    return {"synced_to": storage_mod.sync_to_remote(name)}


@register("storage_delete", SHORT)
def storage_delete(name: str) -> bool:
    from skypilot_amd.data import storage as st
    return st.delete_storage(name)


@register("volumes_list", SHORT)
def volumes_list():
    from skypilot_amd.data import volumes
    return volumes.list_volumes()


@register("volumes_create", SHORT)
def volumes_create(name: str, size_gb: Optional[int] = None):
    from skypilot_amd.data import volumes
    return volumes.create(name, size_gb)


@register("volumes_delete", SHORT)
def volumes_delete(name: str) -> bool:
    from skypilot_amd.data import volumes
    return volumes.delete(name)


@register("recipes_list", SHORT)
def recipes_list():
    from skypilot_amd import recipes
    return recipes.list_recipes()


# ---- managed jobs (controller recursion; see jobs/) -----------------------
@register("jobs_launch", LONG)
def jobs_launch(task: Dict[str, Any], name: Optional[str] = None
                ) -> Dict[str, Any]:
    from skypilot_amd import admin_policy
    task = admin_policy.apply(task, name, "jobs_launch")
    from skypilot_amd.jobs import server as jobs_server
    return jobs_server.launch(task, name)


@register("jobs_queue", SHORT)
def jobs_queue() -> List[Dict[str, Any]]:
    from skypilot_amd.jobs import server as jobs_server
    return jobs_server.queue()


@register("jobs_cancel", SHORT)
def jobs_cancel(job_ids: Optional[List[int]] = None,
                all_jobs: bool = False) -> int:
    from skypilot_amd.jobs import server as jobs_server
    return jobs_server.cancel(job_ids, all_jobs)


@register("jobs_logs", SHORT)
def jobs_logs(job_id: int):
    from skypilot_amd.jobs import server as jobs_server
    return jobs_server.logs(job_id)


@register("jobs_group_launch", LONG)
def jobs_group_launch(name: str, tasks: List[Dict[str, Any]]
                      ) -> Dict[str, Any]:
    """reference: JobGroup (jobs/job_group_networking.py)."""
    from skypilot_amd.jobs import groups
    return groups.launch(name, tasks)


@register("jobs_group_status", SHORT)
def jobs_group_status(name: str) -> Dict[str, Any]:
    from skypilot_amd.jobs import groups
    return groups.status(name)


@register("jobs_group_down", LONG)
def jobs_group_down(name: str) -> Dict[str, Any]:
    from skypilot_amd.jobs import groups
    return groups.down(name)


@register("jobs_pool_apply", LONG)
def jobs_pool_apply(name: str, template: Dict[str, Any],
                    num_workers: int = 2,
                    min_workers: Optional[int] = None,
                    max_workers: Optional[int] = None):
    from skypilot_amd.jobs import pools
    return pools.apply(name, template, num_workers,
                       min_workers=min_workers, max_workers=max_workers)


@register("jobs_pool_status", SHORT)
def jobs_pool_status(name: Optional[str] = None):
    from skypilot_amd.jobs import pools
    return pools.status(name)


@register("jobs_pool_down", LONG)
def jobs_pool_down(name: str) -> int:
    from skypilot_amd.jobs import pools
    return pools.down(name)


# ---- serve ----------------------------------------------------------------
@register("serve_up", LONG)
def serve_up(task: Dict[str, Any], service_name: str) -> Dict[str, Any]:
    from skypilot_amd.serve import server as serve_server
    return serve_server.up(task, service_name)


@register("serve_update", LONG)
def serve_update(task: Dict[str, Any], service_name: str) -> Dict[str, Any]:
    from skypilot_amd.serve import server as serve_server
    return serve_server.update(task, service_name)


@register("serve_down", LONG)
def serve_down(service_name: str) -> None:
    from skypilot_amd.serve import server as serve_server
    serve_server.down(service_name)


@register("serve_logs", SHORT)
def serve_logs(service_name: str, replica_id: Optional[int] = None):
    from skypilot_amd.serve import server as serve_server
    return serve_server.logs(service_name, replica_id)


@register("serve_status", SHORT)
def serve_status(service_name: Optional[str] = None):
    from skypilot_amd.serve import server as serve_server
    return serve_server.status(service_name)
