"""`sky` CLI (reference: sky/client/cli/command.py — launch:1318,
exec:1563, status:2090, jobs_launch:5787, serve_up:7282, api_start:7912).

Run as `python -m skypilot_amd.cli ...` or via the repo's `bin/sky`.
"""
from __future__ import annotations

import json

import click

from skypilot_amd.client import sdk
from skypilot_amd.task import Task


def _load_task(entrypoint: str, env: tuple, overrides: dict) -> Task:
    """YAML file -> Task, with CLI overrides merged (reference:
    _make_task_or_dag_from_entrypoint_with_overrides, command.py:945)."""
    import yaml as _yaml
    if entrypoint.endswith((".yaml", ".yml")):
        with open(entrypoint) as f:
            cfg = _yaml.safe_load(f) or {}
    else:
        cfg = {"run": entrypoint}
    env_overrides = dict(kv.split("=", 1) for kv in env)
    res = cfg.setdefault("resources", {}) or {}
    for k, v in overrides.items():
        if v is not None:
            res[k] = v
    cfg["resources"] = res
    if overrides.get("_num_nodes"):
        cfg["num_nodes"] = overrides["_num_nodes"]
    cfg.pop("_num_nodes", None)
    res.pop("_num_nodes", None)
    return Task.from_yaml_config(cfg, env_overrides)


def _print_result(result):
    click.echo(json.dumps(result, indent=2, default=str))


@click.group()
def cli():
    """skypilot-amd: run tasks on an MI355X pool."""


# ---------------------------------------------------------------------------
@cli.command()
@click.argument("entrypoint")
@click.option("--cluster", "-c", default=None)
@click.option("--gpus", default=None, help='e.g. "MI355X:8"')
@click.option("--num-nodes", type=int, default=None)
@click.option("--env", multiple=True, help="KEY=VALUE")
@click.option("--down", is_flag=True, help="tear down after the job")
@click.option("--idle-minutes-to-autostop", "-i", type=int, default=None)
@click.option("--retry-until-up", "-r", is_flag=True,
              help="retry provisioning with backoff until capacity frees")
@click.option("--detach-run", "-d", is_flag=True,
              help="don't stream logs after submit")
@click.option("--async", "async_", is_flag=True,
              help="print request id and return")
@click.option("--dryrun", is_flag=True,
              help="optimize + print the placement plan, provision nothing")
def launch(entrypoint, cluster, gpus, num_nodes, env, down,
           idle_minutes_to_autostop, retry_until_up, detach_run, async_,
           dryrun):
    """Launch a task (provision + sync + setup + run)."""
    task = _load_task(entrypoint, env,
                      {"accelerators": gpus, "_num_nodes": num_nodes})
    if dryrun:
        plan = sdk.get(sdk.launch(task, cluster, dryrun=True))
        _print_result(plan["handle"])
        return
    rid = sdk.launch(task, cluster, down=down, retry_until_up=retry_until_up,
                     idle_minutes_to_autostop=idle_minutes_to_autostop)
    if async_:
        click.echo(rid)
        return
    result = sdk.stream_and_get(rid)
    click.echo(f"Job submitted: cluster={result.get('cluster_name')} "
               f"job_id={result.get('job_id')}")
    if not detach_run and result.get("job_id") is not None:
        sdk.tail_logs(result.get("cluster_name") or cluster,
                      result["job_id"])


@cli.command("dashboard")
def dashboard_cmd():
    """Print (and try to open) the web dashboard URL (reference: sky
    dashboard)."""
    from skypilot_amd.server.app import server_url
    url = server_url() + "/dashboard"
    click.echo(url)
    try:
        import webbrowser
        webbrowser.open(url)
    except Exception:  # noqa: BLE001 — headless is fine, URL printed
        pass


@cli.command("ssh")
@click.argument("cluster")
@click.option("--cmd", default=None,
              help="run this command instead of an interactive shell")
def ssh_cmd(cluster, cmd):
    """Open an interactive shell on the cluster head (reference: sky
    ssh — tunneled through the API server, no direct node access
    needed)."""
    raise SystemExit(sdk.ssh_shell(cluster, cmd=cmd))


@cli.command("exec")
@click.argument("cluster")
@click.argument("entrypoint")
@click.option("--env", multiple=True)
@click.option("--gpus", default=None)
@click.option("--async", "async_", is_flag=True)
def exec_cmd(cluster, entrypoint, env, gpus, async_):
    """Run a task on an existing cluster (no provision/setup)."""
    task = _load_task(entrypoint, env, {"accelerators": gpus})
    rid = sdk.exec(task, cluster)
    if async_:
        click.echo(rid)
        return
    result = sdk.get(rid)
    click.echo(f"Job submitted: job_id={result.get('job_id')}")
    sdk.tail_logs(cluster, result["job_id"])


@cli.command()
@click.argument("clusters", nargs=-1)
@click.option("--refresh", "-r", is_flag=True)
@click.option("--all-workspaces", "-u", is_flag=True)
def status(clusters, refresh, all_workspaces):
    """Show clusters (optionally filtered by name; -u for every
    workspace)."""
    records = sdk.get(sdk.status(cluster_names=list(clusters) or None,
                                 refresh=refresh,
                                 all_workspaces=all_workspaces))
    if not records:
        click.echo("No existing clusters.")
        return
    fmt = "{:<16} {:<9} {:<22} {:<8} {:<10}"
    click.echo(fmt.format("NAME", "STATUS", "RESOURCES", "GPUS",
                          "WORKSPACE"))
    for r in records:
        h = r["handle"]
        res = f"{h.get('num_nodes', 1)}x local"
        if h.get("gpus_per_node"):
            res += f" MI355X:{h['gpus_per_node']}"
        click.echo(fmt.format(r["name"], r["status"], res,
                              ",".join(map(str, h.get("gpu_ids", []))),
                              r.get("workspace", "default")))


@cli.command()
@click.argument("clusters", nargs=-1, required=True)
def start(clusters):
    """Restart a stopped cluster (same GPU lease)."""
    for cl in clusters:
        sdk.get(sdk.start(cl))
        click.echo(f"Cluster {cl} started.")


@cli.command()
@click.argument("clusters", nargs=-1)
@click.option("--all", "-a", "all_", is_flag=True)
def stop(clusters, all_):
    """Stop a cluster (keep its state dir for sky start)."""
    if all_:
        clusters = [r["name"] for r in sdk.get(sdk.status())
                    if r["status"] == "UP"]
    if not clusters:
        click.echo("No clusters." if all_ else
                   "Pass cluster names or --all.")
        return
    for cl in clusters:
        sdk.get(sdk.stop(cl))
        click.echo(f"Cluster {cl} stopped.")


@cli.command()
@click.argument("clusters", nargs=-1)
@click.option("--all", "-a", "all_", is_flag=True,
              help="tear down every cluster in the workspace")
@click.option("--yes", "-y", is_flag=True)
def down(clusters, all_, yes):
    """Tear the cluster down and release its GPUs."""
    if all_:
        clusters = [r["name"] for r in sdk.get(sdk.status())]
    if not clusters:
        click.echo("No clusters." if all_ else
                   "Pass cluster names or --all.")
        return
    # only mass teardown prompts; single-name down stays scriptable
    if all_ and not yes and not click.confirm(
            f"Tear down {', '.join(clusters)}?"):
        return
    for cl in clusters:
        sdk.get(sdk.down(cl))
        click.echo(f"Cluster {cl} terminated.")


@cli.command()
@click.argument("cluster")
@click.option("--idle-minutes", "-i", type=int, default=None)
@click.option("--down", is_flag=True)
@click.option("--cancel", is_flag=True,
              help="cancel a scheduled autostop (reference: sky "
                   "autostop --cancel)")
def autostop(cluster, idle_minutes, down, cancel):
    """Stop the cluster after N idle minutes (-i), or --cancel."""
    if cancel:
        sdk.get(sdk.autostop(cluster, -1, False))
        click.echo(f"Autostop cancelled on {cluster}.")
        return
    if idle_minutes is None:
        raise click.UsageError("pass -i <minutes> or --cancel")
    sdk.get(sdk.autostop(cluster, idle_minutes, down))
    click.echo(f"Autostop set on {cluster}: {idle_minutes}m "
               f"({'down' if down else 'stop'})")


@cli.command()
@click.argument("cluster")
def queue(cluster):
    """Show a cluster's job queue."""
    jobs = sdk.get(sdk.queue(cluster))
    fmt = "{:<6} {:<18} {:<12} {:<10}"
    click.echo(fmt.format("ID", "NAME", "STATUS", "GPUS"))
    for j in jobs:
        click.echo(fmt.format(j["job_id"], str(j.get("name") or "-"),
                              j["status"],
                              ",".join(map(str, j.get("gpus", [])))))


@cli.command()
@click.argument("cluster")
@click.argument("job_id", type=int, required=False)
@click.option("--no-follow", is_flag=True)
@click.option("--status", "status_only", is_flag=True,
              help="print the job's status instead of its logs "
                   "(exit 0 iff SUCCEEDED; reference: sky logs --status)")
def logs(cluster, job_id, no_follow, status_only):
    """Stream a job's logs."""
    if status_only:
        j = sdk.get(sdk.job_status(cluster, job_id if job_id else 1))
        if j is None:
            click.echo("no such job")
            raise SystemExit(1)
        click.echo(j["status"])
        raise SystemExit(0 if j["status"] == "SUCCEEDED" else 1)
    sdk.tail_logs(cluster, job_id, follow=not no_follow)


@cli.command()
@click.argument("cluster")
@click.argument("job_ids", nargs=-1, type=int)
@click.option("--all", "all_jobs", is_flag=True)
def cancel(cluster, job_ids, all_jobs):
    """Cancel queued/running jobs on a cluster."""
    n = sdk.get(sdk.cancel(cluster, list(job_ids) or None, all_jobs))
    click.echo(f"Cancelled {n} job(s).")


@cli.command()
def check():
    """Verify pool credentials/capabilities (local, ssh, k8s)."""
    _print_result(sdk.get(sdk.check()))


@cli.command("show-gpus")
def show_gpus():
    """List pool GPUs, topology and current leases."""
    gpus = sdk.get(sdk.show_gpus())
    fmt = "{:<6} {:<10} {:<10} {:<6} {:<12}"
    click.echo(fmt.format("GPU", "NAME", "MEM_GB", "NUMA", "USED_BY"))
    for g in gpus:
        click.echo(fmt.format(g["index"], g["name"], g["memory_gb"],
                              g["numa_node"], str(g.get("used_by") or "-")))


# ---- api ------------------------------------------------------------------
@cli.group()
def api():
    """API server management."""


@api.command("start")
def api_start():
    started = sdk.api_start()
    click.echo("API server started." if started
               else "API server already running.")


@api.command("stop")
def api_stop():
    click.echo("Stopped." if sdk.api_stop() else "Not running.")


@api.command("status")
def api_status():
    click.echo("healthy" if sdk.api_healthy() else "not running")


@api.command("logs")
@click.option("--tail", "-n", type=int, default=100)
def api_logs(tail):
    """Show the local API server log (reference: sky api logs)."""
    import os
    path = os.path.expanduser("~/.sky_amd_api.log")
    if not os.path.exists(path):
        click.echo("no server log at ~/.sky_amd_api.log "
                   "(server not auto-started from this client?)")
        return
    with open(path, "rb") as f:
        data = f.read().decode(errors="replace")
    for line in data.splitlines()[-tail:]:
        click.echo(line)


# ---- jobs -----------------------------------------------------------------
@cli.group()
def jobs():
    """Managed jobs (auto-recovery)."""


@jobs.command("launch")
@click.argument("entrypoint")
@click.option("--name", "-n", default=None)
@click.option("--env", multiple=True)
@click.option("--gpus", default=None)
@click.option("--async", "async_", is_flag=True)
def jobs_launch(entrypoint, name, env, gpus, async_):
    task = _load_task(entrypoint, env, {"accelerators": gpus})
    rid = sdk.jobs_launch(task, name)
    if async_:
        click.echo(rid)
        return
    _print_result(sdk.stream_and_get(rid))


@jobs.command("queue")
def jobs_queue():
    rows = sdk.get(sdk.jobs_queue())
    fmt = "{:<6} {:<18} {:<12} {:<8} {:<10}"
    click.echo(fmt.format("ID", "NAME", "STATUS", "RECOV", "CLUSTER"))
    for j in rows:
        click.echo(fmt.format(j["job_id"], str(j.get("name") or "-"),
                              j["status"], j.get("recovery_count", 0),
                              str(j.get("cluster_name") or "-")))


@jobs.command("logs")
@click.argument("job_id", type=int)
def jobs_logs(job_id):
    result = sdk.get(sdk.jobs_logs(job_id))
    if result.get("controller_log"):
        click.echo("=== controller log ===")
        click.echo(result["controller_log"])
    if result.get("task_log"):
        click.echo("=== task log ===")
        click.echo(result["task_log"])


@jobs.command("cancel")
@click.argument("job_ids", nargs=-1, type=int)
@click.option("--all", "all_jobs", is_flag=True)
def jobs_cancel(job_ids, all_jobs):
    n = sdk.get(sdk.jobs_cancel(list(job_ids) or None, all_jobs))
    click.echo(f"Cancelled {n} managed job(s).")


@jobs.group("pool")
def jobs_pool():
    """Worker pools for managed jobs."""


@jobs.group("group")
def jobs_group():
    """Job groups: co-located concurrent tasks (one shared cluster)."""


@jobs_group.command("launch")
@click.argument("entrypoint")
@click.option("--name", "-n", required=True)
def jobs_group_launch_cmd(entrypoint, name):
    """Launch a job group from a YAML with a `tasks:` list."""
    import yaml as _yaml
    with open(entrypoint) as f:
        cfg = _yaml.safe_load(f) or {}
    tasks = cfg.get("tasks") or []
    _print_result(sdk.get(sdk.jobs_group_launch(name, tasks)))


@jobs_group.command("status")
@click.argument("name")
def jobs_group_status_cmd(name):
    _print_result(sdk.get(sdk.jobs_group_status(name)))


@jobs_group.command("down")
@click.argument("name")
def jobs_group_down_cmd(name):
    _print_result(sdk.get(sdk.jobs_group_down(name)))


@jobs_pool.command("apply")
@click.argument("entrypoint")
@click.option("--pool", "-p", "name", required=True)
@click.option("--workers", type=int, default=2)
@click.option("--min-workers", type=int, default=None,
              help="autoscaling floor (queue-length autoscaler)")
@click.option("--max-workers", type=int, default=None,
              help="autoscaling ceiling")
def jobs_pool_apply(entrypoint, name, workers, min_workers, max_workers):
    task = _load_task(entrypoint, (), {})
    _print_result(sdk.get(sdk.jobs_pool_apply(
        name, task, workers, min_workers=min_workers,
        max_workers=max_workers)))


@jobs_pool.command("status")
@click.argument("name", required=False)
def jobs_pool_status(name):
    _print_result(sdk.get(sdk.jobs_pool_status(name)))


@jobs_pool.command("down")
@click.argument("name")
def jobs_pool_down(name):
    n = sdk.get(sdk.jobs_pool_down(name))
    click.echo(f"Tore down {n} pool worker(s).")


# ---- serve ----------------------------------------------------------------
@cli.group()
def serve():
    """Sky Serve: replicated serving with autoscaling."""


@serve.command("up")
@click.argument("entrypoint")
@click.option("--service-name", "-n", required=True)
@click.option("--env", multiple=True)
def serve_up(entrypoint, service_name, env):
    task = _load_task(entrypoint, env, {})
    _print_result(sdk.stream_and_get(sdk.serve_up(task, service_name)))


@serve.command("update")
@click.argument("entrypoint")
@click.option("--service-name", "-n", required=True)
def serve_update(entrypoint, service_name):
    task = _load_task(entrypoint, (), {})
    _print_result(sdk.get(sdk.serve_update(task, service_name)))


@serve.command("down")
@click.argument("service_name", required=False)
@click.option("--all", "-a", "all_", is_flag=True,
              help="tear down every service")
@click.option("--yes", "-y", is_flag=True)
def serve_down(service_name, all_, yes):
    if all_:
        names = [sv["name"] for sv in sdk.get(sdk.serve_status(None))]
        if not names:
            click.echo("No services.")
            return
        if not yes and not click.confirm(
                f"Tear down {', '.join(names)}?"):
            return
    elif service_name:
        names = [service_name]
    else:
        raise click.UsageError("pass a service name or --all")
    for n in names:
        sdk.get(sdk.serve_down(n))
        click.echo(f"Service {n} torn down.")


@serve.command("logs")
@click.argument("service_name")
@click.option("--replica", type=int, default=None)
def serve_logs(service_name, replica):
    result = sdk.get(sdk.serve_logs(service_name, replica))
    if result.get("controller_log"):
        click.echo("=== controller ===")
        click.echo(result["controller_log"])
    for rid, log in (result.get("replica_logs") or {}).items():
        click.echo(f"=== replica {rid} ===")
        click.echo(log)


@serve.command("status")
@click.argument("service_name", required=False)
def serve_status(service_name):
    _print_result(sdk.get(sdk.serve_status(service_name)))


# ---- storage --------------------------------------------------------------
@cli.group()
def storage():
    """Storage management."""


@storage.command("ls")
def storage_ls():
    _print_result(sdk.get(sdk.storage_list()))


@storage.command("sync")
@click.argument("name")
def storage_sync_cmd(name):
    """Push a store to its S3-compatible remote (rclone)."""
    _print_result(sdk.get(sdk.storage_sync(name)))


@storage.command("delete")
@click.argument("name")
def storage_delete(name):
    sdk.get(sdk.storage_delete(name))
    click.echo(f"Storage {name} deleted.")


# ---- volumes --------------------------------------------------------------
@cli.group()
def volumes():
    """Persistent volumes."""


@volumes.command("ls")
def volumes_ls():
    _print_result(sdk.get(sdk.volumes_list()))


@volumes.command("create")
@click.argument("name")
@click.option("--size-gb", type=int, default=None)
def volumes_create(name, size_gb):
    _print_result(sdk.get(sdk.volumes_create(name, size_gb)))


@volumes.command("delete")
@click.argument("name")
def volumes_delete(name):
    sdk.get(sdk.volumes_delete(name))
    click.echo(f"Volume {name} deleted.")


@cli.command("workspace")
def workspace_cmd():
    """Show the active workspace (SKY_AMD_WORKSPACE env or config
    `workspace:`)."""
    from skypilot_amd import global_state
    click.echo(global_state.current_workspace())


@cli.group()
def users():
    """User, role and service-account management (admin)."""


@users.command("list")
def users_list():
    with sdk._client() as c:
        _print_result(c.get("/api/users").json())


@users.command("set-role")
@click.argument("name")
@click.argument("role", type=click.Choice(["admin", "user", "viewer"]))
def users_set_role(name, role):
    with sdk._client() as c:
        r = c.post("/api/users/role", json={"name": name, "role": role})
        if r.status_code != 200:
            raise click.ClickException(r.text)
    click.echo(f"{name} -> {role}")


@users.command("token")
@click.argument("name")
@click.option("--role", default="user",
              type=click.Choice(["admin", "user", "viewer"]))
def users_token(name, role):
    """Mint a service-account token (shown once)."""
    with sdk._client() as c:
        r = c.post("/api/users/token", json={"name": name, "role": role})
        if r.status_code != 200:
            raise click.ClickException(r.text)
        click.echo(r.json()["token"])


@users.command("tokens")
def users_tokens():
    with sdk._client() as c:
        _print_result(c.get("/api/users/tokens").json())


@users.command("revoke")
@click.argument("name")
def users_revoke(name):
    with sdk._client() as c:
        r = c.post("/api/users/token/revoke", json={"name": name})
        _print_result(r.json())


@cli.command("cost-report")
def cost_report_cmd():
    """GPU-hour usage per cluster (live + torn down)."""
    rows = sdk.get(sdk.cost_report())
    fmt = "{:<20} {:<10} {:<11} {:>5} {:>10} {:>10}"
    click.echo(fmt.format("NAME", "USER", "STATUS", "GPUS", "HOURS",
                          "GPU-HOURS"))
    for r in rows:
        click.echo(fmt.format(r["name"], str(r.get("user") or "-"),
                              r["status"], r["gpus"],
                              r["duration_hours"], r["gpu_hours"]))


@cli.command("recipes")
def recipes_cmd():
    """List bundled task recipes (examples/*.yaml)."""
    rows = sdk.get(sdk.recipes_list())
    fmt = "{:<28} {:<22} {:<12}"
    click.echo(fmt.format("NAME", "TASK", "GPUS"))
    for r in rows:
        click.echo(fmt.format(r["name"], r["task_name"], r["accelerators"]))


def main():
    cli()


if __name__ == "__main__":
    main()
