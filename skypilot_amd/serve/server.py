"""Serve server entrypoints (reference: sky/serve/server/impl.py up:140,
down, status)."""
from __future__ import annotations

import os
import socket
import subprocess
import sys
import time
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state
from skypilot_amd.exceptions import ServeError
from skypilot_amd.serve import serve_state as st
from skypilot_amd.serve.service_spec import ServiceSpec


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def up(task: Dict[str, Any], service_name: str) -> Dict[str, Any]:
    if st.get_service(service_name) is not None:
        existing = st.get_service(service_name)
        if existing["status"] not in (st.SHUTDOWN, st.FAILED):
            raise ServeError(f"service {service_name!r} already exists")
        st.remove_service(service_name)
    if not task.get("service"):
        raise ServeError("task YAML needs a `service:` section")
    spec = ServiceSpec.from_config(task["service"])  # validates
    lb_port = _free_port()
    st.add_service(service_name, task, task["service"], lb_port)
    pkg_root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    env = dict(os.environ)
    env["PYTHONPATH"] = pkg_root + (
        ":" + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
    log = open(global_state.root_dir() / f"serve-{service_name}.log", "ab")
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.serve.controller",
         service_name],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env=env)
    log.close()
    st.update_service(service_name, controller_pid=proc.pid)
    return {"service_name": service_name,
            "endpoint": f"http://127.0.0.1:{lb_port}",
            "lb_port": lb_port, "controller_pid": proc.pid,
            "min_replicas": spec.policy.min_replicas}


def update(task: Dict[str, Any], service_name: str) -> Dict[str, Any]:
    svc = st.get_service(service_name)
    if svc is None:
        raise ServeError(f"service {service_name!r} not found")
    if not task.get("service"):
        raise ServeError("task YAML needs a `service:` section")
    ServiceSpec.from_config(task["service"])  # validate
    version = st.bump_version(service_name, task, task["service"])
    return {"service_name": service_name, "version": version}


def down(service_name: str) -> None:
    svc = st.get_service(service_name)
    if svc is None:
        raise ServeError(f"service {service_name!r} not found")
    st.update_service(service_name, status=st.SHUTTING_DOWN)
    # The controller notices, drains, tears replicas down and exits.
    pid = svc.get("controller_pid")
    deadline = time.time() + 60
    while pid and time.time() < deadline:
        try:
            os.kill(pid, 0)
            time.sleep(0.5)
        except ProcessLookupError:
            break
        except PermissionError:
            break
    st.remove_service(service_name)
    # Belt-and-braces: if the controller exited before finishing its
    # teardown (or was killed), reap any replica clusters it left —
    # otherwise `sky status` shows orphaned sky-serve-* clusters.
    from skypilot_amd import global_state
    from skypilot_amd.backends.pool_backend import PoolBackend
    prefix = f"sky-serve-{service_name}-"
    for c in global_state.list_clusters(all_workspaces=True):
        if c["name"].startswith(prefix):
            try:
                PoolBackend().teardown(c["handle"], terminate=True)
            except Exception:  # noqa: BLE001
                global_state.remove_cluster(c["name"])


def logs(service_name: str, replica_id: int = None,
         tail_lines: int = 200):
    svc = st.get_service(service_name)
    if svc is None:
        raise ServeError(f"service {service_name!r} not found")
    out = {"service": service_name}
    log_path = global_state.root_dir() / f"serve-{service_name}.log"
    if log_path.exists():
        out["controller_log"] = "\n".join(
            log_path.read_text(errors="replace").splitlines()[-tail_lines:])
    reps = st.list_replicas(service_name)
    if replica_id is not None:
        reps = [r for r in reps if r["replica_id"] == replica_id]
    rep_logs = {}
    for r in reps:
        rec = global_state.get_cluster(r["cluster_name"]) if             r.get("cluster_name") else None
        if not rec:
            continue
        try:
            from skypilot_amd.backends.pool_backend import PoolBackend
            agent = PoolBackend()._agent(rec["handle"])
            jobs = agent.get_job_queue()
            if jobs:
                chunks = []
                for c in agent.tail_logs(jobs[0]["job_id"], follow=False):
                    chunks.append(c)
                    if sum(len(x) for x in chunks) > 1 << 20:
                        break
                rep_logs[r["replica_id"]] = b"".join(chunks).decode(
                    errors="replace")[-32768:]
        except Exception as e:  # noqa: BLE001
            rep_logs[r["replica_id"]] = f"(error: {e})"
    out["replica_logs"] = rep_logs
    return out


def status(service_name: Optional[str] = None) -> List[Dict[str, Any]]:
    st.reconcile()
    services = ([st.get_service(service_name)] if service_name
                else st.list_services())
    out = []
    for s in services:
        if s is None:
            continue
        out.append({
            "name": s["name"],
            "status": s["status"],
            "endpoint": f"http://127.0.0.1:{s['lb_port']}",
            "replicas": st.list_replicas(s["name"]),
        })
    return out
