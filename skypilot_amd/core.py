"""Server-side implementations of status/start/stop/down/queue/cancel/...

Reference: sky/core.py (2,022 LoC of the same operations).
"""
from __future__ import annotations

import time
from pathlib import Path
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state, users, provision
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.exceptions import ClusterDoesNotExist, ClusterNotUpError
from skypilot_amd.utils.gpu_topology import detect_gpus, xgmi_topology


def _get_record(cluster_name: str) -> Dict[str, Any]:
    record = global_state.get_cluster(cluster_name)
    if record is None:
        raise ClusterDoesNotExist(f"cluster {cluster_name!r} not found")
    return record


def status(cluster_names: Optional[List[str]] = None,
           refresh: bool = False,
           all_workspaces: bool = False) -> List[Dict[str, Any]]:
    records = global_state.list_clusters(all_workspaces=all_workspaces)
    if cluster_names:
        records = [r for r in records if r["name"] in cluster_names]
    if refresh:
        for r in records:
            live = provision.query_instances(
                r["handle"].get("cloud"), r["name"], r["handle"])
            if live != r["status"]:
                global_state.set_cluster_status(r["name"], live)
                r["status"] = live
            # Autostop flag file written by the agent (skylet StopEvent).
            cdir = r["handle"].get("cluster_dir")
            if cdir and (Path(cdir) / "autostop_triggered").exists():
                import json as _json
                info = _json.loads(
                    (Path(cdir) / "autostop_triggered").read_text())
                if info.get("down"):
                    down(r["name"])
                    r["status"] = "TERMINATED"
                else:
                    stop(r["name"])
                    r["status"] = global_state.STOPPED
    return records


def start(cluster_name: str) -> Dict[str, Any]:
    users.check_cluster_owner(cluster_name)
    record = _get_record(cluster_name)
    handle = record["handle"]
    flag = Path(handle["cluster_dir"]) / "autostop_triggered"
    flag.unlink(missing_ok=True)
    handle = provision.run_instances(
        handle.get("cloud"), cluster_name, handle.get("num_nodes", 1),
        "MI355X" if handle.get("gpus_per_node") else None,
        handle.get("gpus_per_node", 0), handle)
    global_state.add_or_update_cluster(cluster_name, global_state.UP,
                                       handle, record["resources"])
    global_state.add_cluster_event(cluster_name, "START")
    return handle


def stop(cluster_name: str) -> None:
    users.check_cluster_owner(cluster_name)
    record = _get_record(cluster_name)
    PoolBackend().teardown(record["handle"], terminate=False)


def down(cluster_name: str) -> None:
    users.check_cluster_owner(cluster_name)
    record = _get_record(cluster_name)
    PoolBackend().teardown(record["handle"], terminate=True)


def autostop(cluster_name: str, idle_minutes: int, down_: bool = False
             ) -> None:
    users.check_cluster_owner(cluster_name)
    record = _get_record(cluster_name)
    if record["status"] != global_state.UP:
        raise ClusterNotUpError(f"cluster {cluster_name!r} is not UP")
    PoolBackend().set_autostop(record["handle"], idle_minutes, down_)


def queue(cluster_name: str) -> List[Dict[str, Any]]:
    record = _get_record(cluster_name)
    if record["status"] != global_state.UP:
        raise ClusterNotUpError(f"cluster {cluster_name!r} is not UP")
    return PoolBackend().job_queue(record["handle"])


def cancel(cluster_name: str, job_ids: Optional[List[int]] = None,
           all_jobs: bool = False) -> int:
    users.check_cluster_owner(cluster_name)
    record = _get_record(cluster_name)
    return PoolBackend().cancel_jobs(record["handle"],
                                     None if all_jobs else job_ids)


def tail_logs(cluster_name: str, job_id: Optional[int] = None,
              follow: bool = True):
    record = _get_record(cluster_name)
    return PoolBackend().tail_logs(record["handle"], job_id, follow)


def job_status(cluster_name: str, job_id: int) -> Optional[Dict[str, Any]]:
    record = _get_record(cluster_name)
    return PoolBackend()._agent(record["handle"]).get_job(job_id)


def check() -> Dict[str, Any]:
    """reference: sky/check.py — per-pool capability check."""
    gpus = detect_gpus()
    pools: Dict[str, Any] = {
        "local": {
            "enabled": True,
            "gpus": [{"index": g.index, "name": g.name,
                      "memory_gb": g.memory_gb,
                      "numa_node": g.numa_node} for g in gpus],
        },
    }
    try:
        from skypilot_amd.provision import ssh_pool
        hosts = ssh_pool.parse_hosts()
        pools["ssh"] = {"enabled": bool(hosts),
                        "hosts": [h["ip"] for h in hosts]}
    except Exception as e:  # noqa: BLE001
        pools["ssh"] = {"enabled": False, "error": str(e)}
    try:
        import shutil as _sh
        from skypilot_amd.provision import k8s
        st = k8s.k8s_settings()
        pools["kubernetes"] = {
            "enabled": bool(_sh.which("kubectl") and st.get("image")),
            "namespace": st.get("namespace"),
            "image": st.get("image"),
        }
    except Exception as e:  # noqa: BLE001
        pools["kubernetes"] = {"enabled": False, "error": str(e)}
    return {"pools": pools, "timestamp": time.time()}


def show_gpus() -> List[Dict[str, Any]]:
    taken: Dict[int, str] = {}
    for c in global_state.list_clusters():
        if c["status"] == global_state.UP:
            for g in c["handle"].get("gpu_ids", []):
                taken[g] = c["name"]
    return [{
        "index": g.index, "name": g.name, "memory_gb": g.memory_gb,
        "numa_node": g.numa_node,
        "used_by": taken.get(g.index),
    } for g in detect_gpus()]


def cluster_events(cluster_name: str) -> List[Dict[str, Any]]:
    return global_state.get_cluster_events(cluster_name)


def topology() -> Dict[str, Any]:
    return {"xgmi": xgmi_topology(), "gpus": [g.__dict__ for g in
                                              detect_gpus()]}


def cost_report() -> List[Dict[str, Any]]:
    """GPU-hour accounting per cluster, live and historical
    (reference: sky/client/cli cost-report backed by cluster_history;
    a local pool has no $ prices, so the unit is GPU-hours)."""
    now = time.time()
    rows: List[Dict[str, Any]] = []
    for r in global_state.list_clusters():
        h = r["handle"]
        ngpu = h.get("gpus_per_node", 0) * h.get("num_nodes", 1)
        dur_h = (now - (r.get("launched_at") or now)) / 3600
        rows.append({
            "name": r["name"], "user": r.get("user"), "status":
            r["status"], "gpus": ngpu,
            "duration_hours": round(dur_h, 3),
            "gpu_hours": round(ngpu * dur_h, 3), "live": True,
        })
    for hrec in global_state.list_cluster_history(limit=200):
        res = hrec.get("resources") or {}
        acc = str(res.get("accelerators") or "")
        ngpu = 0
        if ":" in acc:
            ngpu = int(acc.split(":")[1])
        elif acc:
            ngpu = 1
        dur_h = max(0.0, ((hrec.get("torn_down_at") or 0) -
                          (hrec.get("launched_at") or 0)) / 3600)
        rows.append({
            "name": hrec["name"], "user": None, "status": "TERMINATED",
            "gpus": ngpu, "duration_hours": round(dur_h, 3),
            "gpu_hours": round(ngpu * dur_h, 3), "live": False,
        })
    return rows
