"""Local MI355X pool provisioner.

Reference shape: sky/provision/__init__.py per-cloud module API
(run_instances / terminate_instances / stop_instances / wait_instances /
get_cluster_info).  On the local pool, "provisioning" allocates GPUs
from the node inventory, creates the cluster runtime dir, and starts the
node-agent daemon — no cloud API round-trips, which is what makes
job-start latency beat the reference's EXEC path.
"""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import sys
from pathlib import Path
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state
from skypilot_amd.agent.client import AgentClient
from skypilot_amd.exceptions import ResourcesUnavailableError
from skypilot_amd.utils.gpu_topology import detect_gpus

CLOUD_NAME = "local"


def cluster_dir(cluster_name: str) -> Path:
    d = global_state.root_dir() / "clusters" / cluster_name
    d.mkdir(parents=True, exist_ok=True)
    return d


def _allocated_gpus_elsewhere(except_cluster: str) -> List[int]:
    out: List[int] = []
    for c in global_state.list_clusters():
        if c["name"] == except_cluster or c["status"] != global_state.UP:
            continue
        if c["handle"].get("cloud") == CLOUD_NAME:
            out.extend(c["handle"].get("gpu_ids", []))
    return out


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_instances(cluster_name: str, num_nodes: int, accelerator: str | None,
                  acc_count: int, existing_handle: Optional[Dict] = None,
                  use_spot: bool = False) -> Dict[str, Any]:
    """Allocate GPUs + start the agent. Returns the cluster handle.
    Guarded by a per-cluster file lock (reference: utils/locks.py used at
    cloud_vm_ray_backend.py:3433) so concurrent launches can't
    double-allocate GPUs or double-start the agent."""
    import fcntl
    lock_dir = global_state.root_dir() / "locks"
    lock_dir.mkdir(parents=True, exist_ok=True)
    lock_f = open(lock_dir / f"provision-{cluster_name}.lock", "w")
    fcntl.flock(lock_f, fcntl.LOCK_EX)
    try:
        return _run_instances_locked(cluster_name, num_nodes, accelerator,
                                     acc_count, existing_handle, use_spot)
    finally:
        fcntl.flock(lock_f, fcntl.LOCK_UN)
        lock_f.close()


def _spot_victims(requester: str) -> list:
    """Live spot clusters, newest first (cheapest to preempt by
    seniority: reference spot semantics kill the market's choice; on a
    pool we take the most recently launched)."""
    out = []
    for c in global_state.list_clusters():
        if c["name"] == requester or c["status"] != global_state.UP:
            continue
        h = c["handle"]
        if h.get("cloud") == CLOUD_NAME and h.get("use_spot")                 and h.get("gpu_ids"):
            out.append(c)
    out.sort(key=lambda c: c.get("launched_at") or 0, reverse=True)
    return out


def _preempt(victim: Dict) -> None:
    """Kill a spot cluster like a revoked spot instance: agent + record
    gone (the managed-jobs monitor then sees the cluster missing and
    enters RECOVERING; reference: jobs/controller.py preemption path)."""
    name = victim["name"]
    global_state.add_cluster_event(name, "PREEMPTED",
                                   "spot capacity reclaimed")
    try:
        terminate_instances(name, victim["handle"])
    except Exception:  # noqa: BLE001
        pass
    global_state.remove_cluster(name)


def _fractional_load(except_cluster: str):
    """Per-GPU fractional usage + the set of fully-leased GPUs."""
    frac = {}
    full = set()
    for c in global_state.list_clusters(all_workspaces=True):
        if c["name"] == except_cluster or c["status"] not in (
                global_state.UP, global_state.INIT):
            continue
        h = c["handle"]
        if h.get("cloud") != CLOUD_NAME:
            continue
        f = h.get("gpu_fraction")
        for g in h.get("gpu_ids", []):
            if f:
                frac[g] = frac.get(g, 0.0) + f
            else:
                full.add(g)
    return frac, full


def _run_instances_locked(cluster_name, num_nodes, accelerator, acc_count,
                          existing_handle=None, use_spot=False
                          ) -> Dict[str, Any]:
    gpus = detect_gpus()
    need = num_nodes * acc_count
    gpu_fraction = None
    if existing_handle and existing_handle.get("gpu_ids") is not None:
        gpu_ids = existing_handle["gpu_ids"]
        gpu_fraction = existing_handle.get("gpu_fraction")
    elif 0 < acc_count < 1:
        # Fractional share of one GPU (reference: fractional
        # accelerators): bin-packing — pick the MOST-loaded GPU that
        # still fits, so whole GPUs stay free for whole-GPU leases.
        frac, full = _fractional_load(cluster_name)
        cand = [(frac.get(g.index, 0.0), g.index) for g in gpus
                if g.index not in full
                and frac.get(g.index, 0.0) + acc_count <= 1.0 + 1e-6]
        if not cand:
            raise ResourcesUnavailableError(
                f"no GPU has {acc_count:g} capacity free "
                f"(fractional shares)")
        cand.sort(reverse=True)
        gpu_ids = [cand[0][1]]
        gpu_fraction = float(acc_count)
    elif need > 0:
        frac_used, full_taken = _fractional_load(cluster_name)
        taken = set(_allocated_gpus_elsewhere(cluster_name))
        taken |= set(frac_used)  # fractionally-shared GPUs are not whole
        free = [g.index for g in gpus if g.index not in taken]
        if len(free) < need and not use_spot:
            # On-demand requests reclaim spot capacity (reference:
            # spot preemption; spot requests never preempt anyone).
            for victim in _spot_victims(cluster_name):
                _preempt(victim)
                taken = set(_allocated_gpus_elsewhere(cluster_name))
                free = [g.index for g in gpus if g.index not in taken]
                if len(free) >= need:
                    break
        if len(free) < need:
            raise ResourcesUnavailableError(
                f"need {need}x{accelerator or 'GPU'}, pool has "
                f"{len(free)} free of {len(gpus)}")
        gpu_ids = free[:need]
    else:
        gpu_ids = []

    cdir = cluster_dir(cluster_name)
    (cdir / "workdir").mkdir(exist_ok=True)
    handle = {
        "cloud": CLOUD_NAME,
        "cluster_dir": str(cdir),
        "gpu_ids": gpu_ids,
        "num_nodes": num_nodes,
        "gpus_per_node": acc_count,
        "gpu_fraction": gpu_fraction,
        "use_spot": bool(use_spot),
        "head_ip": "127.0.0.1",
        "node_ips": ["127.0.0.1"] * num_nodes,
    }
    handle["agent_port"] = _ensure_agent(cdir, gpu_ids,
                                         existing_handle or {})
    handle["agent_token"] = _agent_token(cdir)
    return handle


def _agent_meta(cdir: Path) -> Optional[Dict]:
    meta = cdir / "agent.json"
    if not meta.exists():
        return None
    try:
        return json.loads(meta.read_text())
    except Exception:  # noqa: BLE001
        return None


def _agent_pid_alive(cdir: Path) -> Optional[Dict]:
    """Process-level liveness only (the pid exists)."""
    info = _agent_meta(cdir)
    if info is None:
        return None
    try:
        pid = info["pid"]
        os.kill(pid, 0)
        # a SIGKILLed-but-unreaped agent is a zombie, not a live agent
        with open(f"/proc/{pid}/stat") as f:
            if f.read().rsplit(")", 1)[1].split()[0] == "Z":
                return None
        return info
    except (ProcessLookupError, PermissionError, KeyError, OSError,
            IndexError):
        return None


def _agent_alive(cdir: Path) -> Optional[int]:
    """Full liveness: pid exists AND the HTTP endpoint answers."""
    info = _agent_pid_alive(cdir)
    if info is None:
        return None
    try:
        if AgentClient(info["port"], timeout=5.0).healthy():
            return info["port"]
    except Exception:  # noqa: BLE001
        pass
    return None


def _agent_token(cdir: Path) -> str:
    """Per-cluster shared secret for the agent HTTP surface."""
    tf = cdir / "agent_token"
    if tf.exists():
        return tf.read_text().strip()
    import secrets
    tok = secrets.token_hex(16)
    tf.touch(mode=0o600)
    tf.write_text(tok)
    return tok


def _ensure_agent(cdir: Path, gpu_ids: List[int],
                  existing_handle: Dict) -> int:
    tok = _agent_token(cdir)
    port = _agent_alive(cdir)
    if port is not None:
        return port
    port = _free_port()
    log = open(cdir / "agent.log", "ab")
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.agent.daemon",
         "--cluster-dir", str(cdir), "--port", str(port),
         "--gpu-ids", ",".join(str(g) for g in gpu_ids)],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env={**os.environ, "SKY_AMD_HOME": str(global_state.root_dir()),
             "SKY_AMD_AGENT_TOKEN": tok})
    log.close()
    # Record the pid IMMEDIATELY (the daemon re-writes the same file at
    # startup): if this provision request is cancelled before the agent
    # finishes booting, teardown/orphan-sweep must still find the pid —
    # the boot-window gap leaked agent processes in CI.
    (cdir / "agent.json").write_text(
        json.dumps({"port": port, "pid": proc.pid}))
    AgentClient(port, token=tok).wait_ready(timeout=30)
    return port


def _drain_jobs(handle: Dict[str, Any]) -> None:
    """Cancel every non-terminal job straight through the on-disk job
    table (guarded killpg of the drivers).  Teardown used to kill only
    the AGENT; job drivers run in their own sessions and survived a
    cancel->teardown sequence (leaked `sleep`-ing drivers in CI)."""
    cdir = handle.get("cluster_dir")
    if not cdir or not Path(cdir).exists():
        return
    try:
        from skypilot_amd.agent import job_lib
        t = job_lib.JobTable(cdir)
        for j in t.list():
            if j["status"] in job_lib.NONTERMINAL:
                t.cancel(j["job_id"])
    except Exception:  # noqa: BLE001 — teardown is best-effort
        pass


def stop_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    """Stop = kill the agent + all jobs; runtime dir and GPU lease kept."""
    _drain_jobs(handle)
    _kill_agent(handle)


def terminate_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    _drain_jobs(handle)
    _kill_agent(handle)
    import shutil
    cdir = handle.get("cluster_dir")
    if cdir and Path(cdir).exists():
        try:
            from skypilot_amd.data.storage import unmount_cluster_mounts
            unmount_cluster_mounts(cdir)
        except Exception:  # noqa: BLE001
            pass
        shutil.rmtree(cdir, ignore_errors=True)


def _kill_agent(handle: Dict[str, Any]) -> None:
    cdir = handle.get("cluster_dir")
    if not cdir:
        return
    port = handle.get("agent_port")
    if port:
        try:
            AgentClient(port,
                        token=handle.get("agent_token")).cancel_all()
        except Exception:  # noqa: BLE001
            pass
    meta = Path(cdir) / "agent.json"
    if meta.exists():
        try:
            pid = json.loads(meta.read_text()).get("pid")
            if pid:
                os.kill(pid, signal.SIGTERM)
        except (OSError, ValueError, ProcessLookupError):
            pass
        meta.unlink(missing_ok=True)


def query_instances(cluster_name: str, handle: Dict[str, Any]) -> str:
    """UP = agent process alive AND answering HTTP.  A live pid with a
    dead/hung HTTP endpoint (partition, wedged agent) reports INIT so
    `sky status -r` surfaces the degradation without hanging
    (reference: abnormal clusters map to INIT)."""
    cdir = handle.get("cluster_dir")
    if not cdir:
        return global_state.STOPPED
    info = _agent_pid_alive(Path(cdir))
    if info is None:
        return global_state.STOPPED
    try:
        if AgentClient(info["port"], timeout=2.0).healthy():
            return global_state.UP
    except Exception:  # noqa: BLE001
        pass
    return global_state.INIT


def get_cluster_info(handle: Dict[str, Any]) -> Dict[str, Any]:
    return {
        "head_ip": handle.get("head_ip", "127.0.0.1"),
        "node_ips": handle.get("node_ips", ["127.0.0.1"]),
        "gpu_ids": handle.get("gpu_ids", []),
    }
