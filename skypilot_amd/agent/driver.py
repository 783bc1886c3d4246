"""Job driver — the gang launcher that replaces the reference's Ray
placement-group codegen (reference: sky/backends/task_codegen.py:301
RayCodeGen, :595-685 env injection).

One driver process per job.  It starts `num_nodes` copies of the task's
run command, each with the SKYPILOT_* env contract (reference:
sky/skylet/constants.py:575-580) plus per-"node" GPU slices
(HIP_VISIBLE_DEVICES) and NUMA binding for the MI355X topology.  On the
single-node pool, "nodes" are process slices of the 8-GPU box; SSH pools
run one slice per remote machine.

Usage: python -m skypilot_amd.agent.driver <cluster_dir> <job_id>
"""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

from skypilot_amd.agent import job_lib
from skypilot_amd.utils.gpu_topology import plan_ranks


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_job(cluster_dir: str, job_id: int) -> int:
    table = job_lib.JobTable(cluster_dir)
    job = table.get(job_id)
    if job is None:
        print(f"job {job_id} not found", file=sys.stderr)
        return 1
    spec = job["spec"]
    num_nodes = int(spec.get("num_nodes", 1))
    gpn_raw = float(spec.get("gpus_per_node", 0) or 0)
    gpu_fraction = gpn_raw if 0 < gpn_raw < 1 else None
    # fractional share: the one shared device is fully visible; the
    # fraction travels in SKYPILOT_GPU_FRACTION (best-effort sharing,
    # reference: fractional k8s accelerators)
    gpus_per_node = 1 if gpu_fraction else int(gpn_raw)
    gpu_ids = spec.get("gpu_ids") or []
    workdir = spec.get("workdir") or str(Path(cluster_dir) / "workdir")
    Path(workdir).mkdir(parents=True, exist_ok=True)
    log_dir = Path(cluster_dir) / "logs" / str(job_id)
    log_dir.mkdir(parents=True, exist_ok=True)
    run_cmd = spec.get("run") or ""
    setup_cmd = spec.get("setup")
    envs = dict(spec.get("envs") or {})
    envs.update(spec.get("secrets") or {})
    task_id = spec.get("task_id") or f"sky-{int(time.time())}-{job_id}"

    table.set_driver(job_id, os.getpid(), gpu_ids, str(log_dir))
    cur = table.get(job_id)
    if cur and cur["status"] == job_lib.CANCELLED:
        return 0

    # -- setup phase (runs once per node; on the local pool: once) ----------
    if setup_cmd:
        table.set_status(job_id, job_lib.SETTING_UP)
        with open(log_dir / "setup.log", "ab") as f:
            rc = subprocess.run(["bash", "-c", setup_cmd], cwd=workdir,
                                env={**os.environ, **envs}, stdout=f,
                                stderr=subprocess.STDOUT).returncode
        if rc != 0:
            table.set_status(job_id, job_lib.FAILED_SETUP, exit_code=rc)
            return rc

    table.set_status(job_id, job_lib.RUNNING)
    if not run_cmd:
        table.set_status(job_id, job_lib.SUCCEEDED, exit_code=0)
        return 0

    # -- gang launch --------------------------------------------------------
    # Two dispatch modes per node rank:
    #   local slice  — a process on this box (single-node pool, SSH head)
    #   peer agent   — a leaf job POSTed to another node's agent over
    #                  HTTP (multi-pod k8s gangs: every pod runs an
    #                  agent; rank 0 is this pod, ranks 1.. dispatch to
    #                  peer_agents[i-1] = "ip:port").  The gang env
    #                  travels in the leaf spec's gang_env.
    node_ips = spec.get("node_ips") or ["127.0.0.1"] * num_nodes
    peer_agents = spec.get("peer_agents") or []
    master_addr = spec.get("master_addr") or "127.0.0.1"
    gang_env_override = spec.get("gang_env")  # set on leaf jobs
    master_port = int(spec.get("master_port") or _free_port())
    procs = []
    remote = []  # (AgentClient, remote_job_id, node_rank)
    for node_rank in range(num_nodes):
        node_gpus = gpu_ids[node_rank * gpus_per_node:
                            (node_rank + 1) * gpus_per_node]
        env = dict(os.environ)
        env.update(envs)
        env.update({
            "SKYPILOT_NODE_IPS": "\n".join(node_ips),
            "SKYPILOT_NODE_RANK": str(node_rank),
            "SKYPILOT_NUM_NODES": str(num_nodes),
            "SKYPILOT_NUM_GPUS_PER_NODE": str(gpus_per_node),
            **({"SKYPILOT_GPU_FRACTION": str(gpu_fraction)}
               if gpu_fraction else {}),
            "SKYPILOT_TASK_ID": task_id,
            "SKYPILOT_INTERNAL_JOB_ID": str(job_id),
            # Convenience for torchrun on the one-box pool: a unique
            # rendezvous port per job (reference leaves this to the user).
            "SKYPILOT_MASTER_PORT": str(master_port),
            "MASTER_ADDR": master_addr,
            "MASTER_PORT": str(master_port),
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        })
        if gang_env_override:
            env.update(gang_env_override)
        if spec.get("managed_job_id"):
            env["SKYPILOT_MANAGED_JOB_ID"] = str(spec["managed_job_id"])
        if node_rank > 0 and node_rank <= len(peer_agents):
            from skypilot_amd.agent.client import AgentClient
            host, _, port = peer_agents[node_rank - 1].rpartition(":")
            peer = AgentClient(int(port), host=host,
                               token=spec.get("agent_token"))
            leaf_env = {k: v for k, v in env.items()
                        if k.startswith(("SKYPILOT_", "MASTER_"))}
            leaf_env.update(envs)
            rjid = peer.queue_job({
                "run": run_cmd, "setup": spec.get("setup"),
                "envs": {}, "num_nodes": 1,
                "gpus_per_node": gpus_per_node,
                "gang_env": leaf_env, "task_id": task_id,
            }, name=f"{task_id}-rank{node_rank}")
            remote.append((peer, rjid, node_rank))
            continue
        if node_gpus:
            env["HIP_VISIBLE_DEVICES"] = ",".join(str(g) for g in node_gpus)
            env["CUDA_VISIBLE_DEVICES"] = env["HIP_VISIBLE_DEVICES"]
        plans = plan_ranks(node_gpus) if node_gpus else []
        prefix = []
        if plans and all(p.numa_node == plans[0].numa_node for p in plans):
            prefix = plans[0].numactl_prefix()
        log_file = log_dir / (f"run.log" if num_nodes == 1
                              else f"{node_rank}-node.log")
        f = open(log_file, "ab")
        p = subprocess.Popen(prefix + ["bash", "-c", run_cmd], cwd=workdir,
                             env=env, stdout=f, stderr=subprocess.STDOUT,
                             start_new_session=True)
        p._logf = f  # keep ref
        procs.append(p)

    # Record node-process pgids so the agent can reap orphans if this
    # driver dies (reference: skylet/subprocess_daemon.py).
    spec["node_pids"] = [p.pid for p in procs]
    with table._conn() as c:
        c.execute("UPDATE jobs SET spec=? WHERE job_id=?",
                  (json.dumps(spec), job_id))

    def forward_term(signum, frame):
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
        for peer, rjid, _ in remote:
            try:
                peer.cancel_job(rjid)
            except Exception:  # noqa: BLE001
                pass

    signal.signal(signal.SIGTERM, forward_term)

    def _pull_peer_log(peer, rjid, node_rank):
        try:
            with open(log_dir / f"{node_rank}-node.log", "ab") as lf:
                for chunk in peer.tail_logs(rjid, follow=False):
                    lf.write(chunk if isinstance(chunk, bytes)
                             else chunk.encode())
        except Exception:  # noqa: BLE001
            pass

    # Gang semantics (reference: a Ray placement-group task error fails
    # the gang): first non-zero rank kills local slices and cancels
    # peer leaf jobs instead of waiting out the survivors.
    rcs = []
    try:
        pending_p = list(procs)
        pending_r = list(remote)
        failing = False
        while pending_p or pending_r:
            progressed = False
            for p in list(pending_p):
                rc = p.poll()
                if rc is not None:
                    rcs.append(rc)
                    pending_p.remove(p)
                    progressed = True
                    failing = failing or rc != 0
            for item in list(pending_r):
                peer, rjid, node_rank = item
                j = peer.get_job(rjid)
                if j and j["status"] in job_lib.TERMINAL:
                    rc = (0 if j["status"] == job_lib.SUCCEEDED
                          else (j.get("exit_code") or 1))
                    rcs.append(rc)
                    pending_r.remove(item)
                    progressed = True
                    failing = failing or rc != 0
                    _pull_peer_log(peer, rjid, node_rank)
            if failing and (pending_p or pending_r):
                for p in pending_p:
                    try:
                        os.killpg(p.pid, signal.SIGTERM)
                    except ProcessLookupError:
                        pass
                for peer, rjid, _ in pending_r:
                    try:
                        peer.cancel_job(rjid)
                    except Exception:  # noqa: BLE001
                        pass
            if not progressed:
                time.sleep(0.5)
    finally:
        for p in procs:
            if p.poll() is None:
                try:
                    os.killpg(p.pid, signal.SIGTERM)
                except ProcessLookupError:
                    pass
            p._logf.close()
        for peer, rjid, _ in remote:
            try:
                j = peer.get_job(rjid)
                if j and j["status"] not in job_lib.TERMINAL:
                    peer.cancel_job(rjid)
            except Exception:  # noqa: BLE001
                pass

    worst = max((abs(r) for r in rcs), default=0)
    final = job_lib.SUCCEEDED if all(r == 0 for r in rcs) else job_lib.FAILED
    cur = table.get(job_id)
    if cur and cur["status"] == job_lib.CANCELLED:
        final = job_lib.CANCELLED
    else:
        table.set_status(job_id, final, exit_code=worst)
    _run_event_callback(spec, task_id, job_id, final, workdir, log_dir)
    _ship_logs(task_id, job_id, final, log_dir)
    return 0 if final == job_lib.CANCELLED else worst


def _ship_logs(task_id, job_id, status, log_dir):
    """External log shipping (reference: sky/logs/agent.py
    FluentbitAgent installed at provision time; here a plain HTTP sink
    configured as `logs: {endpoint: <url>}` in ~/.sky_amd/config.yaml).
    Ships each completed job's log files as JSON lines; failures never
    affect the job."""
    try:
        from skypilot_amd import config as sky_config
        endpoint = sky_config.get_nested(["logs", "endpoint"])
        if not endpoint:
            return
        import json as _json
        import urllib.request
        for lf in sorted(Path(log_dir).glob("*.log")):
            body = _json.dumps({
                "task_id": task_id, "job_id": job_id, "status": status,
                "file": lf.name,
                "content": lf.read_text(errors="replace")[-65536:],
            }).encode()
            req = urllib.request.Request(
                endpoint, data=body,
                headers={"Content-Type": "application/json"})
            urllib.request.urlopen(req, timeout=10).read()
    except Exception:  # noqa: BLE001 — best-effort shipping
        pass


def _run_event_callback(spec, task_id, job_id, status, workdir, log_dir):
    """reference: jobs/utils.py:1090 event callbacks with SKYPILOT_TASK_ID
    / JOB_STATUS env."""
    cb = spec.get("event_callback")
    if not cb:
        return
    env = dict(os.environ)
    env.update({
        "SKYPILOT_TASK_ID": task_id,
        "SKYPILOT_INTERNAL_JOB_ID": str(job_id),
        "JOB_STATUS": status,
    })
    try:
        with open(Path(log_dir) / "event_callback.log", "ab") as f:
            subprocess.run(["bash", "-c", cb], cwd=workdir, env=env,
                           stdout=f, stderr=subprocess.STDOUT, timeout=120)
    except (OSError, subprocess.TimeoutExpired):
        pass


def main():
    cluster_dir, job_id = sys.argv[1], int(sys.argv[2])
    # The agent starts us with start_new_session=True, so we are already
    # our own process group leader — cancel kills the whole tree by pgid.
    sys.exit(run_job(cluster_dir, job_id))


if __name__ == "__main__":
    main()
