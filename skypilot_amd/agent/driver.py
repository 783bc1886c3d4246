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
    gpus_per_node = int(spec.get("gpus_per_node", 0))
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
    node_ips = spec.get("node_ips") or ["127.0.0.1"] * num_nodes
    master_port = _free_port()
    procs = []
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
            "SKYPILOT_TASK_ID": task_id,
            "SKYPILOT_INTERNAL_JOB_ID": str(job_id),
            # Convenience for torchrun on the one-box pool: a unique
            # rendezvous port per job (reference leaves this to the user).
            "SKYPILOT_MASTER_PORT": str(master_port),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(master_port),
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        })
        if spec.get("managed_job_id"):
            env["SKYPILOT_MANAGED_JOB_ID"] = str(spec["managed_job_id"])
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

    signal.signal(signal.SIGTERM, forward_term)

    rcs = []
    try:
        for p in procs:
            rcs.append(p.wait())
    finally:
        for p in procs:
            if p.poll() is None:
                try:
                    os.killpg(p.pid, signal.SIGTERM)
                except ProcessLookupError:
                    pass
            p._logf.close()

    worst = max((abs(r) for r in rcs), default=0)
    final = job_lib.SUCCEEDED if all(r == 0 for r in rcs) else job_lib.FAILED
    cur = table.get(job_id)
    if cur and cur["status"] == job_lib.CANCELLED:
        final = job_lib.CANCELLED
    else:
        table.set_status(job_id, final, exit_code=worst)
    _run_event_callback(spec, task_id, job_id, final, workdir, log_dir)
    return 0 if final == job_lib.CANCELLED else worst


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
