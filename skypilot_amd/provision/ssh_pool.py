"""SSH node-pool provisioner — bare MI355X machines over SSH.

Reference: sky/provision/ssh/ + sky/ssh_node_pools/core.py (the
reference adopts bare machines into k3s; here the node agent IS the
runtime, so adoption = push the framework + start the agent + tunnel its
port — no Kubernetes needed for a flat pool).

Pool config: ~/.sky_amd/ssh_node_pools.yaml
    default:
      hosts:
        - ip: 10.0.0.5
          user: root
          identity_file: ~/.ssh/id_rsa
          gpus: 8
"""
from __future__ import annotations

import os
import shlex
import socket
import subprocess
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml

from skypilot_amd import global_state
from skypilot_amd.agent.client import AgentClient
from skypilot_amd.exceptions import ResourcesUnavailableError
from skypilot_amd.utils.command_runner import SSHCommandRunner

CLOUD_NAME = "ssh"
POOLS_PATH = "~/.sky_amd/ssh_node_pools.yaml"
REMOTE_ROOT = "~/.sky_amd_node"
AGENT_PORT = 46590  # fixed remote port (skylet's, reference constants:184)


def load_pools(path: Optional[str] = None) -> Dict[str, Any]:
    p = Path(os.path.expanduser(path or POOLS_PATH))
    if not p.exists():
        return {}
    with open(p) as f:
        return yaml.safe_load(f) or {}


def ensure_keypair() -> str:
    """Framework SSH keypair (reference: sky/authentication.py — an
    ed25519 pair generated once under the state dir; its pubkey is what
    gets installed on pool hosts).  Used for any host that does not set
    identity_file."""
    import subprocess
    d = global_state.root_dir() / "ssh"
    d.mkdir(parents=True, exist_ok=True)
    key = d / "sky-key"
    if not key.exists():
        subprocess.run(
            ["ssh-keygen", "-t", "ed25519", "-N", "", "-q", "-f",
             str(key), "-C", "skypilot-amd"],
            check=True, capture_output=True)
        key.chmod(0o600)
    return str(key)


def parse_hosts(pool: Optional[str] = None) -> List[Dict[str, Any]]:
    pools = load_pools()
    if not pools:
        return []
    name = pool or next(iter(pools))
    cfg = pools.get(name) or {}
    hosts = []
    for h in cfg.get("hosts", []):
        if isinstance(h, str):
            hosts.append({"ip": h, "user": "root", "gpus": 8,
                          "identity_file": ensure_keypair()})
        else:
            hosts.append({"ip": h["ip"], "user": h.get("user", "root"),
                          "identity_file": (h.get("identity_file")
                                            or ensure_keypair()),
                          "port": int(h.get("port", 22)),
                          "gpus": int(h.get("gpus", 8))})
    return hosts


def _runner(host: Dict[str, Any]) -> SSHCommandRunner:
    return SSHCommandRunner(host["ip"], user=host.get("user"),
                            port=host.get("port", 22),
                            identity_file=host.get("identity_file"))


def _hosts_in_use(except_cluster: str) -> List[str]:
    used = []
    for c in global_state.list_clusters():
        if c["name"] == except_cluster or c["status"] != global_state.UP:
            continue
        if c["handle"].get("cloud") == CLOUD_NAME:
            used.extend(c["handle"].get("hosts_ips", []))
    return used


def run_instances(cluster_name: str, num_nodes: int, accelerator,
                  acc_count: int, existing_handle: Optional[Dict] = None,
                  use_spot: bool = False) -> Dict[str, Any]:
    hosts = parse_hosts()
    if not hosts:
        raise ResourcesUnavailableError(
            f"no SSH hosts configured in {POOLS_PATH}")
    if existing_handle and existing_handle.get("hosts"):
        chosen = existing_handle["hosts"]
    else:
        used = set(_hosts_in_use(cluster_name))
        free = [h for h in hosts if h["ip"] not in used
                and h["gpus"] >= acc_count]
        if len(free) < num_nodes:
            raise ResourcesUnavailableError(
                f"need {num_nodes} SSH hosts with {acc_count}+ GPUs, "
                f"{len(free)} free")
        chosen = free[:num_nodes]

    head = chosen[0]
    runner = _runner(head)
    # Bootstrap: push the framework + start the agent on the head node.
    repo_root = Path(__file__).resolve().parents[2]
    runner.run(f"mkdir -p {REMOTE_ROOT}", check=True, timeout=60)
    runner.rsync(str(repo_root) + "/", f"{REMOTE_ROOT}/repo")
    gpu_ids = ",".join(str(i) for i in range(acc_count))
    # Per-cluster agent token (reference: skylet is only reachable via
    # the SSH tunnel; the token also protects against other local users
    # of the remote box).
    import secrets as _secrets
    token = _secrets.token_hex(16)
    runner.run(f"mkdir -p {REMOTE_ROOT}/cluster && umask 077 && "
               f"[ -s {REMOTE_ROOT}/cluster/agent_token ] || "
               f"echo {token} > {REMOTE_ROOT}/cluster/agent_token",
               check=True, timeout=60)
    token = runner.run(f"cat {REMOTE_ROOT}/cluster/agent_token",
                       check=True, timeout=60).stdout.strip()
    start_cmd = (
        f"cd {REMOTE_ROOT}/repo && "
        f"nohup python3 -m skypilot_amd.agent.daemon "
        f"--cluster-dir {REMOTE_ROOT}/cluster --port {AGENT_PORT} "
        f"--gpu-ids {shlex.quote(gpu_ids)} "
        f"> {REMOTE_ROOT}/agent.log 2>&1 & echo started")
    runner.run(f"pgrep -f 'skypilot_amd.agent.daemon.*{AGENT_PORT}' "
               f">/dev/null || ({start_cmd})", check=True, timeout=120)
    local_port = _tunnel(head, AGENT_PORT)
    AgentClient(local_port, token=token).wait_ready(timeout=60)

    cdir = global_state.root_dir() / "clusters" / cluster_name
    cdir.mkdir(parents=True, exist_ok=True)
    return {
        "cloud": CLOUD_NAME,
        "cluster_dir": str(cdir),
        "hosts": chosen,
        "hosts_ips": [h["ip"] for h in chosen],
        "gpu_ids": list(range(acc_count)),
        "num_nodes": num_nodes,
        "gpus_per_node": acc_count,
        "head_ip": head["ip"],
        "node_ips": [h["ip"] for h in chosen],
        "agent_port": local_port,
        "agent_token": token,
    }


def _tunnel(host: Dict[str, Any], remote_port: int) -> int:
    """SSH -L tunnel to the remote agent (reference: skylet gRPC channel
    through an SSH tunnel, cloud_vm_ray_backend.py:2414)."""
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        local_port = s.getsockname()[1]
    args = ["ssh", "-N", "-o", "StrictHostKeyChecking=no",
            "-o", "UserKnownHostsFile=/dev/null", "-o", "LogLevel=ERROR",
            "-L", f"{local_port}:127.0.0.1:{remote_port}",
            "-p", str(host.get("port", 22))]
    if host.get("identity_file"):
        args += ["-i", os.path.expanduser(host["identity_file"])]
    args.append(f"{host.get('user', 'root')}@{host['ip']}")
    subprocess.Popen(args, stdout=subprocess.DEVNULL,
                     stderr=subprocess.DEVNULL, start_new_session=True)
    time.sleep(0.5)
    return local_port


def _kill_remote_agents(handle: Dict[str, Any]) -> None:
    for h in handle.get("hosts", []):
        try:
            # Kill by the exact pid the agent recorded (never by pattern).
            _runner(h).run(
                f"pid=$(python3 -c \"import json;print(json.load(open("
                f"'{REMOTE_ROOT}/cluster/agent.json'))['pid'])\" "
                f"2>/dev/null); [ -n \"$pid\" ] && kill $pid || true",
                timeout=30)
        except Exception:  # noqa: BLE001
            pass


def stop_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    # Drain jobs through the live agent (guarded driver killpg), THEN
    # kill the agent: a stopped cluster with a live agent would flip
    # back to UP on the next status refresh (query_instances probes the
    # agent port).
    port = handle.get("agent_port")
    if port:
        try:
            AgentClient(port).cancel_all()
        except Exception:  # noqa: BLE001
            pass
    _kill_remote_agents(handle)


def terminate_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    stop_instances(cluster_name, handle)


def query_instances(cluster_name: str, handle: Dict[str, Any]) -> str:
    port = handle.get("agent_port")
    if port and AgentClient(port).healthy():
        return global_state.UP
    return global_state.STOPPED


def get_cluster_info(handle: Dict[str, Any]) -> Dict[str, Any]:
    return {"head_ip": handle.get("head_ip"),
            "node_ips": handle.get("node_ips", []),
            "gpu_ids": handle.get("gpu_ids", [])}
