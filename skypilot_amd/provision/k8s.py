"""Kubernetes pool provisioner — MI355X pods via kubectl.

Reference: sky/provision/kubernetes/instance.py (pod lifecycle) — the
largest reference provisioner.  Here the node agent replaces both Ray
and skylet, so a "cluster" is one pod running the agent with
`amd.com/gpu` resources; the control plane reaches it through a
`kubectl port-forward` tunnel (same shape as the SSH pool's tunnel).

Config (~/.sky_amd/config.yaml):
    kubernetes:
      context: my-ctx          # optional
      namespace: default
      image: <image with this framework + ROCm>
      gpu_resource: amd.com/gpu
"""
from __future__ import annotations

import json
import socket
import subprocess
import time
from typing import Any, Dict, List, Optional

import yaml

from skypilot_amd import config as sky_config
from skypilot_amd import global_state
from skypilot_amd.agent.client import AgentClient
from skypilot_amd.exceptions import ResourcesUnavailableError

CLOUD_NAME = "kubernetes"
AGENT_PORT = 46590
REMOTE_REPO = "/sky_amd_repo"


def k8s_settings() -> Dict[str, Any]:
    return {
        "context": sky_config.get_nested(["kubernetes", "context"]),
        "namespace": sky_config.get_nested(["kubernetes", "namespace"],
                                           "default"),
        "image": sky_config.get_nested(["kubernetes", "image"]),
        "gpu_resource": sky_config.get_nested(
            ["kubernetes", "gpu_resource"], "amd.com/gpu"),
    }


def _kubectl_base(settings: Dict[str, Any]) -> List[str]:
    cmd = ["kubectl"]
    if settings.get("context"):
        cmd += ["--context", settings["context"]]
    cmd += ["-n", settings.get("namespace", "default")]
    return cmd


def render_pod_manifest(cluster_name: str, token: str, acc_count: int,
                        settings: Dict[str, Any]) -> Dict[str, Any]:
    """Pod spec: one agent container with N AMD GPUs (reference pattern:
    provision/kubernetes/instance.py pod template)."""
    resources: Dict[str, Any] = {}
    if acc_count > 0:
        resources = {"limits": {settings["gpu_resource"]: acc_count}}
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": f"sky-amd-{cluster_name}",
            "labels": {"app": "sky-amd", "sky-amd-cluster": cluster_name},
        },
        "spec": {
            "restartPolicy": "Never",
            "containers": [{
                "name": "agent",
                "image": settings.get("image") or "sky-amd:latest",
                "command": ["python3", "-m", "skypilot_amd.agent.daemon",
                            "--cluster-dir", "/tmp/sky_amd_cluster",
                            "--port", str(AGENT_PORT),
                            "--host", "0.0.0.0",  # peers dial podIP
                            "--gpu-ids",
                            ",".join(str(i) for i in range(acc_count))],
                "workingDir": REMOTE_REPO,
                "resources": resources,
                "env": [{"name": "HSA_ENABLE_IPC_MODE_LEGACY",
                         "value": "0"},
                        {"name": "SKY_AMD_AGENT_TOKEN",
                         "value": token}],
            }],
        },
    }


def _run_kubectl(args: List[str], settings, input_text: str = None,
                 timeout: float = 60) -> subprocess.CompletedProcess:
    return subprocess.run(_kubectl_base(settings) + args,
                          input=input_text, capture_output=True, text=True,
                          timeout=timeout)


def _pod_name(cluster_name: str, rank: int) -> str:
    return (f"sky-amd-{cluster_name}" if rank == 0
            else f"sky-amd-{cluster_name}-{rank}")


def _wait_running(pod: str, settings, deadline: float) -> str:
    """Wait for Running; returns the pod IP."""
    while time.time() < deadline:
        out = _run_kubectl(["get", "pod", pod, "-o", "json"], settings)
        if out.returncode == 0:
            st = json.loads(out.stdout).get("status", {})
            if st.get("phase") == "Running":
                return st.get("podIP", "")
            if st.get("phase") in ("Failed", "Unknown"):
                raise ResourcesUnavailableError(
                    f"pod {pod} phase {st.get('phase')}")
        time.sleep(2)
    raise ResourcesUnavailableError(f"pod {pod} never became Running")


def run_instances(cluster_name: str, num_nodes: int, accelerator,
                  acc_count: int, existing_handle: Optional[Dict] = None,
                  use_spot: bool = False) -> Dict[str, Any]:
    """Multi-pod gang: one agent pod per node.  Pod 0 is the head (the
    control plane reaches its agent through a kubectl port-forward
    tunnel); the head's job driver dispatches rank>0 leaf jobs to the
    peer pods' agents at podIP:AGENT_PORT (agent/driver.py peer_agents
    path) — no Ray, no ssh, just the agent HTTP surface
    (reference: multi-node pods via Ray in
    sky/provision/kubernetes/instance.py)."""
    settings = k8s_settings()
    import secrets as _secrets
    token = (existing_handle or {}).get("agent_token") or _secrets.token_hex(16)
    pods = [_pod_name(cluster_name, i) for i in range(num_nodes)]
    for pod in pods:
        manifest = render_pod_manifest(cluster_name, token, acc_count,
                                       settings)
        manifest["metadata"]["name"] = pod
        proc = _run_kubectl(["apply", "-f", "-"], settings,
                            input_text=yaml.safe_dump(manifest))
        if proc.returncode != 0:
            raise ResourcesUnavailableError(
                f"kubectl apply failed: {proc.stderr[:400]}")
    deadline = time.time() + 300
    ips = [_wait_running(pod, settings, deadline) for pod in pods]

    local_port = _port_forward(pods[0], settings)
    AgentClient(local_port, token=token).wait_ready(timeout=60)
    cdir = global_state.root_dir() / "clusters" / cluster_name
    cdir.mkdir(parents=True, exist_ok=True)
    return {
        "cloud": CLOUD_NAME,
        "cluster_dir": str(cdir),
        "pod": pods[0],
        "pods": pods,
        "namespace": settings.get("namespace", "default"),
        "gpu_ids": list(range(acc_count * num_nodes)),
        "num_nodes": num_nodes,
        "gpus_per_node": acc_count,
        "head_ip": ips[0] or "127.0.0.1",
        "node_ips": ips,
        "agent_port": local_port,
        "master_addr": ips[0] or "127.0.0.1",
        "peer_agents": [f"{ip}:{AGENT_PORT}" for ip in ips[1:]],
        "agent_token": token,
    }


def _port_forward(pod: str, settings) -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        local_port = s.getsockname()[1]
    subprocess.Popen(
        _kubectl_base(settings) +
        ["port-forward", f"pod/{pod}", f"{local_port}:{AGENT_PORT}"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        start_new_session=True)
    time.sleep(1.0)
    return local_port


def stop_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    # k8s has no "stopped pod": drain jobs, then delete the pods (the
    # cluster record survives, `sky start` re-provisions fresh pods).
    # Leaving the pod running made a STOPPED cluster flip back to UP on
    # the next refresh (query_instances probes the agent).
    port = handle.get("agent_port")
    if port:
        try:
            AgentClient(port).cancel_all()
        except Exception:  # noqa: BLE001
            pass
    settings = k8s_settings()
    for pod in handle.get("pods") or ([handle["pod"]]
                                      if handle.get("pod") else []):
        try:
            _run_kubectl(["delete", "pod", pod, "--wait=false"], settings)
        except (OSError, subprocess.TimeoutExpired):
            pass


def terminate_instances(cluster_name: str, handle: Dict[str, Any]) -> None:
    stop_instances(cluster_name, handle)


def query_instances(cluster_name: str, handle: Dict[str, Any]) -> str:
    port = handle.get("agent_port")
    if port and AgentClient(port).healthy():
        return global_state.UP
    settings = k8s_settings()
    pod = handle.get("pod")
    if pod:
        try:
            out = _run_kubectl(["get", "pod", pod, "-o", "json"], settings)
            if out.returncode == 0 and json.loads(out.stdout).get(
                    "status", {}).get("phase") == "Running":
                return global_state.INIT  # pod up, tunnel down
        except (OSError, subprocess.TimeoutExpired, json.JSONDecodeError):
            pass
    return global_state.STOPPED


def get_cluster_info(handle: Dict[str, Any]) -> Dict[str, Any]:
    return {"head_ip": "127.0.0.1",
            "node_ips": handle.get("node_ips", []),
            "gpu_ids": handle.get("gpu_ids", [])}
