"""Peer-agent gang dispatch: the head driver runs rank 0 locally and
POSTs rank>0 leaf jobs to peer node agents (the multi-pod k8s gang
path, agent/driver.py peer_agents; reference: Ray placement-group gangs
in sky/provision/kubernetes).  Exercised with two real local agents."""
import json
import os
import signal
import subprocess
import sys
import time
from pathlib import Path

import pytest

from skypilot_amd.agent.client import AgentClient


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture()
def two_agents(tmp_path):
    agents = []
    try:
        for name in ("head", "peer"):
            cdir = tmp_path / name
            cdir.mkdir()
            port = _free_port()
            log = open(cdir / "agent.log", "ab")
            p = subprocess.Popen(
                [sys.executable, "-m", "skypilot_amd.agent.daemon",
                 "--cluster-dir", str(cdir), "--port", str(port)],
                stdout=log, stderr=subprocess.STDOUT,
                start_new_session=True,
                env={**os.environ, "SKY_AMD_HOME": str(tmp_path / "home")})
            log.close()
            AgentClient(port).wait_ready(timeout=30)
            agents.append((cdir, port, p))
        yield agents
    finally:
        for _, _, p in agents:
            try:
                os.killpg(p.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass


def test_two_node_gang_via_peer_agent(two_agents, tmp_path):
    (head_dir, head_port, _), (peer_dir, peer_port, _) = two_agents
    out = tmp_path / "out"
    out.mkdir()
    run = ("echo rank=$SKYPILOT_NODE_RANK of=$SKYPILOT_NUM_NODES "
           "master=$MASTER_ADDR task=$SKYPILOT_TASK_ID > "
           f"{out}/$SKYPILOT_NODE_RANK.txt")
    head = AgentClient(head_port)
    jid = head.queue_job({
        "run": run, "num_nodes": 2, "gpus_per_node": 0,
        "node_ips": ["127.0.0.1", "127.0.0.1"],
        "master_addr": "10.9.8.7",
        "peer_agents": [f"127.0.0.1:{peer_port}"],
        "task_id": "gang-test-1",
    }, name="gang")
    job = head.wait_job(jid, timeout=90, poll=0.5)
    assert job["status"] == "SUCCEEDED", job
    r0 = (out / "0.txt").read_text()
    r1 = (out / "1.txt").read_text()
    assert "rank=0 of=2 master=10.9.8.7 task=gang-test-1" in r0
    assert "rank=1 of=2 master=10.9.8.7 task=gang-test-1" in r1
    # the leaf ran on the PEER agent (its job table has it)
    peer_jobs = AgentClient(peer_port).get_job_queue()
    assert any((j.get("name") or "").endswith("rank1") for j in peer_jobs)
    # the peer's log tail was pulled next to the head's rank logs
    logdir = head_dir / "logs" / str(jid)
    assert (logdir / "1-node.log").exists()


def test_gang_failure_propagates_and_cancels(two_agents, tmp_path):
    (head_dir, head_port, _), (peer_dir, peer_port, _) = two_agents
    head = AgentClient(head_port)
    # rank 0 fails fast; the leaf (sleep) must be cancelled, job FAILED
    jid = head.queue_job({
        "run": 'if [ "$SKYPILOT_NODE_RANK" = 0 ]; then exit 3; '
               "else sleep 300; fi",
        "num_nodes": 2, "gpus_per_node": 0,
        "peer_agents": [f"127.0.0.1:{peer_port}"],
        "task_id": "gang-test-2",
    }, name="gang-fail")
    job = head.wait_job(jid, timeout=90, poll=0.5)
    assert job["status"] == "FAILED"
    deadline = time.time() + 30
    while time.time() < deadline:
        peer_jobs = AgentClient(peer_port).get_job_queue()
        leaf = [j for j in peer_jobs
                if (j.get("name") or "").endswith("rank1")]
        if leaf and leaf[0]["status"] in ("CANCELLED", "FAILED",
                                          "FAILED_DRIVER"):
            break
        time.sleep(0.5)
    else:
        raise AssertionError(f"leaf never cancelled: {peer_jobs}")


def test_k8s_multipod_provision_with_stub_kubectl(tmp_path, monkeypatch):
    """Multi-pod run_instances: N pods applied, IPs collected, handle
    carries peer_agents + master_addr (stub kubectl, no cluster)."""
    import stat

    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path / "home"))
    stub = tmp_path / "bin" / "kubectl"
    stub.parent.mkdir()
    stub.write_text(f"""#!/bin/bash
# stub kubectl: record applies, answer get-pod with Running + an IP
d={tmp_path}/k8s; mkdir -p $d
args="$*"
case "$args" in
  *"apply -f -"*) cat > $d/apply_$RANDOM.yaml; echo created;;
  *"get pod"*)
    pod=$(echo "$args" | sed 's/.*get pod \\([^ ]*\\).*/\\1/')
    n=$(echo "$pod" | grep -o '[0-9]*$'); n=${{n:-0}}
    echo '{{"status": {{"phase": "Running", "podIP": "10.244.0.'$((10+n))'"}}}}';;
  *"port-forward"*) sleep 5;;
  *"delete pod"*) echo deleted;;
esac
""")
    stub.chmod(stub.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", f"{stub.parent}:{os.environ['PATH']}")

    from skypilot_amd.provision import k8s
    monkeypatch.setattr(k8s, "_port_forward", lambda pod, s: 45999)
    monkeypatch.setattr(k8s.AgentClient, "wait_ready",
                        lambda self, timeout=30: None)
    handle = k8s.run_instances("gangk", 3, "MI355X", 8)
    assert handle["num_nodes"] == 3
    assert handle["pods"] == ["sky-amd-gangk", "sky-amd-gangk-1",
                              "sky-amd-gangk-2"]
    assert handle["master_addr"] == "10.244.0.10"
    assert handle["peer_agents"] == ["10.244.0.11:46590",
                                     "10.244.0.12:46590"]
    applies = list((tmp_path / "k8s").glob("apply_*.yaml"))
    assert len(applies) == 3
    import yaml as _yaml
    names = sorted(_yaml.safe_load(a.read_text())["metadata"]["name"]
                   for a in applies)
    assert names == sorted(handle["pods"])
    k8s.terminate_instances("gangk", handle)
