"""Integration: the API server as a real uvicorn process (the path
`sky api start` uses), exercised over real HTTP — TestClient-based
suites bypass the server process entirely."""
import json
import os
import signal
import subprocess
import sys
import time
from pathlib import Path

import httpx
import pytest


@pytest.fixture()
def real_server(tmp_path, monkeypatch):
    home = tmp_path / "home"
    monkeypatch.setenv("SKY_AMD_HOME", str(home))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "8")
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    monkeypatch.setenv("SKY_AMD_API_PORT", str(port))
    log = open(tmp_path / "server.log", "ab")
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.server.app", "--port",
         str(port)],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env=dict(os.environ))
    base = f"http://127.0.0.1:{port}"
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            if httpx.get(base + "/health", timeout=2).status_code == 200:
                break
        except Exception:
            time.sleep(0.3)
    else:
        proc.kill()
        raise RuntimeError("server never became healthy: "
                           + (tmp_path / "server.log").read_text()[-2000:])
    yield base
    try:
        os.killpg(proc.pid, signal.SIGTERM)
    except ProcessLookupError:
        pass
    # reap any agents
    for meta in home.glob("clusters/*/agent.json"):
        try:
            os.kill(json.loads(meta.read_text())["pid"], signal.SIGTERM)
        except (OSError, ValueError):
            pass


def _wait_req(base, rid, timeout=90):
    deadline = time.time() + timeout
    while time.time() < deadline:
        st = httpx.get(base + "/api/get",
                       params={"request_id": rid}, timeout=10).json()
        if st["status"] in ("SUCCEEDED", "FAILED", "CANCELLED"):
            return st
        time.sleep(0.5)
    raise TimeoutError(rid)


def test_real_server_launch_roundtrip(real_server):
    base = real_server
    h = httpx.get(base + "/health", timeout=5).json()
    assert h["ok"] and h["api_version"] >= 1
    r = httpx.post(base + "/api/v1/launch",
                   json={"task": {"run": "echo real-http-ok"},
                         "cluster_name": "real-c"}, timeout=10)
    assert r.status_code == 200
    st = _wait_req(base, r.json()["request_id"])
    assert st["status"] == "SUCCEEDED", st
    # status via the real server
    r = httpx.post(base + "/api/v1/status", json={}, timeout=10)
    st = _wait_req(base, r.json()["request_id"])
    names = [c["name"] for c in st["result"]]
    assert "real-c" in names
    # metrics endpoint live
    m = httpx.get(base + "/metrics", timeout=5).text
    assert "sky_amd_clusters_up" in m
    # workspace header travels with the request (LONG subprocess path)
    r = httpx.post(base + "/api/v1/launch",
                   json={"task": {"run": "true"}, "cluster_name": "ws-c"},
                   headers={"X-Skypilot-Workspace": "hdr-ws"}, timeout=10)
    st = _wait_req(base, r.json()["request_id"])
    assert st["status"] == "SUCCEEDED", st
    r = httpx.post(base + "/api/v1/status", json={"all_workspaces": True},
                   timeout=10)
    st = _wait_req(base, r.json()["request_id"])
    ws = {c["name"]: c.get("workspace") for c in st["result"]}
    assert ws.get("ws-c") == "hdr-ws", ws
    for name in ("real-c", "ws-c"):
        r = httpx.post(base + "/api/v1/down",
                       json={"cluster_name": name}, timeout=10)
        st = _wait_req(base, r.json()["request_id"])
        assert st["status"] == "SUCCEEDED"
