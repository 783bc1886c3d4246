"""Fault-injection tests (reference: tests/chaos/chaos_proxy.py — here we
kill framework processes directly and assert the state machines
reconcile)."""
import json
import os
import signal
import time
from pathlib import Path

from tests.test_orchestrator import client, sky_env, _wait_job_done  # noqa: F401


def test_driver_killed_marks_failed_driver(client):
    """Kill the job driver process: reconcile must mark FAILED_DRIVER
    (reference: job_lib.py:833 liveness by driver PID)."""
    from skypilot_amd.client import sdk
    res = sdk.get(sdk.launch({"name": "victim", "run": "sleep 600"},
                             "t-chaos1"), timeout=60)
    jid = res["job_id"]
    deadline = time.time() + 30
    driver_pid = None
    while time.time() < deadline:
        j = sdk.get(sdk.job_status("t-chaos1", jid))
        if j and j["status"] == "RUNNING" and j.get("driver_pid"):
            driver_pid = j["driver_pid"]
            break
        time.sleep(0.3)
    assert driver_pid, "job never started"
    os.kill(driver_pid, signal.SIGKILL)  # exact pid, never a pattern
    j = _wait_job_done("t-chaos1", jid, timeout=60)
    assert j["status"] == "FAILED_DRIVER"
    sdk.get(sdk.down("t-chaos1"))


def test_agent_killed_detected_by_refresh(client):
    """Kill the node agent: `sky status -r` must reconcile the cluster to
    STOPPED; `sky start` restarts the agent on the same GPU lease."""
    from skypilot_amd.client import sdk
    sdk.get(sdk.launch({"name": "al", "run": "true",
                        "resources": {"accelerators": "MI355X:2"}},
                       "t-chaos2"), timeout=60)
    home = Path(os.environ["SKY_AMD_HOME"])
    meta = home / "clusters" / "t-chaos2" / "agent.json"
    pid = json.loads(meta.read_text())["pid"]
    os.kill(pid, signal.SIGKILL)
    time.sleep(0.5)
    records = sdk.get(sdk.status(refresh=True))
    rec = next(r for r in records if r["name"] == "t-chaos2")
    assert rec["status"] == "STOPPED"
    # restart reuses the recorded GPU lease
    handle = sdk.get(sdk.start("t-chaos2"))
    assert handle["gpu_ids"] == rec["handle"]["gpu_ids"]
    records = sdk.get(sdk.status())
    rec = next(r for r in records if r["name"] == "t-chaos2")
    assert rec["status"] == "UP"
    sdk.get(sdk.down("t-chaos2"))


def test_network_partition_reports_init_and_recovers(client):
    """SIGSTOP the agent (alive pid, dead HTTP = partition/wedge): the
    status refresh must complete promptly and report INIT, then UP again
    after SIGCONT (reference: abnormal clusters -> INIT)."""
    from skypilot_amd.client import sdk
    sdk.get(sdk.launch({"name": "ap", "run": "true",
                        "resources": {"cpus": 1}}, "t-chaos3"),
            timeout=60)
    home = Path(os.environ["SKY_AMD_HOME"])
    meta = home / "clusters" / "t-chaos3" / "agent.json"
    pid = json.loads(meta.read_text())["pid"]
    os.kill(pid, signal.SIGSTOP)
    try:
        t0 = time.time()
        records = sdk.get(sdk.status(refresh=True), timeout=60)
        elapsed = time.time() - t0
        rec = next(r for r in records if r["name"] == "t-chaos3")
        assert rec["status"] == "INIT", rec
        assert elapsed < 30, f"refresh hung {elapsed:.0f}s on partition"
    finally:
        os.kill(pid, signal.SIGCONT)
    records = sdk.get(sdk.status(refresh=True), timeout=60)
    rec = next(r for r in records if r["name"] == "t-chaos3")
    assert rec["status"] == "UP"
    sdk.get(sdk.down("t-chaos3"))


def test_log_shipping_to_http_sink(client, tmp_path):
    """`logs: {endpoint}` config ships completed-job logs to an external
    HTTP sink (reference: sky/logs/agent.py FluentbitAgent)."""
    import http.server
    import threading

    received = []

    class Sink(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            received.append(json.loads(self.rfile.read(n)))
            self.send_response(200)
            self.end_headers()

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), Sink)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        home = Path(os.environ["SKY_AMD_HOME"])
        (home / "config.yaml").write_text(
            f"logs:\n  endpoint: http://127.0.0.1:{srv.server_port}/\n")
        from skypilot_amd import config as sky_config
        sky_config.load(refresh=True)
        from skypilot_amd.client import sdk
        sdk.get(sdk.launch({"run": "echo shipped-$SKYPILOT_TASK_ID"},
                           "ship-c"), timeout=60)
        deadline = time.time() + 30
        while time.time() < deadline and not received:
            time.sleep(0.3)
        assert received, "no logs shipped"
        rec = next(r for r in received if "shipped-" in r["content"])
        assert rec["status"] == "SUCCEEDED"
        sdk.get(sdk.down("ship-c"))
    finally:
        srv.shutdown()
        (home / "config.yaml").unlink(missing_ok=True)
        from skypilot_amd import config as sky_config
        sky_config.load(refresh=True)


def test_agent_restart_mid_job_reconciles_running_job(client):
    """Agent killed WHILE a job runs (ROUND2_PLAN verification debt):
    restart must bring the agent back on the same lease, and the job
    must reach a terminal state instead of sticking in RUNNING.  The
    driver runs in its own session, so the normal outcome is that the
    job SURVIVES the agent restart and SUCCEEDS (the driver keeps
    writing the shared job table); FAILED_DRIVER is the acceptable
    outcome if the driver did die with the agent."""
    from skypilot_amd.client import sdk
    sdk.get(sdk.launch({"name": "mid", "run": "sleep 30",
                        "resources": {"cpus": 1}},
                       "t-chaos5"), timeout=60)
    home = Path(os.environ["SKY_AMD_HOME"])
    meta = home / "clusters" / "t-chaos5" / "agent.json"
    pid = json.loads(meta.read_text())["pid"]
    # wait for the job to reach RUNNING
    deadline = time.time() + 60
    while time.time() < deadline:
        j = sdk.get(sdk.job_status("t-chaos5", 1))
        if j and j["status"] == "RUNNING":
            break
        time.sleep(0.3)
    assert j["status"] == "RUNNING", j
    os.kill(pid, signal.SIGKILL)
    time.sleep(0.5)
    # refresh sees the dead agent; start brings it back
    recs = sdk.get(sdk.status(refresh=True))
    rec = next(r for r in recs if r["name"] == "t-chaos5")
    assert rec["status"] == "STOPPED"
    sdk.get(sdk.start("t-chaos5"))
    # the restarted agent's reconcile must move the orphaned job out of
    # RUNNING (driver pid is gone)
    deadline = time.time() + 60
    final = None
    while time.time() < deadline:
        j = sdk.get(sdk.job_status("t-chaos5", 1))
        if j and j["status"] in ("FAILED_DRIVER", "FAILED", "CANCELLED",
                                 "SUCCEEDED"):
            final = j["status"]
            break
        time.sleep(0.5)
    assert final in ("SUCCEEDED", "FAILED_DRIVER"), final
    # and the cluster accepts new jobs
    rid = sdk.exec({"run": "echo back"}, "t-chaos5")
    sdk.get(rid, timeout=60)
    deadline = time.time() + 60
    while time.time() < deadline:
        j2 = sdk.get(sdk.job_status("t-chaos5", 2))
        if j2 and j2["status"] == "SUCCEEDED":
            break
        time.sleep(0.3)
    assert j2["status"] == "SUCCEEDED", j2
    sdk.get(sdk.down("t-chaos5"))
