"""Sky-Serve tests: replica manager, LB proxy, recovery, autoscaler unit
tests (reference: SURVEY.md §2.8/§3.3)."""
import time

import pytest

from tests.test_orchestrator import client, sky_env  # fixtures  # noqa: F401

REPLICA_SERVER = (
    "python3 -c '\n"
    "import os, http.server as h\n"
    "rid = os.environ.get(\"SKYPILOT_TASK_ID\", \"?\")\n"
    "class H(h.BaseHTTPRequestHandler):\n"
    "    def do_GET(self):\n"
    "        self.send_response(200); self.end_headers()\n"
    "        self.wfile.write((\"replica:\" + rid).encode())\n"
    "    def log_message(self, *a):\n"
    "        pass\n"
    "h.HTTPServer((\"127.0.0.1\", int(os.environ[\"PORT\"])), H)"
    ".serve_forever()'\n"
)


def _service_task(replicas=2):
    return {
        "name": "echo-svc",
        "service": {
            "readiness_probe": {"path": "/", "initial_delay_seconds": 60},
            "replicas": replicas,
        },
        "run": REPLICA_SERVER,
    }


def _wait_ready(service, n, timeout=90):
    from skypilot_amd.client import sdk
    deadline = time.time() + timeout
    while time.time() < deadline:
        stats = sdk.get(sdk.serve_status(service))
        if stats:
            ready = [r for r in stats[0]["replicas"]
                     if r["status"] == "READY"]
            if len(ready) >= n and stats[0]["status"] == "READY":
                return stats[0]
        time.sleep(1)
    raise TimeoutError(f"service {service} never reached {n} ready; "
                       f"last={stats}")


def test_serve_up_proxy_and_down(client):
    import httpx
    from skypilot_amd.client import sdk
    res = sdk.get(sdk.serve_up(_service_task(2), "svc1"), timeout=120)
    endpoint = res["endpoint"]
    stat = _wait_ready("svc1", 2)
    assert stat["status"] == "READY"
    # Proxy through the LB: both replicas answer over multiple requests.
    seen = set()
    for _ in range(10):
        r = httpx.get(endpoint + "/", timeout=10)
        assert r.status_code == 200
        assert r.text.startswith("replica:")
        seen.add(r.text)
    assert len(seen) >= 1  # least-load may favor one when idle
    sdk.get(sdk.serve_down("svc1"), timeout=120)
    stats = sdk.get(sdk.serve_status("svc1"))
    assert stats == []
    # Replica clusters cleaned up.
    records = sdk.get(sdk.status())
    assert not any(r["name"].startswith("sky-serve-svc1") for r in records)


def test_serve_replica_recovery(client):
    from skypilot_amd.client import sdk
    sdk.get(sdk.serve_up(_service_task(1), "svc2"), timeout=120)
    stat = _wait_ready("svc2", 1)
    victim = stat["replicas"][0]
    # Kill the replica's cluster out from under the controller.
    sdk.get(sdk.down(victim["cluster_name"]))
    time.sleep(1)
    # Controller must notice and bring up a replacement replica.
    deadline = time.time() + 120
    while time.time() < deadline:
        stats = sdk.get(sdk.serve_status("svc2"))
        ready = [r for r in stats[0]["replicas"]
                 if r["status"] == "READY"
                 and r["replica_id"] != victim["replica_id"]]
        if ready:
            break
        time.sleep(1)
    else:
        pytest.fail("replacement replica never became READY")
    sdk.get(sdk.serve_down("svc2"), timeout=120)


def test_request_rate_autoscaler_hysteresis():
    from skypilot_amd.serve.autoscalers import RequestRateAutoscaler
    from skypilot_amd.serve.service_spec import ReplicaPolicy
    pol = ReplicaPolicy(min_replicas=1, max_replicas=4,
                        target_qps_per_replica=2.0,
                        upscale_delay_seconds=0,
                        downscale_delay_seconds=3600)
    a = RequestRateAutoscaler(pol)
    # qps 7 -> ceil(7/2)=4 replicas, no delay configured upward.
    assert a.target_replicas(7.0, 1) in (1, 4)
    assert a.target_replicas(7.0, 1) == 4  # second tick past 0s delay
    # Downscale is held back by the long delay.
    assert a.target_replicas(0.0, 4) == 4


def test_service_spec_validation():
    from skypilot_amd.exceptions import TaskValidationError
    from skypilot_amd.serve.service_spec import ServiceSpec
    spec = ServiceSpec.from_config({
        "readiness_probe": "/health",
        "replica_policy": {"min_replicas": 2, "max_replicas": 8,
                           "target_qps_per_replica": 3}})
    assert spec.readiness_probe.path == "/health"
    assert spec.policy.max_replicas == 8
    with pytest.raises(TaskValidationError):
        ServiceSpec.from_config({"replicas": 2,
                                 "replica_policy": {"min_replicas": 1}})
    with pytest.raises(TaskValidationError):
        ServiceSpec.from_config({"bogus_key": 1})


def test_tls_spec_and_selfsigned_materialization(tmp_path):
    """`tls: true` generates a self-signed pair; explicit paths pass
    through (reference: serve schema tls keyfile/certfile)."""
    from skypilot_amd.serve.service_spec import ServiceSpec, TLSConfig
    spec = ServiceSpec.from_config({"ports": 9999, "tls": True})
    assert spec.tls is not None and spec.tls.auto
    tls = spec.tls.ensure_materialized(tmp_path / "tls")
    import pathlib
    assert pathlib.Path(tls.certfile).exists()
    assert pathlib.Path(tls.keyfile).exists()
    cert_pem = pathlib.Path(tls.certfile).read_text()
    assert "BEGIN CERTIFICATE" in cert_pem
    explicit = TLSConfig.from_config(
        {"certfile": tls.certfile, "keyfile": tls.keyfile})
    assert explicit.ensure_materialized(tmp_path).certfile == tls.certfile


def test_tls_lb_serves_https(tmp_path):
    """A uvicorn server with the generated pair answers HTTPS (the exact
    config the serve LB passes)."""
    import threading

    import httpx
    import uvicorn
    from fastapi import FastAPI

    from skypilot_amd.serve.service_spec import TLSConfig
    tls = TLSConfig(auto=True).ensure_materialized(tmp_path)
    app = FastAPI()

    @app.get("/ping")
    def ping():
        return {"ok": True}

    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    config = uvicorn.Config(app, host="127.0.0.1", port=port,
                            log_level="error", ssl_certfile=tls.certfile,
                            ssl_keyfile=tls.keyfile)
    server = uvicorn.Server(config)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    import time
    deadline = time.time() + 15
    last = None
    while time.time() < deadline:
        try:
            r = httpx.get(f"https://127.0.0.1:{port}/ping", verify=False)
            assert r.json()["ok"]
            break
        except Exception as e:  # noqa: BLE001
            last = e
            time.sleep(0.3)
    else:
        raise AssertionError(f"https never came up: {last}")
    server.should_exit = True
    t.join(10)


def test_lb_under_replica_churn(client):
    """Requests keep flowing through the LB while a replica dies and is
    replaced: error rate stays low (survivor takes traffic) and the
    service returns to full readiness (round-2 verification-debt
    item)."""
    import threading

    import httpx

    from skypilot_amd.client import sdk
    res = sdk.get(sdk.serve_up(_service_task(2), "churn"), timeout=180)
    endpoint = res["endpoint"]
    stat = _wait_ready("churn", 2)
    stop = threading.Event()
    results = {"ok": 0, "fail": 0}

    def hammer():
        while not stop.is_set():
            try:
                r = httpx.get(endpoint + "/", timeout=5)
                if r.status_code == 200:
                    results["ok"] += 1
                else:
                    results["fail"] += 1
            except Exception:  # noqa: BLE001
                results["fail"] += 1
            time.sleep(0.05)

    t = threading.Thread(target=hammer)
    t.start()
    try:
        time.sleep(2)
        victim = stat["replicas"][0]
        sdk.get(sdk.down(victim["cluster_name"]))
        # keep hammering through detection + replacement
        deadline = time.time() + 120
        while time.time() < deadline:
            stats = sdk.get(sdk.serve_status("churn"))
            ready = [r for r in stats[0]["replicas"]
                     if r["status"] == "READY"]
            if len(ready) >= 2 and all(
                    r["replica_id"] != victim["replica_id"]
                    or r["status"] == "READY" for r in ready):
                break
            time.sleep(1)
        time.sleep(2)
    finally:
        stop.set()
        t.join(10)
    total = results["ok"] + results["fail"]
    assert total > 20, results
    # the LB may lose a few in-flight requests at the kill instant but
    # must keep the service usable throughout
    assert results["fail"] / total < 0.3, results
    assert results["ok"] > 0
    sdk.get(sdk.serve_down("churn"), timeout=120)


def test_serve_rolling_update(client):
    """`sky serve update` bumps the service version and the controller
    replaces stale replicas one at a time while the service stays
    reachable (reference: sky/serve rolling updates / version
    tracking)."""
    import httpx
    from skypilot_amd.client import sdk
    task = _service_task(2)
    task["envs"] = {"SVC_TAG": "v1"}
    task["run"] = task["run"].replace("\"replica:\"", "\"replica-\" + "
                                      "os.environ.get(\"SVC_TAG\",\"\") + "
                                      "\":\"")
    res = sdk.get(sdk.serve_up(task, "svc-roll"), timeout=120)
    endpoint = res["endpoint"]
    _wait_ready("svc-roll", 2)
    r = httpx.get(endpoint + "/", timeout=10)
    assert r.text.startswith("replica-v1:")
    # rolling update to v2
    task2 = dict(task)
    task2["envs"] = {"SVC_TAG": "v2"}
    up2 = sdk.get(sdk.serve_update(task2, "svc-roll"), timeout=60)
    assert up2["version"] == 2
    deadline = time.time() + 180
    ok = False
    while time.time() < deadline:
        stats = sdk.get(sdk.serve_status("svc-roll"))
        reps = stats[0]["replicas"]
        ready = [x for x in reps if x["status"] == "READY"]
        # availability: never zero ready replicas during the roll
        assert len(ready) >= 1, reps
        if (len(ready) >= 2
                and all(x.get("version", 1) == 2 for x in ready)):
            ok = True
            break
        time.sleep(1)
    assert ok, stats
    # traffic now serves the new version
    deadline = time.time() + 30
    while time.time() < deadline:
        txt = httpx.get(endpoint + "/", timeout=10).text
        if txt.startswith("replica-v2:"):
            break
        time.sleep(0.5)
    assert txt.startswith("replica-v2:"), txt
    sdk.get(sdk.serve_down("svc-roll"), timeout=120)
