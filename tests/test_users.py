"""RBAC: roles, service-account tokens, per-request authorization and
cluster-ownership enforcement (reference: sky/users/rbac.py,
sky/users/server.py token auth)."""
import pytest

from tests.test_orchestrator import sky_env, client  # noqa: F401 (fixtures)


def test_bootstrap_admin_and_roles(sky_env, client):
    # server's own identity bootstrapped as admin
    me = client.get("/api/users").json()
    assert any(u["role"] == "admin" for u in me)
    # a fresh header identity is auto-created as plain user
    r = client.get("/api/users", headers={"X-Skypilot-User": "alice"})
    assert r.status_code == 200
    roles = {u["name"]: u["role"] for u in client.get("/api/users").json()}
    assert roles["alice"] == "user"


def test_viewer_denied_mutations(sky_env, client):
    client.post("/api/users/role",
                json={"name": "bob", "role": "viewer"},
                headers=None)
    # bob must exist first (auto-create), then demote
    client.get("/api/users", headers={"X-Skypilot-User": "bob"})
    assert client.post("/api/users/role",
                       json={"name": "bob", "role": "viewer"}
                       ).status_code == 200
    hdr = {"X-Skypilot-User": "bob"}
    r = client.post("/api/v1/launch",
                    json={"task": {"run": "true"}}, headers=hdr)
    assert r.status_code == 403
    # read-only still fine
    assert client.post("/api/v1/status", json={}, headers=hdr
                       ).status_code == 200


def test_non_admin_cannot_manage_users(sky_env, client):
    hdr = {"X-Skypilot-User": "carol"}
    client.get("/api/users", headers=hdr)  # auto-create as 'user'
    r = client.post("/api/users/token",
                    json={"name": "ci"}, headers=hdr)
    assert r.status_code == 403
    r = client.post("/api/users/role",
                    json={"name": "carol", "role": "admin"}, headers=hdr)
    assert r.status_code == 403


def test_service_account_token_auth(sky_env, client):
    r = client.post("/api/users/token",
                    json={"name": "ci-bot", "role": "user"})
    assert r.status_code == 200
    tok = r.json()["token"]
    assert tok.startswith("sky_")
    hdr = {"Authorization": f"Bearer {tok}"}
    assert client.post("/api/v1/status", json={},
                       headers=hdr).status_code == 200
    toks = client.get("/api/users/tokens").json()
    assert any(t["name"] == "ci-bot" for t in toks)
    # bad token rejected
    bad = {"Authorization": "Bearer sky_deadbeef"}
    assert client.post("/api/v1/status", json={},
                       headers=bad).status_code == 401
    # revoke kills it
    client.post("/api/users/token/revoke", json={"name": "ci-bot"})
    assert client.post("/api/v1/status", json={},
                       headers=hdr).status_code == 401


def test_token_auth_mode_rejects_header_identity(sky_env, client,
                                                 monkeypatch):
    """In token auth mode (non-local binds) the X-Skypilot-User header
    must NOT grant identity — only bearer tokens."""
    r = client.post("/api/users/token", json={"name": "tok-mode",
                                              "role": "user"})
    tok = r.json()["token"]
    monkeypatch.setenv("SKY_AMD_AUTH_MODE", "token")
    # header identity rejected
    assert client.post("/api/v1/status", json={},
                       headers={"X-Skypilot-User": "imposter"}
                       ).status_code == 401
    # no identity at all rejected too
    assert client.post("/api/v1/status", json={}).status_code == 401
    assert client.get("/api/requests").status_code == 401
    # bearer token still works
    assert client.post("/api/v1/status", json={},
                       headers={"Authorization": f"Bearer {tok}"}
                       ).status_code == 200
    # the dashboard carries the same gate
    assert client.get("/dashboard").status_code == 401
    assert client.get("/dashboard",
                      headers={"Authorization": f"Bearer {tok}"}
                      ).status_code == 200
    # trusted-proxy header accepted only when explicitly configured
    hdr = {"X-Auth-Request-Email": "eve@corp"}
    assert client.post("/api/v1/status", json={},
                       headers=hdr).status_code == 401
    monkeypatch.setenv("SKY_AMD_TRUST_PROXY_AUTH", "1")
    assert client.post("/api/v1/status", json={},
                       headers=hdr).status_code == 200


def test_request_access_scoped_to_owner(sky_env, client):
    """/api/get, /api/cancel and /api/requests are owner-or-admin only."""
    hdr_a = {"X-Skypilot-User": "req-owner"}
    hdr_b = {"X-Skypilot-User": "req-snoop"}
    rid = client.post("/api/v1/status", json={},
                      headers=hdr_a).json()["request_id"]
    # another plain user cannot read, cancel, or list it
    assert client.get("/api/get", params={"request_id": rid},
                      headers=hdr_b).status_code == 403
    assert client.post("/api/cancel", json={"request_id": rid},
                       headers=hdr_b).status_code == 403
    listed = client.get("/api/requests", headers=hdr_b).json()
    assert all(r["request_id"] != rid for r in listed)
    # the owner and the admin can
    assert client.get("/api/get", params={"request_id": rid},
                      headers=hdr_a).status_code == 200
    assert client.get("/api/get", params={"request_id": rid}
                      ).status_code == 200


def test_cluster_ownership_enforced(sky_env, client):
    import time
    # admin (default identity) launches a cluster
    r = client.post("/api/v1/launch",
                    json={"task": {"run": "sleep 0.1"},
                          "cluster_name": "owned-c"})
    rid = r.json()["request_id"]
    deadline = time.time() + 60
    while time.time() < deadline:
        st = client.get("/api/get", params={"request_id": rid}).json()
        if st["status"] in ("SUCCEEDED", "FAILED", "CANCELLED"):
            break
        time.sleep(0.5)
    assert st["status"] == "SUCCEEDED", st
    # another plain user may not down it
    hdr = {"X-Skypilot-User": "mallory"}
    r = client.post("/api/v1/down",
                    json={"cluster_name": "owned-c"}, headers=hdr)
    rid = r.json()["request_id"]
    deadline = time.time() + 60
    while time.time() < deadline:
        st = client.get("/api/get", params={"request_id": rid}).json()
        if st["status"] in ("SUCCEEDED", "FAILED", "CANCELLED"):
            break
        time.sleep(0.5)
    assert st["status"] == "FAILED"
    assert "PermissionDenied" in (st.get("error") or "")


def test_api_version_handshake(sky_env, client, monkeypatch, capsys):
    """Old-server warning and too-old-client rejection (reference:
    sky/server API version compat)."""
    from skypilot_amd.client import sdk
    from skypilot_amd.exceptions import ApiServerError
    h = client.get("/health").json()
    assert h["api_version"] >= h["min_client_api_version"]
    # same version: silent
    monkeypatch.setattr(sdk, "_version_checked", False)
    sdk.check_server_compat()
    # client older than server's minimum: hard error
    monkeypatch.setattr(sdk, "_version_checked", False)
    monkeypatch.setattr(sdk, "CLIENT_API_VERSION", 0)
    import pytest as _pytest
    with _pytest.raises(ApiServerError):
        sdk.check_server_compat()
    # server older than client: stderr note, no error
    monkeypatch.setattr(sdk, "_version_checked", False)
    monkeypatch.setattr(sdk, "CLIENT_API_VERSION", 99)
    sdk.check_server_compat()
    assert "older than client" in capsys.readouterr().err


def test_cost_report(sky_env, client):
    """GPU-hour accounting includes live and historical clusters."""
    import time
    from skypilot_amd.client import sdk
    sdk.get(sdk.launch({"run": "true", "resources": {"cpus": 1}},
                       "cost-c"), timeout=60)
    rows = sdk.get(sdk.cost_report())
    live = next(r for r in rows if r["name"] == "cost-c")
    assert live["live"] and live["status"] == "UP"
    sdk.get(sdk.down("cost-c"))
    rows = sdk.get(sdk.cost_report())
    hist = [r for r in rows if r["name"] == "cost-c" and not r["live"]]
    assert hist and hist[0]["status"] == "TERMINATED"
    assert hist[0]["duration_hours"] >= 0


def test_metrics_gauges_and_leader(sky_env, client):
    """Executor free-slot gauges, loop-stall metric and leader flag on
    /metrics; leader election is exclusive across 'processes'."""
    body = client.get("/metrics").text
    assert 'sky_amd_executor_free_slots{queue="long"}' in body
    assert 'sky_amd_executor_free_slots{queue="short"}' in body
    assert "sky_amd_loop_stall_max_seconds" in body
    assert "sky_amd_daemons_leader 1" in body
    from skypilot_amd.server import daemons
    assert daemons.is_leader()
    assert daemons.try_acquire_leadership()  # idempotent for the holder
    # a second server process must NOT win the election while this one
    # holds the flock
    import subprocess
    import sys
    code = "\n".join([
        "import fcntl, sys",
        "f = open(sys.argv[1] + '/daemons-leader.lock', 'w')",
        "try:",
        "    fcntl.flock(f, fcntl.LOCK_EX | fcntl.LOCK_NB)",
        "    print('WON')",
        "except OSError:",
        "    print('LOST')",
    ])
    # point the child at the lock dir the current leader actually used
    actual = daemons._leader_lock_file.name.rsplit("/", 1)[0]
    out = subprocess.run([sys.executable, "-c", code, actual],
                         capture_output=True, text=True)
    assert out.stdout.strip() == "LOST", out


def test_workspaces_scope_clusters(sky_env, client, monkeypatch):
    """Clusters are scoped to the active workspace (reference: sky
    workspaces); --all-workspaces sees everything."""
    from skypilot_amd.client import sdk
    monkeypatch.setenv("SKY_AMD_WORKSPACE", "team-a")
    sdk.get(sdk.launch({"run": "true", "resources": {"cpus": 1}},
                       "ws-a"), timeout=60)
    monkeypatch.setenv("SKY_AMD_WORKSPACE", "team-b")
    sdk.get(sdk.launch({"run": "true", "resources": {"cpus": 1}},
                       "ws-b"), timeout=60)
    names = {r["name"] for r in sdk.get(sdk.status())}
    assert "ws-b" in names and "ws-a" not in names
    monkeypatch.setenv("SKY_AMD_WORKSPACE", "team-a")
    names = {r["name"] for r in sdk.get(sdk.status())}
    assert "ws-a" in names and "ws-b" not in names
    allr = sdk.get(sdk.status(all_workspaces=True))
    ws = {r["name"]: r["workspace"] for r in allr}
    assert ws["ws-a"] == "team-a" and ws["ws-b"] == "team-b"
    sdk.get(sdk.down("ws-a"))
    monkeypatch.setenv("SKY_AMD_WORKSPACE", "team-b")
    sdk.get(sdk.down("ws-b"))


def test_workspace_rbac(sky_env, client, monkeypatch):
    """Private workspaces admit only allowed_users (reference: sky
    workspaces private/allowed_users); open workspaces stay open and
    admins always pass."""
    import os
    from pathlib import Path
    home = Path(os.environ["SKY_AMD_HOME"])
    home.mkdir(parents=True, exist_ok=True)
    (home / "config.yaml").write_text(
        "workspaces:\n  secret:\n    private: true\n"
        "    allowed_users: [alice]\n")
    from skypilot_amd import config as sky_config
    sky_config.load(refresh=True)
    try:
        hdr_a = {"X-Skypilot-User": "alice",
                 "X-Skypilot-Workspace": "secret"}
        hdr_b = {"X-Skypilot-User": "bob",
                 "X-Skypilot-Workspace": "secret"}
        hdr_open = {"X-Skypilot-User": "bob",
                    "X-Skypilot-Workspace": "open-ws"}
        assert client.post("/api/v1/status", json={},
                           headers=hdr_a).status_code == 200
        r = client.post("/api/v1/status", json={}, headers=hdr_b)
        assert r.status_code == 403 and "private" in r.text
        assert client.post("/api/v1/status", json={},
                           headers=hdr_open).status_code == 200
        # admin (server identity) passes everywhere
        assert client.post("/api/v1/status", json={},
                           headers={"X-Skypilot-Workspace": "secret"}
                           ).status_code == 200
    finally:
        (home / "config.yaml").unlink(missing_ok=True)
        sky_config.load(refresh=True)
