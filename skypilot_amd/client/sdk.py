"""Client SDK — one function per REST endpoint, async request IDs.

Reference: sky/client/sdk.py (launch:694, get:2409, stream_and_get).
Every call POSTs to the API server and returns a request_id; results are
fetched with get()/stream_and_get().  If no server is running on
localhost it is auto-started (reference: `sky api start` implicit).
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import time
from typing import Any, Dict, List, Optional

import httpx

from skypilot_amd.exceptions import ApiServerError, SkyAmdError
from skypilot_amd.server.app import server_url
from skypilot_amd.task import Task

_TEST_CLIENT = None  # set by tests to route in-process (TestClient)


def use_test_client(client) -> None:
    global _TEST_CLIENT
    _TEST_CLIENT = client


from contextlib import contextmanager


def _auth_headers() -> dict:
    """Identity for RBAC: a service-account bearer token
    (SKY_AMD_API_TOKEN) wins; otherwise the local username travels in
    X-Skypilot-User (reference: sky client auth headers)."""
    h = {}
    ws = os.environ.get("SKY_AMD_WORKSPACE")
    if ws:
        h["X-Skypilot-Workspace"] = ws
    tok = os.environ.get("SKY_AMD_API_TOKEN")
    if tok:
        h["Authorization"] = f"Bearer {tok}"
        return h
    user = os.environ.get("SKY_AMD_USER") or os.environ.get("USER")
    if user:
        h["X-Skypilot-User"] = user
    return h


@contextmanager
def _client():
    if _TEST_CLIENT is not None:
        yield _TEST_CLIENT  # never closed here; tests own its lifecycle
    else:
        c = httpx.Client(base_url=server_url(), timeout=30.0,
                         headers=_auth_headers())
        try:
            yield c
        finally:
            c.close()


def api_start(wait: float = 15.0) -> bool:
    """Start the local API server if not running."""
    if api_healthy():
        return False
    env = dict(os.environ)
    log = open(os.path.expanduser("~/.sky_amd_api.log"), "ab")
    subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.server.app"],
        stdout=log, stderr=subprocess.STDOUT, start_new_session=True,
        env=env)
    log.close()
    deadline = time.time() + wait
    while time.time() < deadline:
        if api_healthy():
            return True
        time.sleep(0.3)
    raise ApiServerError("API server failed to start; see ~/.sky_amd_api.log")


CLIENT_API_VERSION = 2
_version_checked = False


def check_server_compat() -> None:
    """One-shot client/server API-version handshake (reference:
    sky/server versions compat).  Old server + new client: warn and
    continue; server that no longer serves this client: hard error."""
    global _version_checked
    if _version_checked:
        return
    try:
        with _client() as c:
            info = c.get("/health").json()
    except Exception:  # noqa: BLE001
        return  # health failures surface elsewhere
    _version_checked = True
    srv = info.get("api_version", 1)
    min_client = info.get("min_client_api_version", 1)
    if CLIENT_API_VERSION < min_client:
        raise ApiServerError(
            f"this client speaks API v{CLIENT_API_VERSION} but the "
            f"server requires >= v{min_client}; upgrade the client")
    if srv < CLIENT_API_VERSION:
        import sys
        print(f"[sky] note: server API v{srv} is older than client "
              f"v{CLIENT_API_VERSION}; some flags may be ignored",
              file=sys.stderr)


def api_healthy() -> bool:
    try:
        with _client() as c:
            return c.get("/health").status_code == 200
    except (httpx.HTTPError, OSError):
        return False


def api_stop() -> bool:
    from skypilot_amd import global_state
    meta = global_state.root_dir() / "api" / "server.json"
    if not meta.exists():
        return False
    try:
        pid = json.loads(meta.read_text()).get("pid")
        if pid:
            import signal
            os.kill(pid, signal.SIGTERM)
            return True
    except (OSError, ValueError, ProcessLookupError):
        pass
    return False


def _ensure_server():
    if _TEST_CLIENT is None and not api_healthy():
        api_start()


def _submit(name: str, body: Dict[str, Any]) -> str:
    _ensure_server()
    check_server_compat()
    with _client() as c:
        r = c.post(f"/api/v1/{name}", json=body)
        if r.status_code != 200:
            raise ApiServerError(f"{name}: {r.status_code} {r.text[:400]}")
        return r.json()["request_id"]


def get(request_id: str, timeout: float = 3600.0, poll: float = 0.3) -> Any:
    """Block until the request finishes; return its result
    (reference: sdk.py:2409)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        with _client() as c:
            r = c.get("/api/get", params={"request_id": request_id})
            if r.status_code != 200:
                raise ApiServerError(r.text[:400])
            data = r.json()
        if data["status"] == "SUCCEEDED":
            return data["result"]
        if data["status"] == "FAILED":
            raise SkyAmdError(data["error"] or "request failed")
        if data["status"] == "CANCELLED":
            raise SkyAmdError("request cancelled")
        time.sleep(poll)
    raise TimeoutError(f"request {request_id} timed out")


def stream_and_get(request_id: str, out=None) -> Any:
    out = out or sys.stdout
    if _TEST_CLIENT is not None:
        r = _TEST_CLIENT.get("/api/stream",
                             params={"request_id": request_id})
        out.write(r.text)
        return get(request_id)
    with httpx.Client(base_url=server_url(), timeout=None) as c:
        with c.stream("GET", "/api/stream",
                      params={"request_id": request_id}) as r:
            for chunk in r.iter_text():
                out.write(chunk)
                out.flush()
    return get(request_id)


def cancel_request(request_id: str) -> bool:
    with _client() as c:
        r = c.post("/api/cancel", json={"request_id": request_id})
        return r.json().get("cancelled", False)


UPLOAD_CHUNK_BYTES = 8 << 20


def upload_path(path: str) -> str:
    """Tar+gzip a local directory (or file) and ship it to the API
    server in chunks (reference: client/common.py:154-192 chunked
    upload to /upload).  Returns the upload id for task markers."""
    import tarfile
    import tempfile
    import uuid

    upload_id = uuid.uuid4().hex
    src = os.path.expanduser(path)
    with tempfile.TemporaryDirectory() as td:
        tarball = os.path.join(td, "u.tar.gz")
        with tarfile.open(tarball, "w:gz") as tf:
            if os.path.isdir(src):
                for entry in sorted(os.listdir(src)):
                    tf.add(os.path.join(src, entry), arcname=entry)
            else:
                tf.add(src, arcname=os.path.basename(src))
        size = os.path.getsize(tarball)
        total = max(1, (size + UPLOAD_CHUNK_BYTES - 1) // UPLOAD_CHUNK_BYTES)
        with _client() as c, open(tarball, "rb") as f:
            for i in range(total):
                chunk = f.read(UPLOAD_CHUNK_BYTES)
                r = c.post("/api/upload",
                           params={"upload_id": upload_id,
                                   "chunk_index": i,
                                   "total_chunks": total},
                           content=chunk)
                if r.status_code != 200:
                    raise ApiServerError(
                        f"upload failed: {r.status_code} {r.text[:200]}")
    return upload_id


def _server_is_remote() -> bool:
    """True when the API server is not on this host — local paths in the
    task are then invisible to it and must be uploaded."""
    if os.environ.get("SKY_AMD_FORCE_UPLOAD") == "1":
        return True
    if _TEST_CLIENT is not None:
        return False
    from urllib.parse import urlparse
    host = urlparse(server_url()).hostname or "127.0.0.1"
    return host not in ("127.0.0.1", "localhost", "::1")


def _task_body(task) -> Dict[str, Any]:
    body = task.to_yaml_config() if isinstance(task, Task) else dict(task)
    if _server_is_remote():
        wd = body.get("workdir")
        if isinstance(wd, str) and os.path.exists(os.path.expanduser(wd)):
            body["workdir"] = {"upload": upload_path(wd)}
        fm = body.get("file_mounts")
        if fm:
            body["file_mounts"] = {
                k: ({"upload": upload_path(v)}
                    if isinstance(v, str) and not v.startswith(("s3://",))
                    and os.path.exists(os.path.expanduser(v)) else v)
                for k, v in fm.items()}
    return body


# ---- public API (mirrors reference sdk surface) ---------------------------
def launch(task, cluster_name: Optional[str] = None, *, down: bool = False,
           idle_minutes_to_autostop: Optional[int] = None,
           detach_run: bool = True, retry_until_up: bool = False,
           dryrun: bool = False) -> str:
    return _submit("launch", {
        "task": _task_body(task), "cluster_name": cluster_name,
        "down": down,
        "idle_minutes_to_autostop": idle_minutes_to_autostop,
        "detach_run": detach_run, "retry_until_up": retry_until_up,
        "dryrun": dryrun})


def exec(task, cluster_name: str, *, detach_run: bool = True) -> str:  # noqa: A001
    return _submit("exec", {"task": _task_body(task),
                            "cluster_name": cluster_name,
                            "detach_run": detach_run})


def status(cluster_names: Optional[List[str]] = None,
           refresh: bool = False, all_workspaces: bool = False) -> str:
    return _submit("status", {"cluster_names": cluster_names,
                              "refresh": refresh,
                              "all_workspaces": all_workspaces})


def start(cluster_name: str) -> str:
    return _submit("start", {"cluster_name": cluster_name})


def stop(cluster_name: str) -> str:
    return _submit("stop", {"cluster_name": cluster_name})


def down(cluster_name: str) -> str:
    return _submit("down", {"cluster_name": cluster_name})


def autostop(cluster_name: str, idle_minutes: int, down: bool = False) -> str:
    return _submit("autostop", {"cluster_name": cluster_name,
                                "idle_minutes": idle_minutes, "down": down})


def queue(cluster_name: str) -> str:
    return _submit("queue", {"cluster_name": cluster_name})


def cancel(cluster_name: str, job_ids: Optional[List[int]] = None,
           all_jobs: bool = False) -> str:
    return _submit("cancel", {"cluster_name": cluster_name,
                              "job_ids": job_ids, "all_jobs": all_jobs})


def job_status(cluster_name: str, job_id: int) -> str:
    return _submit("job_status", {"cluster_name": cluster_name,
                                  "job_id": job_id})


def jobs_group_launch(name: str, tasks):
    return _submit("jobs_group_launch", {"name": name, "tasks": tasks})


def jobs_group_status(name: str):
    return _submit("jobs_group_status", {"name": name})


def jobs_group_down(name: str):
    return _submit("jobs_group_down", {"name": name})


def storage_sync(name: str):
    return _submit("storage_sync", {"name": name})


def cost_report():
    return _submit("cost_report", {})


def check() -> str:
    return _submit("check", {})


def show_gpus() -> str:
    return _submit("show_gpus", {})


def cluster_events(cluster_name: str) -> str:
    return _submit("cluster_events", {"cluster_name": cluster_name})


def tail_logs(cluster_name: str, job_id: Optional[int] = None,
              follow: bool = True, out=None):
    """Stream job logs to `out` (direct streaming route, not a request)."""
    out = out or sys.stdout
    _ensure_server()
    if _TEST_CLIENT is not None:
        r = _TEST_CLIENT.get(f"/api/v1-logs/{cluster_name}",
                             params={"job_id": job_id, "follow": follow})
        out.write(r.text)
        return
    with httpx.Client(base_url=server_url(), timeout=None) as c:
        params = {"follow": follow}
        if job_id is not None:
            params["job_id"] = job_id
        with c.stream("GET", f"/api/v1-logs/{cluster_name}",
                      params=params) as r:
            for chunk in r.iter_text():
                out.write(chunk)
                out.flush()


# managed jobs / serve ------------------------------------------------------
def jobs_launch(task, name: Optional[str] = None) -> str:
    return _submit("jobs_launch", {"task": _task_body(task), "name": name})


def jobs_queue() -> str:
    return _submit("jobs_queue", {})


def jobs_cancel(job_ids: Optional[List[int]] = None,
                all_jobs: bool = False) -> str:
    return _submit("jobs_cancel", {"job_ids": job_ids, "all_jobs": all_jobs})


def jobs_logs(job_id: int) -> str:
    return _submit("jobs_logs", {"job_id": job_id})


def jobs_pool_apply(name: str, template, num_workers: int = 2,
                    min_workers: Optional[int] = None,
                    max_workers: Optional[int] = None) -> str:
    return _submit("jobs_pool_apply", {
        "name": name, "template": _task_body(template),
        "num_workers": num_workers, "min_workers": min_workers,
        "max_workers": max_workers})


def jobs_pool_status(name: Optional[str] = None) -> str:
    return _submit("jobs_pool_status", {"name": name})


def jobs_pool_down(name: str) -> str:
    return _submit("jobs_pool_down", {"name": name})


def serve_up(task, service_name: str) -> str:
    return _submit("serve_up", {"task": _task_body(task),
                                "service_name": service_name})


def serve_update(task, service_name: str) -> str:
    return _submit("serve_update", {"task": _task_body(task),
                                    "service_name": service_name})


def serve_down(service_name: str) -> str:
    return _submit("serve_down", {"service_name": service_name})


def serve_logs(service_name: str, replica_id: Optional[int] = None) -> str:
    return _submit("serve_logs", {"service_name": service_name,
                                  "replica_id": replica_id})


def serve_status(service_name: Optional[str] = None) -> str:
    return _submit("serve_status", {"service_name": service_name})


def volumes_list() -> str:
    return _submit("volumes_list", {})


def volumes_create(name: str, size_gb: Optional[int] = None) -> str:
    return _submit("volumes_create", {"name": name, "size_gb": size_gb})


def volumes_delete(name: str) -> str:
    return _submit("volumes_delete", {"name": name})


def recipes_list() -> str:
    return _submit("recipes_list", {})


def storage_list() -> str:
    return _submit("storage_list", {})


def storage_delete(name: str) -> str:
    return _submit("storage_delete", {"name": name})


# interactive shell tunnel (reference: `sky ssh` + the websocket SSH
# proxy in sky/server/server.py; HTTP-streaming PTY sessions here — see
# agent/daemon.py exec sessions) --------------------------------------------
def ssh_start(cluster_name: str, cmd=None, env: Optional[dict] = None) -> str:
    body: Dict[str, Any] = {}
    if cmd:
        body["cmd"] = cmd
    if env:
        body["env"] = env
    with _client() as c:
        r = c.post(f"/api/v1/ssh/{cluster_name}/start", json=body,
                   headers=_auth_headers())
        if r.status_code != 200:
            raise ApiServerError(f"ssh start: {r.status_code} {r.text[:300]}")
        return r.json()["sid"]


def ssh_stdin(cluster_name: str, sid: str, data: bytes) -> None:
    with _client() as c:
        r = c.post(f"/api/v1/ssh/{cluster_name}/{sid}/stdin", content=data,
                   headers=_auth_headers())
        if r.status_code != 200:
            raise ApiServerError(f"ssh stdin: {r.status_code}")


def ssh_stdout(cluster_name: str, sid: str):
    """Yield the session's PTY output until the shell exits."""
    if _TEST_CLIENT is not None:
        with _TEST_CLIENT.stream(
                "GET", f"/api/v1/ssh/{cluster_name}/{sid}/stdout",
                headers=_auth_headers()) as r:
            yield from r.iter_bytes()
        return
    with httpx.Client(base_url=server_url(), timeout=None,
                      headers=_auth_headers()) as c:
        with c.stream("GET",
                      f"/api/v1/ssh/{cluster_name}/{sid}/stdout") as r:
            yield from r.iter_bytes()


def ssh_resize(cluster_name: str, sid: str, rows: int, cols: int) -> None:
    with _client() as c:
        c.post(f"/api/v1/ssh/{cluster_name}/{sid}/resize",
               json={"rows": rows, "cols": cols}, headers=_auth_headers())


def ssh_close(cluster_name: str, sid: str) -> None:
    with _client() as c:
        c.post(f"/api/v1/ssh/{cluster_name}/{sid}/close",
               headers=_auth_headers())


def ssh_shell(cluster_name: str, cmd=None) -> int:
    """Interactive shell on the cluster head (reference: `sky ssh`).

    Puts the local TTY in raw mode and pumps stdin/stdout through the
    server's tunnel; with a non-TTY stdin, pipes it through and exits
    when the remote shell does.  Returns the remote exit code when the
    agent reports one, else 0."""
    import select as _select
    import shutil
    import threading

    sid = ssh_start(cluster_name, cmd=cmd)
    sz = shutil.get_terminal_size()
    try:
        ssh_resize(cluster_name, sid, sz.lines, sz.columns)
    except Exception:  # noqa: BLE001 — resize is best-effort
        pass
    done = threading.Event()

    def pump_out():
        try:
            for chunk in ssh_stdout(cluster_name, sid):
                sys.stdout.buffer.write(chunk)
                sys.stdout.buffer.flush()
        finally:
            done.set()

    t = threading.Thread(target=pump_out, daemon=True)
    t.start()
    is_tty = sys.stdin.isatty()
    old_attrs = None
    if is_tty:
        import termios
        import tty
        old_attrs = termios.tcgetattr(sys.stdin.fileno())
        tty.setraw(sys.stdin.fileno())
    try:
        fd = sys.stdin.fileno()
        while not done.is_set():
            r, _, _ = _select.select([fd], [], [], 0.2)
            if not r:
                continue
            data = os.read(fd, 4096)
            if not data:  # EOF on piped stdin
                ssh_stdin(cluster_name, sid, b"exit\n")
                done.wait(timeout=30)
                break
            ssh_stdin(cluster_name, sid, data)
    finally:
        if old_attrs is not None:
            import termios
            termios.tcsetattr(sys.stdin.fileno(), termios.TCSADRAIN,
                              old_attrs)
        try:
            with _client() as c:
                st = c.get(f"/api/v1/ssh/{cluster_name}/{sid}/status",
                           headers=_auth_headers())
        except Exception:  # noqa: BLE001
            st = None
        ssh_close(cluster_name, sid)
    if st is not None and st.status_code == 200:
        return int(st.json().get("exit_code") or 0)
    return 0
