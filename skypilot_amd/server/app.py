"""API server — FastAPI app + async request store.

Reference: sky/server/server.py (app :1176, routes :2034-2555, request
store /api/get /api/stream /api/cancel :2597-2911).  Every mutating
route schedules a request on the executor and returns a request_id; the
client polls /api/get or streams /api/stream.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import time
from pathlib import Path
from typing import Any, Dict, Optional

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import StreamingResponse

from skypilot_amd import global_state, users
from skypilot_amd.server import executor
from skypilot_amd.server import handlers  # noqa: F401  (registers handlers)
from skypilot_amd.server import requests_db as rdb

DEFAULT_PORT = 46580
API_PREFIX = "/api/v1"
# Bump API_VERSION on wire-format changes; raise MIN_CLIENT_API_VERSION
# only when old clients can no longer be served.
API_VERSION = 2
MIN_CLIENT_API_VERSION = 1


def create_app(start_workers: bool = True) -> FastAPI:
    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def _lifespan(app_):
        # the server's own OS identity bootstraps as admin
        users.ensure_user(global_state.current_user())
        if start_workers:
            executor.start_workers()
        # serve/jobs background refreshers (reference: server/daemons.py)
        from skypilot_amd.server import daemons
        daemons.start()
        yield

    app = FastAPI(title="skypilot-amd API server", lifespan=_lifespan)

    @app.get("/health")
    def health():
        # api_version gates client compatibility (reference:
        # sky/server/constants.py API_VERSION + version-mismatch errors)
        return {"ok": True, "version": "0.1.0", "api_version": API_VERSION,
                "min_client_api_version": MIN_CLIENT_API_VERSION,
                "pid": os.getpid()}

    # ---- identity & RBAC (reference: sky/users/rbac.py + server auth) ----
    # Auth modes (reference: sky server only trusts identity headers set
    # by an authenticating proxy — X-Auth-Request-Email behind
    # oauth2-proxy; otherwise identity comes from tokens):
    #   local — single-user/localhost deployment: the client-supplied
    #           X-Skypilot-User header is trusted (anyone who can reach
    #           the loopback socket IS the operator).  Default for
    #           loopback binds.
    #   token — non-local deployment: identity REQUIRES a bearer
    #           service-account token, except when an authenticating
    #           proxy is declared trusted (SKY_AMD_TRUST_PROXY_AUTH=1),
    #           in which case X-Auth-Request-Email is accepted.
    def _auth_mode() -> str:
        return os.environ.get("SKY_AMD_AUTH_MODE", "local")

    def _identity(request: Request) -> Dict[str, str]:
        """Resolve (user, role) per the active auth mode."""
        auth = request.headers.get("authorization", "")
        if auth.lower().startswith("bearer "):
            sa = users.resolve_token(auth[7:].strip())
            if sa is None:
                raise HTTPException(401, "invalid service-account token")
            return {"user": sa["name"], "role": sa["role"]}
        if _auth_mode() == "token":
            if os.environ.get("SKY_AMD_TRUST_PROXY_AUTH") == "1":
                email = request.headers.get("x-auth-request-email")
                if email:
                    return {"user": email, "role": users.ensure_user(email)}
            raise HTTPException(
                401, "authentication required: pass a service-account "
                     "bearer token (server runs in token auth mode)")
        name = request.headers.get("x-skypilot-user")
        if name:
            return {"user": name, "role": users.ensure_user(name)}
        me = global_state.current_user()
        return {"user": me, "role": users.ensure_user(me)}

    def _may_access_request(ident: Dict[str, str], req) -> bool:
        """Request rows are visible/cancellable by their owner or admin."""
        if ident["role"] == "admin":
            return True
        return req.get("user") in (None, ident["user"])

    def _require_admin(request: Request) -> Dict[str, str]:
        ident = _identity(request)
        if ident["role"] != "admin":
            raise HTTPException(
                403, f"admin role required (you are {ident['role']})")
        return ident

    @app.get("/api/users")
    def api_users(request: Request):
        _identity(request)
        return users.list_users()

    @app.post("/api/users/role")
    def api_users_role(request: Request, body: Dict[str, Any]):
        _require_admin(request)
        try:
            users.set_role(body["name"], body["role"])
        except ValueError as e:
            raise HTTPException(400, str(e))
        return {"ok": True}

    @app.post("/api/users/token")
    def api_users_token(request: Request, body: Dict[str, Any]):
        ident = _require_admin(request)
        try:
            token = users.create_token(body["name"], ident["user"],
                                       body.get("role", "user"))
        except ValueError as e:
            raise HTTPException(400, str(e))
        return {"token": token}

    @app.get("/api/users/tokens")
    def api_users_tokens(request: Request):
        _require_admin(request)
        return users.list_tokens()

    @app.post("/api/users/token/revoke")
    def api_users_token_revoke(request: Request, body: Dict[str, Any]):
        _require_admin(request)
        return {"revoked": users.revoke_token(body["name"])}

    # ---- client file upload (reference: sky server /upload,/upload_v2
    # server.py:1826,1885; chunked client in client/common.py:154-192).
    @app.post("/api/upload")
    async def api_upload(request: Request, upload_id: str,
                         chunk_index: int = 0, total_chunks: int = 1):
        import re as _re
        import tarfile
        _identity(request)
        if not _re.fullmatch(r"[a-f0-9]{32}", upload_id):
            raise HTTPException(400, "bad upload_id")
        if not (0 <= chunk_index < total_chunks <= 4096):
            raise HTTPException(400, "bad chunk bounds")
        body = await request.body()
        max_gb = float(os.environ.get("SKY_AMD_MAX_UPLOAD_GB", "10"))
        if len(body) > 64 << 20 or total_chunks * len(body) > max_gb * 1e9:
            raise HTTPException(413, "upload too large")
        updir = global_state.root_dir() / "api" / "uploads"
        updir.mkdir(parents=True, exist_ok=True)
        part = updir / f".{upload_id}.part{chunk_index}"
        part.write_bytes(body)
        parts = [updir / f".{upload_id}.part{i}"
                 for i in range(total_chunks)]
        if not all(p.exists() for p in parts):
            return {"status": "uploading", "chunk_index": chunk_index}
        tarball = updir / f".{upload_id}.tar.gz"
        with open(tarball, "wb") as out:
            for p in parts:
                out.write(p.read_bytes())
                p.unlink()
        dest = updir / upload_id
        dest.mkdir(parents=True, exist_ok=True)
        with tarfile.open(tarball, "r:gz") as tf:
            for m in tf.getmembers():  # path-traversal guard
                if m.name.startswith("/") or ".." in m.name.split("/"):
                    raise HTTPException(400, f"unsafe path {m.name!r}")
            tf.extractall(dest)
        tarball.unlink()
        return {"status": "completed", "upload_id": upload_id}

    # ---- generic async request plumbing -----------------------------------
    @app.post(API_PREFIX + "/{name}")
    def submit(name: str, request: Request, body: Dict[str, Any] = None):
        ident = _identity(request)
        ws = request.headers.get("x-skypilot-workspace")
        try:
            users.authorize(ident["role"], name)
            users.authorize_workspace(ident["user"], ident["role"], ws)
        except Exception as e:
            raise HTTPException(403, str(e))
        try:
            rid = executor.schedule(
                name, body or {}, user=ident["user"], workspace=ws)
        except KeyError:
            raise HTTPException(404, f"unknown request type {name!r}")
        return {"request_id": rid}

    @app.get("/api/get")
    def api_get(request_id: str, request: Request = None):
        ident = _identity(request)
        req = rdb.get(request_id)
        if req is None:
            raise HTTPException(404, "no such request")
        if not _may_access_request(ident, req):
            raise HTTPException(
                403, f"request {request_id} belongs to {req.get('user')!r}")
        return {
            "request_id": request_id,
            "name": req["name"],
            "status": req["status"],
            "result": req["result"],
            "error": req["error"],
        }

    @app.get("/api/stream")
    async def api_stream(request_id: str, request: Request = None):
        ident = _identity(request)
        req = rdb.get(request_id)
        if req is None:
            raise HTTPException(404, "no such request")
        if not _may_access_request(ident, req):
            raise HTTPException(
                403, f"request {request_id} belongs to {req.get('user')!r}")

        async def gen():
            pos = 0
            path = Path(req["log_path"])
            while True:
                if path.exists():
                    with open(path, "rb") as f:
                        f.seek(pos)
                        chunk = f.read()
                    if chunk:
                        pos += len(chunk)
                        yield chunk
                r = rdb.get(request_id)
                if r["status"] in rdb.TERMINAL:
                    if path.exists():
                        with open(path, "rb") as f:
                            f.seek(pos)
                            tailc = f.read()
                        if tailc:
                            yield tailc
                    break
                await asyncio.sleep(0.2)

        return StreamingResponse(gen(), media_type="text/plain")

    @app.post("/api/cancel")
    def api_cancel(body: Dict[str, Any], request: Request = None):
        ident = _identity(request)
        rid = body["request_id"]
        req = rdb.get(rid)
        if req is not None and not _may_access_request(ident, req):
            raise HTTPException(
                403, f"request {rid} belongs to {req.get('user')!r}")
        return {"cancelled": executor.cancel_request(rid)}

    # dashboard pages carry the same identity gate as the API: in token
    # auth mode an anonymous browser must not see cluster metadata/logs
    @app.get("/dashboard")
    def dashboard(request: Request):
        _identity(request)
        from fastapi.responses import HTMLResponse
        from skypilot_amd.server.dashboard import render
        return HTMLResponse(render())

    @app.get("/dashboard/cluster/{cluster_name}")
    def dashboard_cluster(cluster_name: str, request: Request):
        _identity(request)
        from fastapi.responses import HTMLResponse
        from skypilot_amd.server.dashboard import render_cluster
        return HTMLResponse(render_cluster(cluster_name))

    @app.get("/dashboard/cluster/{cluster_name}/job/{job_id}")
    def dashboard_job(cluster_name: str, job_id: int, request: Request):
        _identity(request)
        from fastapi.responses import HTMLResponse
        from skypilot_amd.server.dashboard import render_job_logs
        return HTMLResponse(render_job_logs(cluster_name, job_id))

    @app.get("/metrics")
    def metrics():
        # Prometheus metrics (reference: sky/server/metrics.py).
        from fastapi.responses import PlainTextResponse
        from skypilot_amd.server import requests_db as _rdb
        lines = []
        by_status: Dict[str, int] = {}
        for r in _rdb.list_requests(1000):
            by_status[r["status"]] = by_status.get(r["status"], 0) + 1
        for st_, n in by_status.items():
            lines.append(
                f'sky_amd_requests_total{{status="{st_.lower()}"}} {n}')
        from skypilot_amd import global_state as _gs
        ups = sum(1 for c in _gs.list_clusters() if c["status"] == "UP")
        lines.append(f"sky_amd_clusters_up {ups}")
        # executor free-slot gauges (reference: executor.py:369-404)
        from skypilot_amd.server import daemons as _d
        from skypilot_amd.server import executor as _ex
        for q, free in _ex.free_slots().items():
            lines.append(
                f'sky_amd_executor_free_slots{{queue="{q.lower()}"}} '
                f"{free}")
        lines.append(
            f"sky_amd_loop_stall_max_seconds "
            f"{_d.loop_stall_max_seconds():.4f}")
        lines.append(
            f"sky_amd_daemons_leader {1 if _d.is_leader() else 0}")
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/api/requests")
    def api_requests(limit: int = 100, request: Request = None):
        ident = _identity(request)
        reqs = [r for r in rdb.list_requests(limit)
                if _may_access_request(ident, r)]
        return [{k: r[k] for k in
                 ("request_id", "name", "status", "created_at",
                  "finished_at", "error")} for r in reqs]

    # ---- interactive shell tunnel (reference: the websocket SSH proxy
    # in sky/server/server.py + `sky ssh`; carried over HTTP streaming
    # here — see agent/daemon.py exec sessions).  The server authenticates
    # the caller, checks cluster ownership, then bridges to the cluster
    # agent with the per-cluster bearer token the client never sees. ------
    def _cluster_agent(cluster_name: str, ident: Dict[str, str]):
        rec = global_state.get_cluster(cluster_name)
        if rec is None:
            raise HTTPException(404, f"no cluster {cluster_name!r}")
        owner = rec.get("user")
        if ident["role"] != "admin" and owner not in (None, ident["user"]):
            raise HTTPException(
                403, f"cluster {cluster_name!r} belongs to {owner!r}")
        handle = rec.get("handle") or {}
        port = handle.get("agent_port")
        if not port:
            raise HTTPException(409, f"cluster {cluster_name!r} has no "
                                     "agent (not UP?)")
        from skypilot_amd.agent.client import AgentClient
        return AgentClient(port, host=handle.get("agent_host", "127.0.0.1"),
                           token=handle.get("agent_token"))

    @app.post(API_PREFIX + "/ssh/{cluster_name}/start")
    def ssh_start(cluster_name: str, request: Request,
                  body: Dict[str, Any] = None):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)
        try:
            body = body or {}
            sid = agent.exec_start(cmd=body.get("cmd"),
                                   env=body.get("env"),
                                   term=body.get("term", "xterm-256color"))
        finally:
            agent.close()
        return {"sid": sid}

    @app.post(API_PREFIX + "/ssh/{cluster_name}/{sid}/stdin")
    async def ssh_stdin(cluster_name: str, sid: str, request: Request):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)
        try:
            data = await request.body()
            ok = agent.exec_stdin(sid, data)
        finally:
            agent.close()
        return {"ok": ok}

    @app.get(API_PREFIX + "/ssh/{cluster_name}/{sid}/stdout")
    def ssh_stdout(cluster_name: str, sid: str, request: Request = None):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)

        def gen():
            try:
                yield from agent.exec_stdout(sid)
            finally:
                agent.close()

        return StreamingResponse(gen(),
                                 media_type="application/octet-stream")

    @app.get(API_PREFIX + "/ssh/{cluster_name}/{sid}/status")
    def ssh_status(cluster_name: str, sid: str, request: Request = None):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)
        try:
            return agent.exec_status(sid)
        finally:
            agent.close()

    @app.post(API_PREFIX + "/ssh/{cluster_name}/{sid}/resize")
    def ssh_resize(cluster_name: str, sid: str, request: Request,
                   body: Dict[str, Any] = None):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)
        try:
            body = body or {}
            agent.exec_resize(sid, int(body.get("rows", 24)),
                              int(body.get("cols", 80)))
        finally:
            agent.close()
        return {"ok": True}

    @app.post(API_PREFIX + "/ssh/{cluster_name}/{sid}/close")
    def ssh_close(cluster_name: str, sid: str, request: Request):
        ident = _identity(request)
        agent = _cluster_agent(cluster_name, ident)
        try:
            agent.exec_close(sid)
        finally:
            agent.close()
        return {"ok": True}

    # ---- log streaming for cluster jobs (proxied to the node agent) -------
    @app.get(API_PREFIX + "-logs/{cluster_name}")
    def cluster_logs(cluster_name: str, job_id: Optional[int] = None,
                     follow: bool = True):
        from skypilot_amd import core
        gen = core.tail_logs(cluster_name, job_id, follow)
        return StreamingResponse(gen, media_type="text/plain")

    return app


def server_url() -> str:
    port = int(os.environ.get("SKY_AMD_API_PORT", DEFAULT_PORT))
    return os.environ.get("SKY_AMD_API_SERVER", f"http://127.0.0.1:{port}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("SKY_AMD_API_PORT",
                                               DEFAULT_PORT)))
    args = ap.parse_args()
    if (args.host not in ("127.0.0.1", "localhost", "::1")
            and "SKY_AMD_AUTH_MODE" not in os.environ):
        # Non-local bind: never trust client-supplied identity headers
        # by default (header spoofing → admin escalation).
        os.environ["SKY_AMD_AUTH_MODE"] = "token"
        print(f"[server] binding {args.host}: auth mode set to 'token' "
              "(set SKY_AMD_AUTH_MODE=local to trust identity headers, "
              "e.g. behind an authenticating proxy)")
    import uvicorn
    (global_state.root_dir() / "api").mkdir(parents=True, exist_ok=True)
    (global_state.root_dir() / "api" / "server.json").write_text(
        json.dumps({"port": args.port, "pid": os.getpid(),
                    "started_at": time.time()}))
    uvicorn.run(create_app(), host=args.host, port=args.port,
                log_level="warning")


if __name__ == "__main__":
    main()
