"""Node agent daemon (skylet equivalent) — HTTP control plane per cluster.

Reference: sky/skylet/skylet.py (gRPC daemon, port 46590) + services.py.
The RPC surface mirrors the skylet protos (SURVEY.md Appendix A:
AddJob/QueueJob/GetJobQueue/CancelJobs/TailLogs/GetJobStatus +
SetAutostop/IsAutostopping) as JSON-over-HTTP on a per-cluster local
port; SSH pools tunnel the same port.  A scheduler thread runs the
GPU-aware FIFO queue (reference: job_lib.py:278 JobScheduler) and an
autostop event loop mirrors skylet/events.py:268 StopEvent.
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import threading
import time
from pathlib import Path

import uvicorn
from fastapi import FastAPI, Request
from fastapi.responses import StreamingResponse

from skypilot_amd.agent import job_lib

SCHEDULER_INTERVAL = 1.0
AUTOSTOP_INTERVAL = 30.0


def create_app(cluster_dir: str, gpu_ids: list[int],
               token: str | None = None) -> FastAPI:
    app = FastAPI()

    if token:
        # Per-cluster shared-secret auth (reference: skylet gRPC is only
        # reachable through an SSH tunnel — cloud_vm_ray_backend.py:2414;
        # an HTTP agent on a pod network needs its own gate).
        from fastapi.responses import JSONResponse

        @app.middleware("http")
        async def _auth(request: Request, call_next):
            if request.url.path != "/health":
                auth = request.headers.get("authorization", "")
                if auth != f"Bearer {token}":
                    return JSONResponse({"detail": "agent token required"},
                                        status_code=401)
            return await call_next(request)
    table = job_lib.JobTable(cluster_dir)
    state = {"autostop_idle_minutes": -1, "autostop_down": False,
             "last_active": time.time(), "autostopping": False,
             "gpu_ids": gpu_ids}

    # ---- scheduler thread (FIFO + GPU accounting) -------------------------
    driver_procs: list = []

    def scheduler_loop():
        while True:
            try:
                # Reap finished/killed driver processes so PID liveness
                # checks don't see zombies (a SIGKILLed driver must be
                # detected as dead by reconcile()).
                for pr in driver_procs[:]:
                    if pr.poll() is not None:
                        driver_procs.remove(pr)
                table.reconcile()
                free = [g for g in state["gpu_ids"]
                        if g not in table.allocated_gpus()]
                for job in table.pending_jobs():
                    spec = job["spec"]
                    need = int(spec.get("num_nodes", 1)) * \
                        int(spec.get("gpus_per_node", 0))
                    if need > len(free):
                        break  # strict FIFO (reference FIFOScheduler)
                    gpu_ids_alloc = free[:need]
                    free = free[need:]
                    spec["gpu_ids"] = gpu_ids_alloc
                    with table._conn() as c:
                        c.execute("UPDATE jobs SET spec=?, status=? "
                                  "WHERE job_id=? AND status='PENDING'",
                                  (json.dumps(spec), job_lib.INIT,
                                   job["job_id"]))
                    dlog = open(Path(cluster_dir) / "driver.log", "ab")
                    driver_procs.append(subprocess.Popen(
                        [sys.executable, "-m", "skypilot_amd.agent.driver",
                         cluster_dir, str(job["job_id"])],
                        stdout=dlog, stderr=subprocess.STDOUT,
                        start_new_session=True))
                    dlog.close()
                    state["last_active"] = time.time()
            except Exception as e:  # noqa: BLE001
                print(f"scheduler error: {e}", file=sys.stderr)
            time.sleep(SCHEDULER_INTERVAL)

    threading.Thread(target=scheduler_loop, daemon=True).start()

    # ---- autostop loop (reference: skylet/events.py StopEvent:268) --------
    def autostop_loop():
        while True:
            time.sleep(AUTOSTOP_INTERVAL)
            try:
                idle_min = state["autostop_idle_minutes"]
                if idle_min < 0:
                    continue
                if not table.is_idle():
                    state["last_active"] = time.time()
                    continue
                idle_for = (time.time() - state["last_active"]) / 60.0
                if idle_for >= idle_min:
                    state["autostopping"] = True
                    flag = Path(cluster_dir) / "autostop_triggered"
                    flag.write_text(
                        json.dumps({"down": state["autostop_down"],
                                    "ts": time.time()}))
            except Exception as e:  # noqa: BLE001
                print(f"autostop error: {e}", file=sys.stderr)

    threading.Thread(target=autostop_loop, daemon=True).start()

    # ---- RPC surface ------------------------------------------------------
    @app.get("/health")
    def health():
        return {"ok": True, "cluster_dir": cluster_dir,
                "gpu_ids": state["gpu_ids"], "pid": os.getpid()}

    @app.post("/jobs/queue")
    def queue_job(body: dict):
        job_id = table.add_job(body.get("name"), body["spec"])
        state["last_active"] = time.time()
        return {"job_id": job_id}

    @app.get("/jobs")
    def get_job_queue():
        return {"jobs": table.list()}

    @app.get("/jobs/{job_id}")
    def get_job(job_id: int):
        j = table.get(job_id)
        return {"job": j}

    @app.post("/jobs/{job_id}/cancel")
    def cancel(job_id: int):
        return {"cancelled": table.cancel(job_id)}

    @app.post("/jobs/cancel_all")
    def cancel_all():
        n = 0
        for j in table.list():
            if j["status"] in job_lib.NONTERMINAL:
                table.cancel(j["job_id"])
                n += 1
        return {"cancelled": n}

    @app.get("/jobs/{job_id}/logs")
    def tail_logs(job_id: int, follow: bool = True, tail: int = 0):
        j = table.get(job_id)
        if j is None:
            return StreamingResponse(iter(()), media_type="text/plain")

        def stream():
            log_dir = None
            # wait for driver to create the log dir
            for _ in range(300):
                jj = table.get(job_id)
                log_dir = jj.get("log_dir")
                if log_dir or jj["status"] in job_lib.TERMINAL:
                    break
                time.sleep(0.2)
            if not log_dir:
                yield b"(no logs)\n"
                return
            def log_files():
                d = Path(log_dir)
                files = [d / "setup.log", d / "run.log"]
                files += sorted(d.glob("*-node.log"))
                return [f for f in files if f.exists()]

            for _ in range(300):
                if log_files():
                    break
                jj = table.get(job_id)
                if jj["status"] in job_lib.TERMINAL:
                    break
                time.sleep(0.2)
            for lf in log_files():
                with open(lf, "rb") as f:
                    while True:
                        chunk = f.read(65536)
                        if chunk:
                            yield chunk
                            continue
                        jj = table.get(job_id)
                        if not follow or jj["status"] in job_lib.TERMINAL:
                            break
                        time.sleep(0.3)

        return StreamingResponse(stream(), media_type="text/plain")

    @app.post("/autostop")
    def set_autostop(body: dict):
        state["autostop_idle_minutes"] = int(body.get("idle_minutes", -1))
        state["autostop_down"] = bool(body.get("down", False))
        state["last_active"] = time.time()
        return {"ok": True}

    @app.get("/autostop")
    def is_autostopping():
        return {"autostopping": state["autostopping"],
                "idle_minutes": state["autostop_idle_minutes"],
                "down": state["autostop_down"]}

    @app.get("/idle")
    def idle():
        return {"idle": table.is_idle()}

    # ---- interactive exec sessions (reference: the websocket SSH proxy
    # in sky/server/server.py + `sky ssh`).  No websocket stack ships in
    # this offline image (uvicorn has neither `websockets` nor `wsproto`),
    # so the same duplex contract is carried over plain HTTP/1.1: stdin
    # arrives as raw POST bodies, stdout is one long chunked-streaming
    # GET, resize/close are control POSTs.  The shell runs on a real PTY
    # in the cluster workdir, under the agent token gate above. ----------
    import fcntl
    import pty
    import secrets
    import select
    import signal
    import struct
    import termios

    exec_sessions: dict = {}

    def _reap_session(sid: str):
        s = exec_sessions.pop(sid, None)
        if not s:
            return
        try:
            os.kill(s["pid"], signal.SIGKILL)
        except ProcessLookupError:
            pass
        try:
            os.waitpid(s["pid"], os.WNOHANG)
        except ChildProcessError:
            pass
        for k in ("fd", "wfd"):
            try:
                if s.get(k) is not None and (k == "fd" or s[k] != s["fd"]):
                    os.close(s[k])
            except OSError:
                pass

    @app.post("/exec/start")
    def exec_start(body: dict = None):
        body = body or {}
        # cap live sessions (abandoned PTYs hold fds until /close or
        # cluster teardown); reap any whose shell already exited first
        for sid_ in list(exec_sessions):
            se = exec_sessions[sid_]
            try:
                os.kill(se["pid"], 0)
            except ProcessLookupError:
                _reap_session(sid_)
        if len(exec_sessions) >= 32:
            from fastapi.responses import JSONResponse
            return JSONResponse({"detail": "too many exec sessions"},
                                status_code=429)
        cmd = body.get("cmd") or ["/bin/bash", "-i"]
        if isinstance(cmd, str):
            cmd = ["/bin/bash", "-lc", cmd]
        cwd = Path(cluster_dir) / "workdir"
        cwd.mkdir(parents=True, exist_ok=True)
        env = dict(os.environ)
        env["TERM"] = body.get("term", "xterm-256color")
        env.update({str(k): str(v) for k, v in (body.get("env") or {}).items()})
        try:
            pid, fd = pty.fork()
            if pid == 0:  # child: become the shell on the PTY slave
                try:
                    os.chdir(cwd)
                    os.execvpe(cmd[0], cmd, env)
                finally:
                    os._exit(127)
            sess = {"fd": fd, "wfd": fd, "pid": pid, "exit_code": None,
                    "tty": True}
        except OSError:
            # No PTY devices (restricted container without /dev/pts):
            # fall back to a pipe-backed session — same wire contract,
            # no terminal semantics (resize is a no-op, no input echo).
            proc = subprocess.Popen(
                cmd, cwd=str(cwd), env=env, stdin=subprocess.PIPE,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                bufsize=0, start_new_session=True)
            sess = {"fd": proc.stdout.fileno(),
                    "wfd": proc.stdin.fileno(), "pid": proc.pid,
                    "exit_code": None, "tty": False, "proc": proc}
        sid = secrets.token_hex(8)
        exec_sessions[sid] = sess
        state["last_active"] = time.time()
        return {"sid": sid, "pid": sess["pid"], "tty": sess["tty"]}

    @app.post("/exec/{sid}/stdin")
    async def exec_stdin(sid: str, request: Request):
        s = exec_sessions.get(sid)
        if s is None:
            return {"ok": False, "error": "no such session"}
        data = await request.body()
        try:
            os.write(s["wfd"], data)
        except OSError:
            return {"ok": False, "error": "session closed"}
        state["last_active"] = time.time()
        return {"ok": True, "n": len(data)}

    @app.get("/exec/{sid}/stdout")
    def exec_stdout(sid: str):
        s = exec_sessions.get(sid)
        if s is None:
            return StreamingResponse(iter(()),
                                     media_type="application/octet-stream")

        def gen():
            fd = s["fd"]
            try:
                while sid in exec_sessions:
                    r, _, _ = select.select([fd], [], [], 0.25)
                    if not r:
                        continue
                    try:
                        chunk = os.read(fd, 65536)
                    except OSError:  # EIO: shell exited, PTY drained
                        break
                    if not chunk:
                        break
                    yield chunk
            finally:
                try:
                    done, st = os.waitpid(s["pid"], os.WNOHANG)
                    if done:
                        s["exit_code"] = os.waitstatus_to_exitcode(st)
                except ChildProcessError:
                    pass

        return StreamingResponse(gen(),
                                 media_type="application/octet-stream")

    @app.get("/exec/{sid}")
    def exec_status(sid: str):
        s = exec_sessions.get(sid)
        if s is None:
            return {"alive": False}
        try:
            os.kill(s["pid"], 0)
            alive = True
        except ProcessLookupError:
            alive = False
        return {"alive": alive, "exit_code": s["exit_code"]}

    @app.post("/exec/{sid}/resize")
    def exec_resize(sid: str, body: dict):
        s = exec_sessions.get(sid)
        if s is None or not s.get("tty"):
            return {"ok": False}
        winsz = struct.pack("HHHH", int(body.get("rows", 24)),
                            int(body.get("cols", 80)), 0, 0)
        fcntl.ioctl(s["fd"], termios.TIOCSWINSZ, winsz)
        return {"ok": True}

    @app.post("/exec/{sid}/close")
    def exec_close(sid: str):
        _reap_session(sid)
        return {"ok": True}

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cluster-dir", required=True)
    ap.add_argument("--port", type=int, required=True)
    ap.add_argument("--gpu-ids", default="")
    ap.add_argument("--host", default="127.0.0.1",
                    help="bind address (k8s gang pods bind 0.0.0.0 so "
                         "peer agents are reachable at podIP)")
    args = ap.parse_args()
    gpu_ids = [int(g) for g in args.gpu_ids.split(",") if g != ""]
    # Shared secret via env (not argv: visible in ps) or cluster file.
    token = os.environ.get("SKY_AMD_AGENT_TOKEN")
    if not token:
        tf = Path(args.cluster_dir) / "agent_token"
        token = tf.read_text().strip() if tf.exists() else None
    app = create_app(args.cluster_dir, gpu_ids, token=token)
    # Record readiness for the provisioner.
    Path(args.cluster_dir).mkdir(parents=True, exist_ok=True)
    (Path(args.cluster_dir) / "agent.json").write_text(
        json.dumps({"port": args.port, "pid": os.getpid()}))
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
