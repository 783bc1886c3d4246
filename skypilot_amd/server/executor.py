"""Request executor — LONG/SHORT queues with per-request isolation.

Reference: sky/server/requests/executor.py (LONG vs SHORT worker pools
:1-20, per-request process for cancellability :302-420,
_request_execution_wrapper :784).

LONG requests (launch/down/jobs/serve — heavy, must be cancellable) run
in a *spawned* process whose pid is recorded so /api/cancel can SIGTERM
it.  SHORT requests (status-like, sub-second) run inline on worker
threads: forking a threaded server process is deadlock-prone (inherited
lock state), and spawn overhead would dominate these calls — same
trade-off the reference makes with its reusable PoolExecutor for SHORT.
stdout/stderr of LONG requests tee to the request log file streamed by
/api/stream.
"""
from __future__ import annotations

import contextlib
import io
import os
import signal
import sys
import threading
import time
import traceback
from typing import Any, Callable, Dict

from skypilot_amd.server import requests_db as rdb

LONG = "long"
SHORT = "short"

_REGISTRY: Dict[str, tuple] = {}


def register(name: str, queue: str):
    def deco(fn):
        _REGISTRY[name] = (fn, queue)
        return fn
    return deco


def queue_of(name: str) -> str:
    return _REGISTRY[name][1]


def handler(name: str) -> Callable:
    return _REGISTRY[name][0]


def _run_request_inline(req: Dict[str, Any]) -> None:
    rid = req["request_id"]
    buf = io.StringIO()
    from skypilot_amd import global_state
    global_state.set_request_user(req.get("user"))  # thread-scoped RBAC id
    global_state.set_request_workspace(req.get("workspace"))
    try:
        fn = handler(req["name"])
        with contextlib.redirect_stdout(buf), contextlib.redirect_stderr(buf):
            result = fn(**req["body"])
        rdb.finish(rid, rdb.SUCCEEDED, result=result)
    except BaseException as e:  # noqa: BLE001
        buf.write(traceback.format_exc())
        rdb.finish(rid, rdb.FAILED, error=f"{type(e).__name__}: {e}")
    finally:
        global_state.set_request_user(None)
        global_state.set_request_workspace(None)
        out = buf.getvalue()
        if out:
            try:
                with open(req["log_path"], "a") as f:
                    f.write(out)
            except OSError:
                pass


class RequestWorker(threading.Thread):
    """One worker per queue (reference: RequestWorker, executor.py:302)."""

    def __init__(self, queue: str, parallelism: int = 4):
        super().__init__(daemon=True)
        self.queue = queue
        self.parallelism = parallelism
        self._stop_evt = threading.Event()
        self._children: list = []
        self._inline: list = []

    def run(self):
        import subprocess
        while not self._stop_evt.is_set():
            try:
                self._tick(subprocess)
            except Exception:  # noqa: BLE001 — a worker must never die
                traceback.print_exc()
                time.sleep(0.5)

    def _tick(self, subprocess):
        if True:
            alive = []
            for p, rid in self._children:
                if p.poll() is None:
                    alive.append((p, rid))
                else:
                    req = rdb.get(rid)
                    if req and req["status"] == rdb.RUNNING:
                        rdb.finish(rid, rdb.FAILED,
                                   error=f"request process died "
                                         f"(exit {p.returncode})")
            self._children = alive
            self._inline = [t for t in self._inline if t.is_alive()]
            busy = len(self._children) + len(self._inline)
            if busy >= self.parallelism:
                time.sleep(0.05)
                return
            req = rdb.claim_next(self.queue, None)
            if req is None:
                time.sleep(0.05)
                return
            if self.queue == SHORT:
                t = threading.Thread(target=_run_request_inline, args=(req,),
                                     daemon=True)
                t.start()
                self._inline.append(t)
            else:
                pkg_root = os.path.dirname(os.path.dirname(
                    os.path.dirname(os.path.abspath(__file__))))
                env = dict(os.environ)
                env["PYTHONPATH"] = pkg_root + (
                    ":" + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
                p = subprocess.Popen(
                    [sys.executable, "-m",
                     "skypilot_amd.server.request_runner",
                     req["request_id"]],
                    stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
                    start_new_session=True, env=env)
                rdb.set_pid(req["request_id"], p.pid)
                self._children.append((p, req["request_id"]))

    def stop(self):
        self._stop_evt.set()


_workers: list = []


def free_slots() -> Dict[str, int]:
    """Per-queue free worker slots (Prometheus gauge; reference:
    executor.py:369-404 free-slot accounting)."""
    out: Dict[str, int] = {}
    for w in _workers:
        busy = len(w._children) + len(w._inline)
        out[w.queue] = max(0, w.parallelism - busy)
    return out


def start_workers(long_parallelism: int = None,
                  short_parallelism: int = None):
    """reference: executor.py:1317 (start) + server/config.py:89
    (compute_server_config): LONG workers scale with CPU (x2, capped at
    4 for local deployments), SHORT workers keep ample idle slots."""
    global _workers
    if _workers:
        return
    ncpu = os.cpu_count() or 2
    if long_parallelism is None:
        long_parallelism = max(1, min(4, ncpu * 2))
    if short_parallelism is None:
        short_parallelism = max(8, min(32, ncpu * 4))
    for q, par in ((LONG, long_parallelism), (SHORT, short_parallelism)):
        w = RequestWorker(q, par)
        w.start()
        _workers.append(w)


def stop_workers():
    global _workers
    for w in _workers:
        w.stop()
    _workers = []


def schedule(name: str, body: Dict[str, Any],
             user: "str | None" = None,
             workspace: "str | None" = None) -> str:
    if name not in _REGISTRY:
        raise KeyError(f"unknown request {name!r}")
    return rdb.create(name, body, queue_of(name), user=user,
                      workspace=workspace)


def cancel_request(rid: str) -> bool:
    marked, pid = rdb.mark_cancelled(rid)
    # Only a DEDICATED request-runner pid may be signalled.  Inline
    # (SHORT) requests run in a thread of this very process and LONG
    # requests have a claim->spawn window before set_pid records the
    # runner: killing the recorded pid blindly SIGTERMed the whole
    # server (observed as intermittent full-test-suite terminations).
    if marked and pid and pid != os.getpid():
        try:
            os.kill(pid, signal.SIGTERM)
        except (ProcessLookupError, PermissionError):
            pass
    return marked
