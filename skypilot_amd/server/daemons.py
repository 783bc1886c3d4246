"""Background refreshers (reference: sky/server/daemons.py:159-314 —
cluster status, managed-jobs and serve status refresh loops) + HA
leader election (reference: sky/utils/leader_election.py — with several
API-server replicas behind one state dir, exactly one runs the
refresh daemons) + event-loop stall detector
(reference: sky/server/loop_stall.py)."""
from __future__ import annotations

import threading
import time

_started = False
_leader_lock_file = None  # keeps the flock alive for process lifetime
_is_leader = False
_loop_stall_max_s = 0.0

REFRESH_INTERVAL = 60.0
STALL_TICK = 0.25


def try_acquire_leadership() -> bool:
    """flock-based leader election over the shared state dir.  The lock
    is held until process exit; non-leaders serve requests but skip the
    background refreshers."""
    global _leader_lock_file, _is_leader
    if _is_leader:
        return True
    import fcntl

    from skypilot_amd import global_state
    d = global_state.root_dir() / "locks"
    d.mkdir(parents=True, exist_ok=True)
    f = open(d / "daemons-leader.lock", "w")
    try:
        fcntl.flock(f, fcntl.LOCK_EX | fcntl.LOCK_NB)
    except OSError:
        f.close()
        return False
    f.write(str(__import__("os").getpid()))
    f.flush()
    _leader_lock_file = f
    _is_leader = True
    return True


def is_leader() -> bool:
    return _is_leader


def loop_stall_max_seconds() -> float:
    """Worst observed scheduling delay of the stall-detector thread —
    a proxy for GIL/event-loop starvation (exported on /metrics)."""
    return _loop_stall_max_s


def _stall_loop():
    global _loop_stall_max_s
    prev = time.monotonic()
    while True:
        time.sleep(STALL_TICK)
        now = time.monotonic()
        stall = (now - prev) - STALL_TICK
        if stall > _loop_stall_max_s:
            _loop_stall_max_s = stall
        prev = now


def _reap_orphan_agents(min_age_s: float = 120.0) -> int:
    """Stop agent daemons whose cluster has NO record (a launch request
    cancelled between agent spawn and state write leaves one behind).
    Identity-checked: the pid is only killed if its cmdline still names
    this exact cluster dir (PID-recycling guard)."""
    import json as _json
    import os
    import signal
    import time as _t
    from skypilot_amd import global_state
    reaped = 0
    cdir = global_state.root_dir() / "clusters"
    if not cdir.exists():
        return 0
    for meta in cdir.glob("*/agent.json"):
        name = meta.parent.name
        try:
            if global_state.get_cluster(name) is not None:
                continue
            if _t.time() - meta.stat().st_mtime < min_age_s:
                continue  # grace: provision may still be writing state
            pid = int(_json.loads(meta.read_text())["pid"])
            with open(f"/proc/{pid}/cmdline", "rb") as fh:
                if str(meta.parent).encode() not in fh.read():
                    continue
            os.kill(pid, signal.SIGTERM)
            reaped += 1
        except (OSError, ValueError, KeyError):
            continue
    return reaped


def _gc_controller_logs(max_age_days: float = 7.0,
                        max_total_mb: float = 512.0) -> int:
    """Controller-log retention (reference: sky/jobs/log_gc.py): drop
    jobs-controller-*.log files for TERMINAL jobs once they are old or
    the total exceeds the budget (oldest first)."""
    import os
    import time as _t
    from skypilot_amd import global_state
    from skypilot_amd.jobs import state as jobs_state
    root = global_state.root_dir()
    terminal = {j["job_id"] for j in jobs_state.list_jobs()
                if j["status"] in jobs_state.TERMINAL}
    logs = []
    for p in root.glob("jobs-controller-*.log"):
        try:
            jid = int(p.stem.rsplit("-", 1)[1])
        except ValueError:
            continue
        if jid in terminal:
            st = p.stat()
            logs.append((st.st_mtime, st.st_size, p))
    logs.sort()
    removed = 0
    now = _t.time()
    total = sum(sz for _, sz, _ in logs)
    for mtime, size, p in logs:
        if (now - mtime > max_age_days * 86400
                or total > max_total_mb * 1e6):
            try:
                p.unlink()
                removed += 1
                total -= size
            except OSError:
                pass
    return removed


def _loop():
    while True:
        time.sleep(REFRESH_INTERVAL)
        try:
            from skypilot_amd import core
            core.status(refresh=True)
        except Exception:  # noqa: BLE001
            pass
        try:
            from skypilot_amd.jobs import state as jobs_state
            jobs_state.reconcile()
            # drain PENDING controllers freed by finished/crashed ones
            from skypilot_amd.jobs import scheduler as jobs_scheduler
            jobs_scheduler.maybe_start_controllers()
            from skypilot_amd.jobs import pools as jobs_pools
            jobs_pools.autoscale()
            _gc_controller_logs()
            _reap_orphan_agents()
            from skypilot_amd.server import requests_db as _rdb
            _rdb.gc_requests()
        except Exception:  # noqa: BLE001
            pass
        try:
            from skypilot_amd.serve import serve_state
            serve_state.reconcile()
        except Exception:  # noqa: BLE001
            pass


def start():
    global _started
    if _started:
        return
    _started = True
    threading.Thread(target=_stall_loop, daemon=True).start()
    if try_acquire_leadership():
        threading.Thread(target=_loop, daemon=True).start()
