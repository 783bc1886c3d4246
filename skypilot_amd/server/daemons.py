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
