"""Background refreshers (reference: sky/server/daemons.py:159-314 —
cluster status, managed-jobs and serve status refresh loops)."""
from __future__ import annotations

import threading
import time

_started = False

REFRESH_INTERVAL = 60.0


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
    threading.Thread(target=_loop, daemon=True).start()
