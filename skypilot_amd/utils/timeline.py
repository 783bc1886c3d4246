"""Chrome trace-event recorder (reference: sky/utils/timeline.py).

Enabled by SKY_AMD_TIMELINE_FILE (reference env:
SKYPILOT_TIMELINE_FILE_PATH, timeline.py:19); decorate hot functions
with @timeline.event("name"); view in chrome://tracing / Perfetto.
"""
from __future__ import annotations

import atexit
import functools
import json
import os
import threading
import time
from typing import List, Optional

_events: List[dict] = []
_lock = threading.Lock()
_path: Optional[str] = os.environ.get("SKY_AMD_TIMELINE_FILE")


def enabled() -> bool:
    return _path is not None


def record(name: str, ph: str, args=None) -> None:
    if _path is None:
        return
    with _lock:
        _events.append({
            "name": name, "ph": ph, "ts": time.time() * 1e6,
            "pid": os.getpid(), "tid": threading.get_ident() % 100000,
            "args": args or {},
        })


class Event:
    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        record(self.name, "B")
        return self

    def __exit__(self, *a):
        record(self.name, "E")


def event(name_or_fn):
    """Decorator or context-manager factory."""
    if callable(name_or_fn):
        fn = name_or_fn
        name = fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            with Event(name):
                return fn(*args, **kwargs)
        return wrapper

    name = name_or_fn

    def deco(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            with Event(name):
                return fn(*args, **kwargs)
        return wrapper
    return deco


@atexit.register
def _flush():
    if _path is None or not _events:
        return
    try:
        with open(_path, "w") as f:
            json.dump({"traceEvents": _events}, f)
    except OSError:
        pass
