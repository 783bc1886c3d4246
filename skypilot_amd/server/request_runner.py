"""LONG-request child process entrypoint: load request by id, execute.

Invoked by the executor as `python -m skypilot_amd.server.request_runner
<request_id>`; a fresh interpreter avoids forking the threaded server
(reference parity: per-request process, sky/server/requests/process.py).
"""
from __future__ import annotations

import os
import sys
import traceback


def main() -> int:
    rid = sys.argv[1]
    from skypilot_amd.server import requests_db as rdb
    req = rdb.get(rid)
    if req is None:
        return 1
    if req["status"] == rdb.CANCELLED:
        return 0  # cancelled in the claim->spawn window: never execute
    rdb.set_pid(rid, os.getpid())  # the runner is the killable pid
    with open(req["log_path"], "ab", buffering=0) as logf:
        os.dup2(logf.fileno(), 1)
        os.dup2(logf.fileno(), 2)
    if req.get("user"):
        os.environ["SKY_AMD_USER"] = req["user"]
    if req.get("workspace"):
        os.environ["SKY_AMD_WORKSPACE"] = req["workspace"]
    try:
        from skypilot_amd.server import executor
        import skypilot_amd.server.handlers  # noqa: F401 (registry)
        fn = executor.handler(req["name"])
        result = fn(**req["body"])
        rdb.finish(rid, rdb.SUCCEEDED, result=result)
        return 0
    except BaseException as e:  # noqa: BLE001
        traceback.print_exc()
        rdb.finish(rid, rdb.FAILED, error=f"{type(e).__name__}: {e}")
        return 1


if __name__ == "__main__":
    sys.exit(main())
