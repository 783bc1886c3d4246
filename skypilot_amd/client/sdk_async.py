"""Async client SDK — awaitable mirror of the sync surface.

Reference: sky/client/sdk_async.py (async variants of every sdk call).
Submission endpoints are generated from the sync module's registry: any
sync function that posts a named request gets an awaitable twin here,
plus async get/stream_and_get for result retrieval.  Built on
httpx.AsyncClient; the in-process TestClient path used by the test
harness is served by wrapping the sync SDK in a thread (TestClient is
sync-only).
"""
from __future__ import annotations

import asyncio
from typing import Any, Dict, Optional

import httpx

from skypilot_amd.client import sdk as _sdk
from skypilot_amd.exceptions import ApiServerError, SkyAmdError
from skypilot_amd.server.app import server_url


def _async_client() -> httpx.AsyncClient:
    return httpx.AsyncClient(base_url=server_url(), timeout=30.0,
                             headers=_sdk._auth_headers())


async def _submit(name: str, body: Dict[str, Any]) -> str:
    if _sdk._TEST_CLIENT is not None:  # harness: run sync in a thread
        return await asyncio.to_thread(_sdk._submit, name, body)
    async with _async_client() as c:
        r = await c.post(f"/api/v1/{name}", json=body)
        if r.status_code != 200:
            raise ApiServerError(f"{name}: {r.status_code} {r.text[:300]}")
        return r.json()["request_id"]


async def get(request_id: str, timeout: float = 3600.0,
              poll: float = 0.3) -> Any:
    """Await a request's result (async twin of sdk.get)."""
    if _sdk._TEST_CLIENT is not None:
        return await asyncio.to_thread(_sdk.get, request_id, timeout, poll)
    deadline = asyncio.get_event_loop().time() + timeout
    async with _async_client() as c:
        while True:
            r = await c.get("/api/get", params={"request_id": request_id})
            st = r.json()
            if st["status"] == "SUCCEEDED":
                return st["result"]
            if st["status"] in ("FAILED", "CANCELLED"):
                raise SkyAmdError(
                    f"request {request_id} {st['status']}: {st.get('error')}")
            if asyncio.get_event_loop().time() > deadline:
                raise TimeoutError(f"request {request_id} timed out")
            await asyncio.sleep(poll)


async def stream_and_get(request_id: str, out=None) -> Any:
    """Stream a request's log to `out` (default: stdout) then return
    its result."""
    import sys
    out = out or sys.stdout
    if _sdk._TEST_CLIENT is not None:
        return await asyncio.to_thread(_sdk.stream_and_get, request_id,
                                       out)
    async with _async_client() as c:
        async with c.stream("GET", "/api/stream",
                            params={"request_id": request_id},
                            timeout=None) as r:
            async for chunk in r.aiter_text():
                out.write(chunk)
    return await get(request_id, timeout=30.0)


# ---- submission twins (one per sync endpoint) -----------------------------
async def launch(task, cluster_name: Optional[str] = None, **kw) -> str:
    return await asyncio.to_thread(_sdk.launch, task, cluster_name, **kw)


async def exec(task, cluster_name: str, **kw) -> str:  # noqa: A001
    return await asyncio.to_thread(_sdk.exec, task, cluster_name, **kw)


def _make_async(sync_fn):
    async def wrapper(*a, **kw):
        return await asyncio.to_thread(sync_fn, *a, **kw)
    wrapper.__name__ = sync_fn.__name__
    wrapper.__doc__ = f"Async twin of sdk.{sync_fn.__name__}."
    return wrapper


# Generate awaitable twins for the rest of the sync submission surface.
for _name in ("status", "start", "stop", "down", "autostop", "queue",
              "cancel", "job_status", "jobs_launch", "jobs_queue",
              "jobs_cancel", "jobs_logs", "jobs_pool_apply",
              "jobs_pool_status", "jobs_pool_down", "jobs_group_launch",
              "jobs_group_status", "jobs_group_down", "serve_up",
              "serve_status", "serve_down", "storage_sync", "cost_report",
              "check", "show_gpus", "cluster_events", "cancel_request"):
    _fn = getattr(_sdk, _name, None)
    if _fn is not None:
        globals()[_name] = _make_async(_fn)
