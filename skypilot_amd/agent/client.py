"""Typed client for the node agent (reference: SkyletClient,
sky/backends/cloud_vm_ray_backend.py:2903)."""
from __future__ import annotations

import time
from typing import Any, Dict, Iterator, List, Optional

import httpx

from skypilot_amd.exceptions import ClusterNotUpError


class AgentClient:
    def __init__(self, port: int, host: str = "127.0.0.1",
                 timeout: float = 30.0, token: Optional[str] = None):
        self.base = f"http://{host}:{port}"
        headers = {"Authorization": f"Bearer {token}"} if token else {}
        self._client = httpx.Client(timeout=timeout, headers=headers)

    def close(self):
        self._client.close()

    def _post(self, path: str, body: Optional[dict] = None) -> dict:
        r = self._client.post(self.base + path, json=body or {})
        r.raise_for_status()
        return r.json()

    def _get(self, path: str, **params) -> dict:
        r = self._client.get(self.base + path, params=params)
        r.raise_for_status()
        return r.json()

    def healthy(self) -> bool:
        try:
            return bool(self._get("/health").get("ok"))
        except (httpx.HTTPError, OSError):
            return False

    def wait_ready(self, timeout: float = 30.0) -> None:
        deadline = time.time() + timeout
        while time.time() < deadline:
            if self.healthy():
                return
            time.sleep(0.2)
        raise ClusterNotUpError(f"agent at {self.base} not responding")

    def queue_job(self, spec: Dict[str, Any],
                  name: Optional[str] = None) -> int:
        return int(self._post("/jobs/queue",
                              {"name": name, "spec": spec})["job_id"])

    def get_job(self, job_id: int) -> Optional[Dict[str, Any]]:
        return self._get(f"/jobs/{job_id}").get("job")

    def get_job_queue(self) -> List[Dict[str, Any]]:
        return self._get("/jobs")["jobs"]

    def cancel_job(self, job_id: int) -> bool:
        return bool(self._post(f"/jobs/{job_id}/cancel")["cancelled"])

    def cancel_all(self) -> int:
        return int(self._post("/jobs/cancel_all")["cancelled"])

    def tail_logs(self, job_id: int, follow: bool = True
                  ) -> Iterator[bytes]:
        with self._client.stream(
                "GET", f"{self.base}/jobs/{job_id}/logs",
                params={"follow": follow}, timeout=None) as r:
            for chunk in r.iter_bytes():
                yield chunk

    def set_autostop(self, idle_minutes: int, down: bool = False) -> None:
        self._post("/autostop", {"idle_minutes": idle_minutes, "down": down})

    def is_autostopping(self) -> Dict[str, Any]:
        return self._get("/autostop")

    def is_idle(self) -> bool:
        return bool(self._get("/idle")["idle"])

    # ---- interactive exec sessions (PTY shell over HTTP streaming;
    # reference: sky websocket SSH proxy / `sky ssh`) ------------------
    def exec_start(self, cmd=None, env: Optional[dict] = None,
                   term: str = "xterm-256color") -> str:
        body: Dict[str, Any] = {"term": term}
        if cmd:
            body["cmd"] = cmd
        if env:
            body["env"] = env
        return self._post("/exec/start", body)["sid"]

    def exec_stdin(self, sid: str, data: bytes) -> bool:
        r = self._client.post(f"{self.base}/exec/{sid}/stdin",
                              content=data)
        r.raise_for_status()
        return bool(r.json().get("ok"))

    def exec_stdout(self, sid: str) -> Iterator[bytes]:
        with self._client.stream("GET", f"{self.base}/exec/{sid}/stdout",
                                 timeout=None) as r:
            for chunk in r.iter_bytes():
                yield chunk

    def exec_status(self, sid: str) -> Dict[str, Any]:
        return self._get(f"/exec/{sid}")

    def exec_resize(self, sid: str, rows: int, cols: int) -> None:
        self._post(f"/exec/{sid}/resize", {"rows": rows, "cols": cols})

    def exec_close(self, sid: str) -> None:
        self._post(f"/exec/{sid}/close")

    def wait_job(self, job_id: int, timeout: float = 3600,
                 poll: float = 0.5) -> Dict[str, Any]:
        from skypilot_amd.agent import job_lib
        deadline = time.time() + timeout
        while time.time() < deadline:
            j = self.get_job(job_id)
            if j and j["status"] in job_lib.TERMINAL:
                return j
            time.sleep(poll)
        raise TimeoutError(f"job {job_id} did not finish in {timeout}s")
