"""Service runtime: controller (replica manager + autoscaler) + load
balancer, one process per service.

Reference: sky/serve/service.py:_start:391 forks controller
(controller.py:40, autoscaler loop :69) and load balancer
(load_balancer.py:24, httpx reverse proxy); replica lifecycle in
replica_managers.py:764 (launch via execution.launch — the recursion —
readiness probes, recovery of failed replicas).

Usage: python -m skypilot_amd.serve.controller <service_name>
"""
from __future__ import annotations

import collections
import os
import socket
import sys
import threading
import time
import traceback

import httpx
import uvicorn
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response

from skypilot_amd import execution, global_state
from skypilot_amd.agent import job_lib
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.serve import serve_state as st
from skypilot_amd.serve.autoscalers import QPS_WINDOW_SECONDS, make_autoscaler
from skypilot_amd.serve.load_balancing_policies import make_policy
from skypilot_amd.serve.service_spec import ServiceSpec
from skypilot_amd.task import Task

CONTROLLER_LOOP_SECONDS = float(
    os.environ.get("SKY_AMD_SERVE_POLL_SECONDS", "2.0"))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class ServiceRuntime:
    def __init__(self, name: str):
        self.name = name
        rec = st.get_service(name)
        if rec is None:
            raise RuntimeError(f"service {name} not found")
        self.task_cfg = dict(rec["task"])
        self.spec = ServiceSpec.from_config(self.task_cfg.get("service")
                                            or {})
        self.lb_port = rec["lb_port"]
        self.backend = PoolBackend()
        self.policy = make_policy(self.spec.load_balancing_policy)
        self.autoscaler = make_autoscaler(self.spec.policy)
        self.request_times = collections.deque()
        self._ready_replicas: list[str] = []  # endpoints
        self._endpoint_to_rid: dict[str, int] = {}
        self._next_rid = 1
        self._shutdown = threading.Event()

    # ---------------- replica lifecycle ----------------
    def _replica_task(self, port: int) -> Task:
        cfg = dict(self.task_cfg)
        cfg.pop("service", None)
        envs = dict(cfg.get("envs") or {})
        envs["PORT"] = str(port)
        envs["SKYPILOT_SERVE_PORT"] = str(port)
        cfg["envs"] = envs
        return Task.from_yaml_config(cfg)

    def _launch_replica(self):
        svc = st.get_service(self.name) or {}
        version = svc.get("version", 1)
        # Pick up the latest task config (rolling updates).
        if svc.get("task"):
            self.task_cfg = dict(svc["task"])
            self.spec = ServiceSpec.from_config(self.task_cfg.get("service")
                                                or {})
        rid = self._next_rid
        self._next_rid += 1
        port = _free_port()
        cluster = f"sky-serve-{self.name}-{rid}"
        st.upsert_replica(self.name, rid, status=st.R_PROVISIONING,
                          cluster_name=cluster, version=version,
                          endpoint=f"http://127.0.0.1:{port}",
                          launched_at=time.time())
        try:
            task = self._replica_task(port)
            job_id, handle = execution.launch(task, cluster, detach_run=True)
            st.upsert_replica(self.name, rid, status=st.R_STARTING)
            threading.Thread(target=self._probe_until_ready,
                             args=(rid, port, handle, job_id),
                             daemon=True).start()
        except Exception:  # noqa: BLE001
            traceback.print_exc()
            st.upsert_replica(self.name, rid, status=st.R_FAILED)

    def _probe_until_ready(self, rid: int, port: int, handle, job_id):
        probe = self.spec.readiness_probe
        deadline = time.time() + probe.initial_delay_seconds
        url = f"http://127.0.0.1:{port}{probe.path}"
        while time.time() < deadline and not self._shutdown.is_set():
            try:
                job = self.backend._agent(handle).get_job(job_id)
                if job and job["status"] in (job_lib.FAILED,
                                             job_lib.FAILED_SETUP,
                                             job_lib.FAILED_DRIVER):
                    st.upsert_replica(self.name, rid, status=st.R_FAILED)
                    return
                r = httpx.get(url, timeout=probe.timeout_seconds)
                if r.status_code < 500:
                    st.upsert_replica(self.name, rid, status=st.R_READY)
                    return
            except (httpx.HTTPError, OSError):
                pass
            time.sleep(1.0)
        st.upsert_replica(self.name, rid, status=st.R_FAILED)

    def _terminate_replica(self, rid: int, cluster: str):
        st.upsert_replica(self.name, rid, status=st.R_SHUTTING_DOWN)
        record = global_state.get_cluster(cluster)
        if record:
            try:
                self.backend.teardown(record["handle"], terminate=True)
            except Exception:  # noqa: BLE001
                traceback.print_exc()
        st.remove_replica(self.name, rid)

    # ---------------- controller loop ----------------
    def _qps(self) -> float:
        now = time.time()
        while self.request_times and \
                now - self.request_times[0] > QPS_WINDOW_SECONDS:
            self.request_times.popleft()
        return len(self.request_times) / QPS_WINDOW_SECONDS

    def controller_loop(self):
        while not self._shutdown.is_set():
            try:
                self._controller_tick()
            except Exception:  # noqa: BLE001
                traceback.print_exc()
            time.sleep(CONTROLLER_LOOP_SECONDS)

    def _controller_tick(self):
        svc = st.get_service(self.name)
        if svc is None or svc["status"] == st.SHUTTING_DOWN:
            self._shutdown.set()
            return
        replicas = st.list_replicas(self.name)
        # Health recheck for READY replicas (quick probe).
        for r in replicas:
            if r["status"] == st.R_READY:
                try:
                    resp = httpx.get(
                        r["endpoint"] + self.spec.readiness_probe.path,
                        timeout=self.spec.readiness_probe.timeout_seconds)
                    if resp.status_code >= 500:
                        raise httpx.HTTPError("5xx")
                except (httpx.HTTPError, OSError):
                    st.upsert_replica(self.name, r["replica_id"],
                                      status=st.R_NOT_READY)
        replicas = st.list_replicas(self.name)
        # Clean up failed / not-ready replicas (recovery = replace).
        for r in replicas:
            if r["status"] in (st.R_FAILED, st.R_NOT_READY):
                self._terminate_replica(r["replica_id"], r["cluster_name"])
        # Rolling update: when the service version moved past a READY
        # replica's version, replace stale replicas one at a time, only
        # while a newer or equal-count READY capacity exists.
        cur_version = (svc or {}).get("version", 1)
        replicas = st.list_replicas(self.name)
        ready_now = [r for r in replicas if r["status"] == st.R_READY]
        stale_ready = [r for r in ready_now
                       if r.get("version", 1) < cur_version]
        fresh_ready = [r for r in ready_now
                       if r.get("version", 1) >= cur_version]
        if stale_ready and (fresh_ready or len(ready_now) > 1 or
                            self.spec.policy.min_replicas == 1):
            victim = stale_ready[0]
            self._terminate_replica(victim["replica_id"],
                                    victim["cluster_name"])
        replicas = st.list_replicas(self.name)
        alive = [r for r in replicas if r["status"] in
                 (st.R_PROVISIONING, st.R_STARTING, st.R_READY)]
        ready = [r for r in replicas if r["status"] == st.R_READY]
        target = self.autoscaler.target_replicas(self._qps(), len(alive))
        for _ in range(max(0, target - len(alive))):
            self._launch_replica()
        for r in alive[target:] if target < len(alive) else []:
            self._terminate_replica(r["replica_id"], r["cluster_name"])
        # Publish LB routing table + service status.
        self._ready_replicas = [r["endpoint"] for r in ready]
        st.update_service(
            self.name,
            status=st.READY if ready else st.REPLICA_INIT)

    # ---------------- load balancer ----------------
    def lb_app(self) -> FastAPI:
        app = FastAPI()
        client = httpx.AsyncClient(timeout=None)

        @app.get("/-/lb-health")
        def lb_health():
            return {"ok": True, "service": self.name,
                    "ready_replicas": len(self._ready_replicas)}

        @app.api_route("/{path:path}", methods=["GET", "POST", "PUT",
                                                "DELETE", "PATCH"])
        async def proxy(path: str, request: Request):
            self.request_times.append(time.time())
            body = await request.body()
            headers = {k: v for k, v in request.headers.items()
                       if k.lower() not in ("host",)}
            # Failure ejection + retry: a dead replica is removed from
            # the ready set immediately (the controller's health
            # recheck re-adds it if it was a blip) and the request is
            # retried on the survivors — without this, every request
            # routed to a just-killed replica 502s until the next
            # recheck (measured ~50% errors during churn).
            tried = set()
            last_err = "no ready replicas"
            while True:
                candidates = [r for r in self._ready_replicas
                              if r not in tried]
                target = self.policy.pick(candidates)
                if target is None:
                    return JSONResponse({"error": last_err},
                                        status_code=503 if not tried
                                        else 502)
                tried.add(target)
                self.policy.on_start(target)
                try:
                    resp = await client.request(
                        request.method, f"{target}/{path}", content=body,
                        headers=headers,
                        params=dict(request.query_params))
                    return Response(
                        content=resp.content,
                        status_code=resp.status_code,
                        headers={"content-type":
                                 resp.headers.get("content-type",
                                                  "text/plain")})
                except (httpx.HTTPError, OSError) as e:
                    last_err = str(e)
                    try:
                        self._ready_replicas.remove(target)
                    except ValueError:
                        pass
                finally:
                    self.policy.on_finish(target)

        return app

    # ---------------- entry ----------------
    def run(self):
        st.update_service(self.name, controller_pid=os.getpid(),
                          status=st.REPLICA_INIT)
        threading.Thread(target=self.controller_loop, daemon=True).start()
        ssl_kw = {}
        if self.spec.tls is not None:
            tls = self.spec.tls.ensure_materialized(
                st.service_dir(self.name) / "tls")
            ssl_kw = {"ssl_certfile": tls.certfile,
                      "ssl_keyfile": tls.keyfile}
            st.update_service(self.name, tls=True)
        from skypilot_amd import config as sky_config
        lb_host = sky_config.get_nested(["serve", "lb_host"],
                                        "127.0.0.1") or "127.0.0.1"
        config = uvicorn.Config(self.lb_app(), host=lb_host,
                                port=self.lb_port, log_level="warning",
                                **ssl_kw)
        server = uvicorn.Server(config)

        def watch_shutdown():
            while not self._shutdown.is_set():
                time.sleep(0.5)
            server.should_exit = True

        threading.Thread(target=watch_shutdown, daemon=True).start()
        server.run()
        # Teardown all replicas on exit.
        for r in st.list_replicas(self.name):
            self._terminate_replica(r["replica_id"], r["cluster_name"])
        st.update_service(self.name, status=st.SHUTDOWN)


def main():
    ServiceRuntime(sys.argv[1]).run()


if __name__ == "__main__":
    main()
