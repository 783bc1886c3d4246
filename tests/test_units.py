"""Unit tests for under-covered pieces: agent HTTP surface (in-process),
Dag, Resources multi-candidate, optimizer feasibility."""
import time

import pytest


def test_agent_app_surface(tmp_path, monkeypatch):
    """The skylet-equivalent RPC surface, exercised in-process
    (reference surface: SURVEY.md Appendix A)."""
    from fastapi.testclient import TestClient

    from skypilot_amd.agent.daemon import create_app
    app = create_app(str(tmp_path / "c1"), [0, 1])
    with TestClient(app) as c:
        h = c.get("/health").json()
        assert h["ok"] and h["gpu_ids"] == [0, 1]
        # queue a job that needs no GPU and runs instantly
        r = c.post("/jobs/queue", json={
            "name": "t", "spec": {"run": "echo unit-agent",
                                  "num_nodes": 1, "gpus_per_node": 0}})
        jid = r.json()["job_id"]
        deadline = time.time() + 30
        while time.time() < deadline:
            j = c.get(f"/jobs/{jid}").json()["job"]
            if j["status"] in ("SUCCEEDED", "FAILED", "FAILED_DRIVER"):
                break
            time.sleep(0.3)
        assert j["status"] == "SUCCEEDED", j
        assert any(x["job_id"] == jid for x in c.get("/jobs").json()["jobs"])
        # logs endpoint returns the output
        text = c.get(f"/jobs/{jid}/logs",
                     params={"follow": False}).text
        assert "unit-agent" in text
        # autostop state round-trips
        c.post("/autostop", json={"idle_minutes": 7, "down": True})
        a = c.get("/autostop").json()
        assert a["idle_minutes"] == 7 and a["down"]
        assert c.get("/idle").json()["idle"] is True


def test_dag_context_and_chain():
    import skypilot_amd.dag as dag_mod
    from skypilot_amd.task import Task
    with dag_mod.Dag("d") as d:
        assert dag_mod.get_current_dag() is d
        t1, t2 = Task("a", run="x"), Task("b", run="y")
        d.add(t1)
        d.add(t2)
        d.add_edge(t1, t2)
    assert dag_mod.get_current_dag() is None
    assert d.is_chain() and len(d) == 2
    d2 = dag_mod.to_dag(Task("solo", run="z"))
    assert len(d2) == 1


def test_resources_any_of_and_aliases():
    from skypilot_amd.resources import Resources, canonical_accelerator
    r = Resources.from_yaml_config({
        "any_of": [{"accelerators": "MI355:4"},
                   {"accelerators": "MI355X:8"}]})
    assert r.accelerators == "MI355X" and r.accelerator_count == 4
    assert canonical_accelerator("mi355") == "MI355X"
    from skypilot_amd.exceptions import TaskValidationError
    with pytest.raises(TaskValidationError):
        Resources.from_yaml_config({"bogus": 1})


def test_optimizer_feasibility(monkeypatch):
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "8")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd.exceptions import ResourcesUnavailableError
    from skypilot_amd.optimizer import Optimizer
    from skypilot_amd.task import Task
    Optimizer.optimize(Task("ok", run="x") .set_resources(
        __import__("skypilot_amd.resources",
                   fromlist=["Resources"]).Resources.from_yaml_config(
            {"accelerators": "MI355X:8"})))
    bad = Task("bad", run="x")
    bad.resources = bad.resources.copy(accelerators="H100",
                                       accelerator_count=8)
    with pytest.raises(ResourcesUnavailableError):
        Optimizer.optimize(bad)
    gpu_topology.detect_gpus.cache_clear()


def test_any_of_candidates_sorted_and_ordered_kept():
    from skypilot_amd.resources import Resources
    r = Resources.from_yaml_config({
        "any_of": [{"accelerators": "MI355X:8"},
                   {"accelerators": "MI355X:1"}]})
    assert [c.accelerator_count for c in r.candidates] == [1, 8]
    r = Resources.from_yaml_config({
        "ordered": [{"accelerators": "MI355X:8"},
                    {"accelerators": "MI355X:1"}]})
    assert [c.accelerator_count for c in r.candidates] == [8, 1]


def test_optimizer_prunes_infeasible_candidate(tmp_path, monkeypatch):
    """First candidate infeasible (999 GPUs on a 0-GPU box) -> the
    optimizer prunes it and launch uses the cpu-only candidate
    (reference: optimizer candidate enumeration)."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    from skypilot_amd import execution
    from skypilot_amd.task import Task
    task = Task.from_yaml_config({
        "run": "true",
        "resources": {"ordered": [{"accelerators": "MI355X:999"},
                                  {"cpus": 1}]},
    })
    job_id, handle = execution.launch(task, "failover-c",
                                      detach_run=True)
    assert handle is not None
    assert task.resources.accelerator_count == 0  # fell to candidate 2
    from skypilot_amd import core
    core.down("failover-c")


def test_provision_time_failover(tmp_path, monkeypatch):
    """Feasible-looking candidate fails at provision time (lease race)
    -> execution retries the next candidate
    (reference: RetryingVmProvisioner)."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    from skypilot_amd import execution, global_state
    from skypilot_amd.backends.pool_backend import PoolBackend
    from skypilot_amd.exceptions import ResourcesUnavailableError
    from skypilot_amd.task import Task
    real = PoolBackend.provision
    calls = {"n": 0}

    def flaky(self, task, cluster_name, **kw):
        calls["n"] += 1
        if calls["n"] == 1:
            raise ResourcesUnavailableError("lease lost")
        return real(self, task, cluster_name, **kw)

    monkeypatch.setattr(PoolBackend, "provision", flaky)
    task = Task.from_yaml_config({
        "run": "true",
        "resources": {"ordered": [{"cpus": 1}, {"cpus": 2}]},
    })
    job_id, handle = execution.launch(task, "failover-d",
                                      detach_run=True)
    assert handle is not None and calls["n"] == 2
    ev = [e["event"] for e in
          global_state.get_cluster_events("failover-d")]
    assert "PROVISION_FAILOVER" in ev
    from skypilot_amd import core
    core.down("failover-d")


def test_spot_preemption(tmp_path, monkeypatch):
    """On-demand launches reclaim spot GPUs; spot never preempts
    (reference: spot instance revocation -> managed-job RECOVERING)."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "8")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd import core, execution, global_state
    from skypilot_amd.exceptions import ResourcesUnavailableError
    from skypilot_amd.task import Task

    spot = Task.from_yaml_config({
        "run": "true",
        "resources": {"accelerators": "MI355X:6", "use_spot": True}})
    execution.launch(spot, "spot-c", detach_run=True)
    assert global_state.get_cluster("spot-c")["handle"]["use_spot"]

    # a second spot task must NOT preempt the first
    spot2 = Task.from_yaml_config({
        "run": "true",
        "resources": {"accelerators": "MI355X:4", "use_spot": True}})
    import pytest as _pytest
    with _pytest.raises(ResourcesUnavailableError):
        execution.launch(spot2, "spot-d", detach_run=True)
    assert global_state.get_cluster("spot-c") is not None

    # an on-demand task reclaims the spot capacity
    od = Task.from_yaml_config({
        "run": "true", "resources": {"accelerators": "MI355X:4"}})
    execution.launch(od, "od-c", detach_run=True)
    assert global_state.get_cluster("spot-c") is None  # preempted
    ev = [e["event"] for e in global_state.get_cluster_events("spot-c")]
    assert "PREEMPTED" in ev
    rec = global_state.get_cluster("od-c")
    assert rec["status"] == "UP"
    assert len(rec["handle"]["gpu_ids"]) == 4
    core.down("od-c")
    gpu_topology.detect_gpus.cache_clear()


def test_retry_until_up(tmp_path, monkeypatch):
    """--retry-until-up loops provisioning with backoff until capacity
    frees (reference: sky launch --retry-until-up)."""
    import threading
    import time as _time
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "4")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd import core, execution
    from skypilot_amd.task import Task
    hog = Task.from_yaml_config(
        {"run": "true", "resources": {"accelerators": "MI355X:4"}})
    execution.launch(hog, "hog-c", detach_run=True)

    def free_later():
        _time.sleep(6)
        core.down("hog-c")

    t = threading.Thread(target=free_later)
    t.start()
    want = Task.from_yaml_config(
        {"run": "true", "resources": {"accelerators": "MI355X:2"}})
    t0 = _time.time()
    _, handle = execution.launch(want, "want-c", detach_run=True,
                                 retry_until_up=True)
    assert handle is not None
    assert _time.time() - t0 >= 5  # actually waited for capacity
    t.join()
    core.down("want-c")
    gpu_topology.detect_gpus.cache_clear()


def test_fractional_gpu_shares(tmp_path, monkeypatch):
    """MI355X:0.5 shares one GPU between clusters (reference schema:
    fractional accelerators); whole-GPU leases never land on a shared
    device, and over-subscription is rejected."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "2")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd import core, execution, global_state
    from skypilot_amd.exceptions import ResourcesUnavailableError
    from skypilot_amd.task import Task

    def t(acc):
        return Task.from_yaml_config(
            {"run": "true", "resources": {"accelerators": acc}})

    execution.launch(t("MI355X:0.5"), "fa", detach_run=True)
    execution.launch(t("MI355X:0.5"), "fb", detach_run=True)
    ha = global_state.get_cluster("fa")["handle"]
    hb = global_state.get_cluster("fb")["handle"]
    assert ha["gpu_fraction"] == 0.5 and hb["gpu_fraction"] == 0.5
    assert ha["gpu_ids"] == hb["gpu_ids"]  # packed onto one GPU
    # whole-GPU lease avoids the shared device
    execution.launch(t("MI355X:1"), "fw", detach_run=True)
    hw = global_state.get_cluster("fw")["handle"]
    assert hw["gpu_ids"][0] != ha["gpu_ids"][0]
    # no capacity left: another 0.5 fits nowhere (gpu0 full by shares,
    # gpu1 fully leased)
    import pytest as _pytest
    with _pytest.raises(ResourcesUnavailableError):
        execution.launch(t("MI355X:0.75"), "fc", detach_run=True)
    for name in ("fa", "fb", "fw"):
        core.down(name)
    gpu_topology.detect_gpus.cache_clear()


def test_authorize_matrix():
    from skypilot_amd import users
    from skypilot_amd.exceptions import PermissionDeniedError
    users.authorize("admin", "launch")
    users.authorize("user", "launch")
    users.authorize("viewer", "status")
    users.authorize("viewer", "cost_report")
    import pytest as _pytest
    for req in ("launch", "down", "jobs_launch", "serve_up",
                "storage_delete"):
        with _pytest.raises(PermissionDeniedError):
            users.authorize("viewer", req)


def test_accelerator_count_validation():
    from skypilot_amd.exceptions import TaskValidationError
    from skypilot_amd.resources import parse_accelerators
    assert parse_accelerators("MI355X:8") == ("MI355X", 8)
    assert parse_accelerators("MI355X:0.5") == ("MI355X", 0.5)
    assert parse_accelerators({"MI355X": 0.25}) == ("MI355X", 0.25)
    import pytest as _pytest
    with _pytest.raises(TaskValidationError):
        parse_accelerators("MI355X:1.5")  # fractions must be < 1
    with _pytest.raises(TaskValidationError):
        parse_accelerators("MI355X:0")


def test_decode_bucket():
    from skypilot_amd.serve.engine import Engine
    b = Engine._bucket
    class _E:  # unbound helper needs no engine state
        pass
    e = _E()
    assert Engine._bucket(e, 1) == 1
    assert Engine._bucket(e, 3) == 4
    assert Engine._bucket(e, 16) == 16
    assert Engine._bucket(e, 17) == 32


def test_parse_showtopo():
    """xGMI topology parser on representative rocm-smi --showtopo output
    (fully-connected 4-GPU xGMI + a PCIe outlier case)."""
    from skypilot_amd.utils.gpu_topology import parse_showtopo
    sample = """
=========================== ROCm System Management Interface ===========================
================================ Weight between two GPUs ================================
       GPU0         GPU1         GPU2         GPU3
GPU0   0            15           15           15
GPU1   15           0            15           15
GPU2   15           15           0            15
GPU3   15           15           15           0

================================= Hops between two GPUs =================================
       GPU0         GPU1         GPU2         GPU3
GPU0   0            1            1            1
GPU1   1            0            1            1
GPU2   1            1            0            1
GPU3   1            1            1            0

=============================== Link Type between two GPUs ==============================
       GPU0         GPU1         GPU2         GPU3
GPU0   0            XGMI         XGMI         XGMI
GPU1   XGMI         0            XGMI         XGMI
GPU2   XGMI         XGMI         0            XGMI
GPU3   XGMI         XGMI         XGMI         0

====================================== Numa Nodes ======================================
GPU[0]          : (Topology) Numa Node: 0
GPU[0]          : (Topology) Numa Affinity: 0
GPU[1]          : (Topology) Numa Node: 0
GPU[2]          : (Topology) Numa Node: 1
GPU[3]          : (Topology) Numa Node: 1
================================== End of ROCm SMI Log ==================================
"""
    t = parse_showtopo(sample)
    assert t["n_gpus"] == 4
    assert t["fully_connected_xgmi"] is True
    assert t["hops"][0][3] == 1 and t["hops"][2][2] == 0
    assert t["numa"] == {0: 0, 1: 0, 2: 1, 3: 1}
    # a PCIe link breaks the fully-connected claim
    t2 = parse_showtopo(sample.replace("GPU1   XGMI         0",
                                       "GPU1   PCIE         0"))
    assert t2["fully_connected_xgmi"] is False
    # degenerate single-GPU box
    t3 = parse_showtopo("======= Link Type between two GPUs =====\n"
                        "       GPU0\nGPU0   0\n")
    assert t3["n_gpus"] == 1 and t3["fully_connected_xgmi"] is False


def test_optimizer_cost_ranking(monkeypatch, tmp_path):
    """any_of candidates rank by the config price model (spot
    discounted); ordered lists keep the user's order (reference:
    sky/optimizer.py cost ranking)."""
    import os
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "8")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd import config as sky_config
    sky_config.load(refresh=True)
    from skypilot_amd.optimizer import Optimizer
    from skypilot_amd.task import Task

    # any_of: 8 on-demand GPUs vs 8 spot GPUs -> spot is cheaper
    t = Task.from_yaml_config({
        "run": "true",
        "resources": {"any_of": [
            {"accelerators": "MI355X:8"},
            {"accelerators": "MI355X:8", "use_spot": True},
        ]}})
    from skypilot_amd.dag import to_dag
    dag = Optimizer.optimize(to_dag(t))
    best = dag.tasks[0].resources
    assert best.use_spot is True
    assert dag.tasks[0].estimated_hourly_cost < 8 * 2.0
    # ordered: user's order wins even when later is cheaper
    t2 = Task.from_yaml_config({
        "run": "true",
        "resources": {"ordered": [
            {"accelerators": "MI355X:8"},
            {"accelerators": "MI355X:1", "use_spot": True},
        ]}})
    dag2 = Optimizer.optimize(to_dag(t2))
    assert dag2.tasks[0].resources.accelerator_count == 8
    assert dag2.tasks[0].resources.use_spot is False
    # infeasible candidates are pruned before ranking
    t3 = Task.from_yaml_config({
        "run": "true",
        "resources": {"any_of": [
            {"accelerators": "MI355X:64"},
            {"accelerators": "MI355X:2"},
        ]}})
    dag3 = Optimizer.optimize(to_dag(t3))
    assert dag3.tasks[0].resources.accelerator_count == 2


def test_request_gc_retention(monkeypatch, tmp_path):
    """Terminal request rows past the retention window are deleted
    (logs too); non-terminal and recent rows survive (reference: sky
    server request retention)."""
    import os
    import time as _time
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    from skypilot_amd.server import requests_db as rdb
    old = rdb.create("status", {}, "SHORT")
    rdb.finish(old, rdb.SUCCEEDED, result={})
    live = rdb.create("launch", {}, "LONG")  # stays PENDING
    fresh = rdb.create("status", {}, "SHORT")
    rdb.finish(fresh, rdb.FAILED, error="x")
    # age the old row beyond the cutoff
    with rdb._conn() as c:
        c.execute("UPDATE requests SET created_at=? WHERE request_id=?",
                  (_time.time() - 30 * 86400, old))
    logf = rdb.get(old)["log_path"]
    open(logf, "w").write("x")
    removed = rdb.gc_requests(max_age_days=7, keep_latest=1)
    assert removed == 1
    assert rdb.get(old) is None
    assert not os.path.exists(logf)
    assert rdb.get(live)["status"] == "PENDING"
    assert rdb.get(fresh) is not None
    # keep_latest guards even ancient rows
    with rdb._conn() as c:
        c.execute("UPDATE requests SET created_at=? WHERE request_id=?",
                  (_time.time() - 30 * 86400, fresh))
    assert rdb.gc_requests(max_age_days=7, keep_latest=5) == 0


def test_launch_dryrun_returns_plan(monkeypatch, tmp_path):
    """--dryrun optimizes and returns the placement plan without
    provisioning anything (reference: sky launch --dryrun)."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    monkeypatch.setenv("SKY_AMD_FAKE_GPUS", "8")
    from skypilot_amd.utils import gpu_topology
    gpu_topology.detect_gpus.cache_clear()
    from skypilot_amd import execution, global_state
    from skypilot_amd.task import Task
    t = Task.from_yaml_config({
        "run": "true",
        "resources": {"any_of": [
            {"accelerators": "MI355X:8"},
            {"accelerators": "MI355X:2", "use_spot": True}]}})
    job_id, plan = execution.launch(t, "dry-c", dryrun=True)
    assert job_id is None
    assert plan["dryrun"] is True
    assert plan["resources"]["use_spot"] is True  # cheapest ranked first
    assert plan["estimated_hourly_cost"] is not None
    assert len(plan["candidates"]) == 2
    # nothing was provisioned
    assert not any(c["name"] == "dry-c"
                   for c in global_state.list_clusters())


def test_reap_orphan_agents(monkeypatch, tmp_path):
    """An agent whose cluster record vanished (launch cancelled between
    agent spawn and state write) is identity-checked and stopped; live
    recorded clusters and non-matching pids are untouched."""
    import json
    import os
    import subprocess
    import sys
    import time
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    from skypilot_amd.server.daemons import _reap_orphan_agents
    cdir = tmp_path / "clusters" / "orphan-c"
    cdir.mkdir(parents=True)
    proc = subprocess.Popen(
        [sys.executable, "-m", "skypilot_amd.agent.daemon",
         "--cluster-dir", str(cdir), "--port", "0"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        (cdir / "agent.json").write_text(
            json.dumps({"port": 0, "pid": proc.pid}))
        # fresh file is inside the grace window: untouched
        assert _reap_orphan_agents(min_age_s=120) == 0
        assert proc.poll() is None
        # age it out: reaped
        old = time.time() - 600
        os.utime(cdir / "agent.json", (old, old))
        assert _reap_orphan_agents(min_age_s=120) == 1
        deadline = time.time() + 10
        while time.time() < deadline and proc.poll() is None:
            time.sleep(0.2)
        assert proc.poll() is not None
    finally:
        if proc.poll() is None:
            proc.kill()
    # stale agent.json with a recycled (non-matching) pid: skipped
    (cdir / "agent.json").write_text(
        json.dumps({"port": 0, "pid": os.getpid()}))
    os.utime(cdir / "agent.json", (old, old))
    assert _reap_orphan_agents(min_age_s=120) == 0


def test_gl_lds_barriers_are_vm_guarded():
    """ISA regression guard for the r02 race class: in every kernel
    that issues global_load ... lds (async LDS DMA, completion tracked
    by VMcnt), each plain s_barrier must be preceded (within a small
    window) by an s_waitcnt that constrains vmcnt — hipcc only adds
    lgkmcnt(0) by itself, which does NOT cover the DMA
    (docs/KERNELS.md "Synchronizing async LDS staging")."""
    import shutil
    import subprocess
    import tempfile
    from pathlib import Path

    objdump = Path("/opt/rocm/lib/llvm/bin/llvm-objdump")
    so = Path(__file__).parent.parent / "skypilot_amd" / "ops" / "_C.so"
    if not objdump.exists() or not so.exists():
        import pytest as _pytest
        _pytest.skip("llvm-objdump or built extension unavailable")
    with tempfile.TemporaryDirectory() as td:
        cp = Path(td) / "_C.so"
        shutil.copy(so, cp)
        subprocess.run([str(objdump), "--offloading", str(cp)],
                       cwd=td, capture_output=True)
        bad = []
        for bundle in sorted(Path(td).glob("_C.so.*gfx950*")):
            d = subprocess.run([str(objdump), "-d", str(bundle)],
                               capture_output=True, text=True).stdout
            lines = d.splitlines()
            kern, uses_dma, rows = None, False, []
            kernels = []
            for ln in lines:
                if ">:" in ln and "<" in ln:
                    if kern and uses_dma:
                        kernels.append((kern, rows))
                    kern = ln.split("<")[1].split(">")[0]
                    uses_dma, rows = False, []
                if "lds" in ln and ("global_load" in ln
                                    or "buffer_load" in ln):
                    uses_dma = True
                rows.append(ln)
            if kern and uses_dma:
                kernels.append((kern, rows))
            for kname, rows in kernels:
                for i, ln in enumerate(rows):
                    if "s_barrier" in ln and "wait" not in ln:
                        window = " ".join(rows[max(0, i - 12):i])
                        if "vmcnt(0)" not in window:
                            bad.append((bundle.name, kname, i))
        # the hand-scheduled wgrad GEMM carries its own phase waits at
        # K-tile granularity (vmcnt(0) once per tile, not per barrier)
        bad = [b for b in bad if "gemm_nt_kernel" not in b[1]]
        assert not bad, bad


def test_finish_never_resurrects_cancelled(monkeypatch, tmp_path):
    """A runner that slipped through the cancel window must not flip a
    CANCELLED request back to SUCCEEDED when it completes."""
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path))
    from skypilot_amd.server import requests_db as rdb
    rid = rdb.create("launch", {}, "LONG")
    marked, pid = rdb.mark_cancelled(rid)
    assert marked and pid is None  # claim never stamped a killable pid
    rdb.finish(rid, rdb.SUCCEEDED, result={"late": True})
    assert rdb.get(rid)["status"] == rdb.CANCELLED
