"""Smoke tier (reference tests/smoke_tests/ are YAML-driven
scripts against real infra): launch the CPU-safe example YAMLs
end-to-end through the orchestrator on the fake pool and assert the
jobs SUCCEED."""
import time
from pathlib import Path

import yaml

from tests.test_orchestrator import client, sky_env  # noqa: F401

EXAMPLES = Path(__file__).parent.parent / "examples"

# examples whose `run` is CPU-safe (no torchrun / GPU binaries)
SMOKE = ["hello.yaml", "autostop_dev.yaml", "pool_batch_eval.yaml"]


def _wait_job(sdk, cluster, jid, timeout=60):
    deadline = time.time() + timeout
    j = None
    while time.time() < deadline:
        j = sdk.get(sdk.job_status(cluster, jid))
        if j and j["status"] in ("SUCCEEDED", "FAILED", "FAILED_SETUP",
                                 "CANCELLED", "FAILED_DRIVER"):
            return j
        time.sleep(0.5)
    return j


def test_example_yamls_smoke(client):
    from skypilot_amd.client import sdk
    for name in SMOKE:
        cfg = yaml.safe_load((EXAMPLES / name).read_text())
        cname = f"smoke-{name.split('.')[0].replace('_', '-')}"
        res = sdk.get(sdk.launch(cfg, cname), timeout=90)
        assert res.get("job_id") is not None, (name, res)
        j = _wait_job(sdk, cname, res["job_id"])
        assert j and j["status"] == "SUCCEEDED", (name, j)
        sdk.get(sdk.down(cname))


def test_example_autostop_recorded(client):
    """autostop_dev.yaml's resources.autostop must reach the agent."""
    from skypilot_amd.client import sdk
    cfg = yaml.safe_load((EXAMPLES / "autostop_dev.yaml").read_text())
    sdk.get(sdk.launch(cfg, "smoke-as"), timeout=90)
    recs = sdk.get(sdk.status())
    rec = next(r for r in recs if r["name"] == "smoke-as")
    assert rec["status"] == "UP"
    # the agent reports the configured idle window
    from skypilot_amd.agent.client import AgentClient
    h = rec["handle"]
    a = AgentClient(h["agent_port"], token=h.get("agent_token"))
    try:
        st = a.is_autostopping()
        assert st["idle_minutes"] == 30, st
    finally:
        a.close()
    sdk.get(sdk.down("smoke-as"))


def test_volume_mount_persists_across_clusters(client, tmp_path):
    """`file_mounts: {path: {volume: name}}` mounts a persistent volume
    (reference: sky/volumes); data written by one cluster is visible to
    the next.  Mount targets live under tmp_path — absolute mount paths
    are created verbatim, so tests must never use root-level ones."""
    from skypilot_amd.client import sdk
    from skypilot_amd.data import volumes
    volumes.create("smoke-vol", size_gb=1)
    m1 = str(tmp_path / "vmnt")
    m2 = str(tmp_path / "vmnt2")
    try:
        task1 = {"run": f"echo vol-data-$((3*3)) > {m1}/out.txt",
                 "file_mounts": {m1: {"volume": "smoke-vol"}},
                 "resources": {"cpus": 1}}
        res = sdk.get(sdk.launch(task1, "vol-c1"), timeout=90)
        j = _wait_job(sdk, "vol-c1", res["job_id"])
        assert j["status"] == "SUCCEEDED", j
        sdk.get(sdk.down("vol-c1"))
        # second cluster sees the data
        task2 = {"run": f"cat {m2}/out.txt",
                 "file_mounts": {m2: {"volume": "smoke-vol"}},
                 "resources": {"cpus": 1}}
        res = sdk.get(sdk.launch(task2, "vol-c2"), timeout=90)
        j = _wait_job(sdk, "vol-c2", res["job_id"])
        assert j["status"] == "SUCCEEDED", j
        vp = volumes.mount_path("smoke-vol")
        assert (vp / "out.txt").read_text().strip() == "vol-data-9"
        sdk.get(sdk.down("vol-c2"))
    finally:
        volumes.delete("smoke-vol")
