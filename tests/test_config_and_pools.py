"""Config layering + SSH pool parsing tests."""
import os

import yaml


def test_config_layering(tmp_path, monkeypatch):
    from skypilot_amd import config
    user = tmp_path / "user.yaml"
    user.write_text(yaml.safe_dump({"train": {"bucket_mb": 128}}))
    monkeypatch.setattr(config, "USER_CONFIG_PATH", str(user))
    cfg = config.load(refresh=True)
    assert cfg["train"]["bucket_mb"] == 128
    assert cfg["pool"]["accelerator"] == "MI355X"  # default preserved
    assert config.get_nested(["train", "bucket_mb"]) == 128
    assert config.get_nested(["train", "bucket_mb"],
                             override_configs={"train": {"bucket_mb": 7}}) == 7
    assert config.get_nested(["no", "such"], default=42) == 42
    config.load(refresh=True)


def test_ssh_pool_parsing(tmp_path, monkeypatch):
    from skypilot_amd.provision import ssh_pool
    pools = tmp_path / "pools.yaml"
    pools.write_text(yaml.safe_dump({
        "default": {"hosts": [
            {"ip": "10.0.0.5", "user": "amd", "gpus": 8},
            "10.0.0.6",
        ]},
    }))
    hosts = ssh_pool.parse_hosts()
    # default path has no pools file in test env -> empty or parsed
    monkeypatch.setattr(ssh_pool, "POOLS_PATH", str(pools))
    hosts = ssh_pool.parse_hosts()
    assert len(hosts) == 2
    assert hosts[0]["user"] == "amd" and hosts[0]["gpus"] == 8
    assert hosts[1]["ip"] == "10.0.0.6"


def test_timeline_records(tmp_path, monkeypatch):
    import importlib
    from skypilot_amd.utils import timeline
    monkeypatch.setattr(timeline, "_path", str(tmp_path / "t.json"))
    with timeline.Event("unit-test"):
        pass
    assert any(e["name"] == "unit-test" for e in timeline._events)


def test_admin_policy_apply(monkeypatch):
    from skypilot_amd import admin_policy, config
    # no policy configured -> identity
    cfg = {"run": "echo hi"}
    assert admin_policy.apply(cfg) == cfg

    class AddLabel(admin_policy.AdminPolicy):
        def validate_and_mutate(self, request):
            tc = dict(request.task_config)
            tc.setdefault("envs", {})["POLICY"] = "1"
            return admin_policy.MutatedUserRequest(task_config=tc)

    monkeypatch.setattr(admin_policy, "load_policy", lambda: AddLabel())
    out = admin_policy.apply(cfg)
    assert out["envs"]["POLICY"] == "1"


def test_volumes_crud(tmp_path, monkeypatch):
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path / "home"))
    from skypilot_amd.data import volumes
    v = volumes.create("vol1", size_gb=10)
    assert v["name"] == "vol1"
    assert any(x["name"] == "vol1" for x in volumes.list_volumes())
    p = volumes.mount_path("vol1")
    (p / "data.txt").write_text("x")
    assert volumes.delete("vol1")
    assert not any(x["name"] == "vol1" for x in volumes.list_volumes())


def test_recipes_list():
    from skypilot_amd import recipes
    rs = recipes.list_recipes()
    names = {r["name"] for r in rs}
    assert "hello" in names and "train_llama3_8b" in names
    assert recipes.get_recipe_path("hello").endswith("hello.yaml")


def test_k8s_pod_manifest_rendering():
    from skypilot_amd.provision import k8s
    m = k8s.render_pod_manifest("c1", 4, {
        "image": "img:1", "gpu_resource": "amd.com/gpu",
        "namespace": "ns"})
    assert m["metadata"]["name"] == "sky-amd-c1"
    ctr = m["spec"]["containers"][0]
    assert ctr["resources"]["limits"]["amd.com/gpu"] == 4
    assert "--gpu-ids" in ctr["command"]
    assert "0,1,2,3" in ctr["command"]
    # CPU-only pod: no GPU limits
    m0 = k8s.render_pod_manifest("c2", 0, {"image": "i",
                                           "gpu_resource": "amd.com/gpu"})
    assert m0["spec"]["containers"][0]["resources"] == {}


def test_provision_dispatch_registry():
    from skypilot_amd import provision
    for name in ("local", "ssh", "kubernetes", "k8s"):
        assert provision._impl(name) is not None
    import pytest as _pytest
    from skypilot_amd.exceptions import ResourcesUnavailableError
    with _pytest.raises(ResourcesUnavailableError):
        provision._impl("aws")
