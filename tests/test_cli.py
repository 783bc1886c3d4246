"""CLI surface tests via click's CliRunner + the in-process harness."""
import time

from click.testing import CliRunner

from tests.test_orchestrator import client, sky_env  # noqa: F401


def test_cli_launch_status_queue_down(client, tmp_path):
    from skypilot_amd.cli.main import cli
    r = CliRunner()
    yaml_path = tmp_path / "t.yaml"
    yaml_path.write_text("name: cli-test\nrun: echo cli-ok\n")
    res = r.invoke(cli, ["launch", str(yaml_path), "-c", "t-cli",
                         "--detach-run"])
    assert res.exit_code == 0, res.output
    assert "job_id=1" in res.output

    res = r.invoke(cli, ["status"])
    assert res.exit_code == 0 and "t-cli" in res.output

    res = r.invoke(cli, ["queue", "t-cli"])
    assert res.exit_code == 0 and "cli-test" in res.output

    deadline = time.time() + 30
    while time.time() < deadline:
        res = r.invoke(cli, ["queue", "t-cli"])
        if "SUCCEEDED" in res.output:
            break
        time.sleep(0.5)
    assert "SUCCEEDED" in res.output

    res = r.invoke(cli, ["logs", "t-cli", "1", "--no-follow"])
    assert "cli-ok" in res.output

    res = r.invoke(cli, ["show-gpus"])
    assert res.exit_code == 0 and "MI355X" in res.output

    res = r.invoke(cli, ["check"])
    assert res.exit_code == 0 and "local" in res.output

    res = r.invoke(cli, ["recipes"])
    assert res.exit_code == 0 and "hello" in res.output

    res = r.invoke(cli, ["down", "t-cli"])
    assert res.exit_code == 0


def test_cli_inline_command(client):
    from skypilot_amd.cli.main import cli
    r = CliRunner()
    res = r.invoke(cli, ["launch", "echo inline-run", "-c", "t-inline",
                         "--detach-run"])
    assert res.exit_code == 0, res.output
    r.invoke(cli, ["down", "t-inline"])


def test_cli_users_cost_group_storage(client, tmp_path):
    """Surface checks for the round-1 additions: sky users / cost-report
    / jobs group / storage sync."""
    from skypilot_amd.cli.main import cli
    r = CliRunner()

    res = r.invoke(cli, ["users", "list"])
    assert res.exit_code == 0 and "admin" in res.output

    res = r.invoke(cli, ["users", "token", "cli-bot", "--role", "viewer"])
    assert res.exit_code == 0 and res.output.strip().startswith("sky_")
    res = r.invoke(cli, ["users", "tokens"])
    assert "cli-bot" in res.output
    res = r.invoke(cli, ["users", "revoke", "cli-bot"])
    assert res.exit_code == 0

    res = r.invoke(cli, ["cost-report"])
    assert res.exit_code == 0 and "GPU-HOURS" in res.output

    gy = tmp_path / "grp.yaml"
    gy.write_text(
        "tasks:\n"
        "  - name: a\n    run: echo A\n"
        "  - name: b\n    run: echo B\n")
    res = r.invoke(cli, ["jobs", "group", "launch", str(gy), "-n", "cg"])
    assert res.exit_code == 0, res.output
    res = r.invoke(cli, ["jobs", "group", "status", "cg"])
    assert res.exit_code == 0 and '"cluster": "sky-group-cg"' in res.output
    res = r.invoke(cli, ["jobs", "group", "down", "cg"])
    assert res.exit_code == 0

    res = r.invoke(cli, ["storage", "ls"])
    assert res.exit_code == 0


def test_logs_status_flag(client):
    """`sky logs --status` prints the job status and exits 0 iff
    SUCCEEDED (reference: sky logs --status)."""
    from click.testing import CliRunner
    from skypilot_amd.cli.main import cli
    runner = CliRunner()
    r = runner.invoke(cli, ["launch", "echo ok", "-c", "st-c",
                            "--detach-run"])
    assert r.exit_code == 0, r.output
    import time
    from skypilot_amd.client import sdk
    deadline = time.time() + 60
    while time.time() < deadline:
        j = sdk.get(sdk.job_status("st-c", 1))
        if j and j["status"] in ("SUCCEEDED", "FAILED"):
            break
        time.sleep(0.3)
    r = runner.invoke(cli, ["logs", "st-c", "1", "--status"])
    assert r.exit_code == 0 and "SUCCEEDED" in r.output, r.output
    runner.invoke(cli, ["down", "st-c", "-y"])


def test_autostop_cancel(client, tmp_path):
    """`sky autostop --cancel` clears a scheduled autostop."""
    from skypilot_amd.cli.main import cli
    r = CliRunner()
    y = tmp_path / "a.yaml"
    y.write_text("run: 'true'\nresources: {cpus: 1}\n")
    assert r.invoke(cli, ["launch", str(y), "-c", "as-c",
                          "--detach-run"]).exit_code == 0
    assert r.invoke(cli, ["autostop", "as-c", "-i", "15"]).exit_code == 0
    from skypilot_amd.client import sdk
    recs = sdk.get(sdk.status())
    h = next(x for x in recs if x["name"] == "as-c")["handle"]
    from skypilot_amd.agent.client import AgentClient
    a = AgentClient(h["agent_port"], token=h.get("agent_token"))
    try:
        assert a.is_autostopping()["idle_minutes"] == 15
        assert r.invoke(cli, ["autostop", "as-c", "--cancel"]
                        ).exit_code == 0
        assert a.is_autostopping()["idle_minutes"] == -1
    finally:
        a.close()
    r.invoke(cli, ["down", "as-c", "-y"])


def test_down_all(client, tmp_path):
    """`sky down --all -y` tears down every workspace cluster."""
    from skypilot_amd.cli.main import cli
    from skypilot_amd.client import sdk
    r = CliRunner()
    y = tmp_path / "b.yaml"
    y.write_text("run: 'true'\nresources: {cpus: 1}\n")
    for c in ("da-1", "da-2"):
        assert r.invoke(cli, ["launch", str(y), "-c", c,
                              "--detach-run"]).exit_code == 0
    res = r.invoke(cli, ["down", "--all", "-y"])
    assert res.exit_code == 0, res.output
    assert "da-1 terminated" in res.output
    assert "da-2 terminated" in res.output
    assert not sdk.get(sdk.status())


def test_status_name_filter(client, tmp_path):
    from skypilot_amd.cli.main import cli
    r = CliRunner()
    y = tmp_path / "c.yaml"
    y.write_text("run: 'true'\nresources: {cpus: 1}\n")
    for c in ("sf-1", "sf-2"):
        assert r.invoke(cli, ["launch", str(y), "-c", c,
                              "--detach-run"]).exit_code == 0
    res = r.invoke(cli, ["status", "sf-1"])
    assert res.exit_code == 0, res.output
    assert "sf-1" in res.output and "sf-2" not in res.output
    r.invoke(cli, ["down", "--all", "-y"])
