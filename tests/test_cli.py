"""CLI end-to-end against a live server (echo engine): every verb."""

import os

import pytest
from click.testing import CliRunner

from agentainer_amd.cli import cli

from test_crash_integration import Server, _free_port


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    root = str(tmp_path_factory.mktemp("cli-root"))
    srv = Server(root, _free_port())
    srv.start()
    yield srv
    srv.terminate()


@pytest.fixture()
def run(server):
    runner = CliRunner()

    def _run(*args, expect=0):
        res = runner.invoke(cli, ["--url", server.base, *args],
                            catch_exceptions=False)
        assert res.exit_code == expect, res.output
        return res.output

    return _run


def test_cli_full_surface(run):
    out = run("deploy", "echo", "--name", "cli-agent", "--auto-restart",
              "--system-prompt", "hi")
    aid = out.strip().split("-> ")[-1]
    assert aid.startswith("agent-")

    assert "cli-agent" in run("list")
    run("start", aid)
    assert "running" in run("list")

    out = run("invoke", aid, "-m", "ping from cli")
    assert "ping from cli" in out

    out = run("invoke", aid, "-m", "streamed ping", "--stream")
    assert "streamed ping" in out

    run("pause", aid)
    run("resume", aid)
    run("restart", aid)

    out = run("requests", aid)
    assert "completed" in out

    out = run("health", aid)
    assert "healthy" in out
    run("health")

    run("metrics", aid)
    run("metrics", aid, "--history")

    out = run("audit", "--action", "deploy")
    assert "deploy" in out
    run("logs", aid)

    out = run("backup", "create", "--name", "clisnap")
    assert "id" in out
    out = run("backup", "list")
    assert "clisnap" in out
    bid = [line.split()[0] for line in out.splitlines()
           if line.startswith("backup-")][0]
    out = run("backup", "restore", bid)
    assert "-restored" in out
    run("backup", "delete", bid)

    run("stop", aid)
    run("remove", aid)
    assert aid not in run("list")


def test_cli_yaml_deploy(run, tmp_path):
    p = tmp_path / "fleet.yaml"
    p.write_text(
        "kind: AgentDeployment\n"
        "spec:\n"
        "  agents:\n"
        "    - name: yam\n"
        "      model: echo\n"
        "      replicas: 2\n")
    out = run("deploy", "--config", str(p))
    assert "yam-1" in out and "yam-2" in out


def test_cli_unreachable_server():
    runner = CliRunner()
    res = runner.invoke(cli, ["--url", "http://127.0.0.1:9", "list"])
    assert res.exit_code == 2


def test_cli_backup_export_import(run, server, tmp_path):
    run("deploy", "echo", "--name", "exp-agent")
    out = run("backup", "create", "--name", "expsnap")
    import json as _json
    bid = _json.loads(out)["id"]
    bundle = str(tmp_path / "b.tar.gz")
    out = run("backup", "export", bid, "-o", bundle)
    assert "exported" in out
    run("backup", "delete", bid)
    out = run("backup", "import", bundle)
    assert bid in out
    assert bid in run("backup", "list")


def test_metrics_device_cli(run):
    """`agentainer metrics --device` hits /metrics/device (empty JSON on
    the echo engine — populated by the 10s device sampler on GPU)."""
    import json as _json

    out = run("metrics", "--device")
    assert isinstance(_json.loads(out), dict)


def test_metrics_requires_agent_or_device(run):
    try:
        run("metrics", expect=2)  # click usage error
    except AssertionError:
        raise
