"""CLI surface tests (click runner; parity with reference cli commands)."""

from __future__ import annotations

import os
import textwrap

import pytest
from click.testing import CliRunner

from modal_amd.cli.entry_point import entrypoint_cli


@pytest.fixture()
def runner():
    return CliRunner()


@pytest.fixture()
def app_file(tmp_path):
    path = tmp_path / "myapp.py"
    path.write_text(
        textwrap.dedent(
            """
            import modal_amd as modal

            app = modal.App("cli-test-app")

            @app.function()
            def double(x: int):
                return x * 2

            @app.local_entrypoint()
            def main(n: int = 3):
                print("entrypoint says", double.remote(n))
            """
        )
    )
    return str(path)


def test_cli_help(runner):
    result = runner.invoke(entrypoint_cli, ["--help"])
    assert result.exit_code == 0
    for cmd in ["run", "deploy", "serve", "app", "volume", "queue", "dict", "secret", "config"]:
        assert cmd in result.output


def test_cli_run_function(runner, app_file, client):
    result = runner.invoke(entrypoint_cli, ["run", f"{app_file}::app.double", "21"])
    assert result.exit_code == 0, result.output
    assert "42" in result.output


def test_cli_run_entrypoint(runner, app_file, client):
    result = runner.invoke(entrypoint_cli, ["run", f"{app_file}::app.main", "--n=5"])
    assert result.exit_code == 0, result.output
    assert "entrypoint says 10" in result.output


def test_cli_deploy_and_app_list(runner, app_file, client):
    result = runner.invoke(entrypoint_cli, ["deploy", app_file, "--name", "cli-deployed"])
    assert result.exit_code == 0, result.output
    result = runner.invoke(entrypoint_cli, ["app", "list"])
    assert "cli-deployed" in result.output


def test_cli_queue_roundtrip(runner, client):
    assert runner.invoke(entrypoint_cli, ["queue", "create", "cliq"]).exit_code == 0
    import modal_amd as modal

    modal.Queue.from_name("cliq").put("hello")
    result = runner.invoke(entrypoint_cli, ["queue", "len", "cliq"])
    assert result.output.strip() == "1"
    result = runner.invoke(entrypoint_cli, ["queue", "peek", "cliq"])
    assert "hello" in result.output
    assert runner.invoke(entrypoint_cli, ["queue", "delete", "cliq", "--yes"]).exit_code == 0


def test_cli_volume_roundtrip(runner, client, tmp_path):
    src = tmp_path / "data.txt"
    src.write_text("cli-volume-data")
    assert runner.invoke(entrypoint_cli, ["volume", "create", "cliv"]).exit_code == 0
    res = runner.invoke(entrypoint_cli, ["volume", "put", "cliv", str(src), "/"])
    assert res.exit_code == 0, res.output
    res = runner.invoke(entrypoint_cli, ["volume", "ls", "cliv"])
    assert "data.txt" in res.output
    dest = tmp_path / "out.txt"
    res = runner.invoke(entrypoint_cli, ["volume", "get", "cliv", "data.txt", str(dest)])
    assert res.exit_code == 0, res.output
    assert dest.read_text() == "cli-volume-data"


def test_cli_secret_and_config(runner, client):
    res = runner.invoke(entrypoint_cli, ["secret", "create", "clis", "KEY=val"])
    assert res.exit_code == 0, res.output
    res = runner.invoke(entrypoint_cli, ["secret", "list"])
    assert "clis" in res.output
    res = runner.invoke(entrypoint_cli, ["config", "show"])
    assert res.exit_code == 0
    assert "heartbeat_interval" in res.output


def test_cli_dict_roundtrip(runner, client):
    assert runner.invoke(entrypoint_cli, ["dict", "create", "clid"]).exit_code == 0
    import modal_amd as modal

    modal.Dict.from_name("clid").put("k", [1, 2])
    res = runner.invoke(entrypoint_cli, ["dict", "get", "clid", "k"])
    assert "[1, 2]" in res.output


def test_cli_shell_piped(runner, client):
    result = runner.invoke(
        entrypoint_cli, ["shell", "--cmd", "/bin/bash"], input="echo piped-$((2+3))\n"
    )
    assert result.exit_code == 0, result.output
    assert "piped-5" in result.output


def test_cli_container_exec_and_stop(runner, client):
    """`container exec` runs in a worker's context; `container stop`
    drains it (parity: reference cli/container.py:297,318)."""
    import modal_amd as modal

    app = modal.App("cexec-app")

    @app.function()
    def noop():
        return 1

    with app.run(client=client):
        assert noop.remote() == 1
        workers = list(client.svc.pool.workers.values())
        assert workers
        task_id = workers[0].task_id
        result = runner.invoke(
            entrypoint_cli, ["container", "exec", task_id, "python3", "-c", "print(6*7)"]
        )
        assert result.exit_code == 0, result.output
        assert "42" in result.output
        result = runner.invoke(entrypoint_cli, ["container", "stop", task_id])
        assert result.exit_code == 0, result.output
        assert "Stopped" in result.output


def test_cli_endpoint_list(runner, client):
    import modal_amd as modal

    app = modal.App("ep-app")

    @app.function()
    @modal.fastapi_endpoint()
    def hello_ep():
        return {"ok": True}

    with app.run(client=client):
        result = runner.invoke(entrypoint_cli, ["endpoint", "list"])
        assert result.exit_code == 0, result.output
        assert "hello-ep" in result.output or "hello_ep" in result.output


def test_cli_app_logs(runner, client):
    """`app logs` replays worker stdout captured by the log plane."""
    import modal_amd as modal

    app = modal.App("logs-app")

    @app.function()
    def chatty():
        print("log-line-sentinel")
        return 1

    with app.run(client=client):
        assert chatty.remote() == 1
        import time

        time.sleep(0.3)  # log forwarding is async
        result = runner.invoke(entrypoint_cli, ["app", "logs", app.app_id])
    assert result.exit_code == 0, result.output
    assert "log-line-sentinel" in result.output


def test_cli_environment_and_profile(runner, client):
    result = runner.invoke(entrypoint_cli, ["environment", "list"])
    assert result.exit_code == 0, result.output
    result = runner.invoke(entrypoint_cli, ["profile", "current"])
    assert result.exit_code == 0, result.output
    result = runner.invoke(entrypoint_cli, ["config", "show"])
    assert result.exit_code == 0, result.output


def test_cli_curl_and_launch(runner, client):
    import modal_amd as modal

    app = modal.App("curl-app")

    @app.function()
    @modal.fastapi_endpoint()
    def hello_curl():
        return {"via": "curl"}

    with app.run(client=client):
        result = runner.invoke(entrypoint_cli, ["curl", hello_curl.web_url])
        assert result.exit_code == 0, result.output
        assert "curl" in result.output
    result = runner.invoke(entrypoint_cli, ["launch", "--help"])
    assert result.exit_code == 0


def test_cli_volume_cp_and_rename(client, tmp_path):
    import modal_amd as modal
    from modal_amd.cli.entry_point import entrypoint_cli

    from click.testing import CliRunner

    vol = modal.Volume.from_name("cpvol", create_if_missing=True)
    src = tmp_path / "a.txt"
    src.write_text("copy-me")
    with vol.batch_upload() as b:
        b.put_file(str(src), "/a.txt")
    runner = CliRunner()
    result = runner.invoke(entrypoint_cli, ["volume", "cp", "cpvol", "/a.txt", "/b.txt"])
    assert result.exit_code == 0, result.output
    assert b"copy-me" == b"".join(vol.read_file("b.txt"))
    result = runner.invoke(entrypoint_cli, ["volume", "rename", "cpvol", "cpvol2"])
    assert result.exit_code == 0, result.output
    vol2 = modal.Volume.from_name("cpvol2")
    assert b"copy-me" == b"".join(vol2.read_file("a.txt"))


def test_cli_nfs_roundtrip(client, tmp_path, monkeypatch):
    from click.testing import CliRunner

    from modal_amd.cli.entry_point import entrypoint_cli

    runner = CliRunner()
    local = tmp_path / "payload.txt"
    local.write_text("nfs-cli-data")
    r = runner.invoke(entrypoint_cli, ["nfs", "create", "cli-nfs"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(entrypoint_cli, ["nfs", "put", "cli-nfs", str(local), "p.txt"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(entrypoint_cli, ["nfs", "list"])
    assert "cli-nfs" in r.output
    out = tmp_path / "back.txt"
    r = runner.invoke(entrypoint_cli, ["nfs", "get", "cli-nfs", "p.txt", str(out)])
    assert r.exit_code == 0, r.output
    assert out.read_text() == "nfs-cli-data"
    r = runner.invoke(entrypoint_cli, ["nfs", "rm", "cli-nfs", "p.txt"])
    assert r.exit_code == 0, r.output


def test_cli_dashboard_and_workspace(client):
    from click.testing import CliRunner

    from modal_amd.cli.entry_point import entrypoint_cli

    runner = CliRunner()
    r = runner.invoke(entrypoint_cli, ["dashboard"])
    assert r.exit_code == 0 and "run dir:" in r.output
    r = runner.invoke(entrypoint_cli, ["workspace", "current"])
    assert r.exit_code == 0 and r.output.strip()
    r = runner.invoke(entrypoint_cli, ["changelog"])
    assert r.exit_code == 0
