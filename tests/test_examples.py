"""The shipped examples run end-to-end via the CLI (CPU-compatible ones)."""

from __future__ import annotations

import pytest
from click.testing import CliRunner

from modal_amd.cli.entry_point import entrypoint_cli


@pytest.fixture()
def runner():
    return CliRunner()


def test_example_hello(runner, client):
    result = runner.invoke(entrypoint_cli, ["run", "examples/hello.py::app.main", "--n=5"])
    assert result.exit_code == 0, result.output
    assert "one call: 49" in result.output
    assert "fan-out: 30" in result.output


def test_example_queue_pipeline(runner, client):
    result = runner.invoke(
        entrypoint_cli, ["run", "examples/queue_pipeline.py::app.main", "--n=50"]
    )
    assert result.exit_code == 0, result.output
    assert "sum of squares: 40425" in result.output


def test_example_sandbox_volume(runner, client):
    result = runner.invoke(entrypoint_cli, ["run", "examples/sandbox_volume.py::app.main"])
    assert result.exit_code == 0, result.output
    assert "hello from the volume" in result.output


def test_example_web_endpoint(runner, client):
    result = runner.invoke(entrypoint_cli, ["run", "examples/web_endpoint.py::app.main"])
    assert result.exit_code == 0, result.output
    assert "'hello': 'MI355X'" in result.output or '"hello":"MI355X"' in result.output


@pytest.mark.timeout(300)
def test_example_train_ddp(runner, client):
    """DDP training example: 2-rank gang on CPU (gloo), params stay in
    sync through bucketed gradient all-reduce."""
    import os

    os.environ["MODAL_AMD_FORCE_CPU"] = "1"
    try:
        result = runner.invoke(
            entrypoint_cli, ["run", "examples/train_ddp.py::app.main", "--steps=4"]
        )
    finally:
        os.environ.pop("MODAL_AMD_FORCE_CPU", None)
    assert result.exit_code == 0, result.output
    assert "ranks in sync: True" in result.output


def test_example_http_server_class(runner, client):
    result = runner.invoke(
        entrypoint_cli, ["run", "examples/http_server_class.py::app.main"]
    )
    assert result.exit_code == 0, result.output
    assert "response: served by EchoServer" in result.output
    assert "via tunnel: served by EchoServer" in result.output
