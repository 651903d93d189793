from __future__ import annotations

import os
import shutil
import tempfile

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


@pytest.fixture()
def run_dir():
    path = tempfile.mkdtemp(prefix="modal-amd-test-")
    yield path
    shutil.rmtree(path, ignore_errors=True)


@pytest.fixture()
def client(run_dir):
    """A fresh in-process scheduler client, torn down with its worker pool."""
    from modal_amd._sync import synchronizer
    from modal_amd.client import _Client
    from modal_amd.scheduler.core import Scheduler

    async def make():
        scheduler = Scheduler(run_dir=run_dir)
        await scheduler.start()
        c = _Client(scheduler, "client")
        _Client.set_default(c)
        return c

    c = synchronizer.run(make())
    yield c
    synchronizer.run(c.close())
    _Client._singleton = None


@pytest.fixture(autouse=True)
def _reset_singleton():
    yield
    from modal_amd.client import _Client

    _Client._singleton = None
