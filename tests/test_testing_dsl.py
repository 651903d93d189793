"""The shipped interception DSL (parity: reference grpc_testing)."""

from __future__ import annotations

import pytest

import modal_amd as modal
from modal_amd.testing import intercept


def test_intercept_records_sequence(client):
    app = modal.App("intercept-app")

    @app.function()
    def f(x):
        return x + 1

    with intercept(client) as recorder:
        with app.run(client=client):
            assert f.remote(1) == 2
        seq = recorder.sequence()
        assert "app_create" in seq
        assert "function_create" in seq
        assert "function_map" in seq
        recorder.assert_called("function_create", times=1)
        assert seq.index("app_create") < seq.index("function_create")


def test_intercept_injects_failure(client):
    from modal_amd.exception import NotFoundError

    with intercept(client) as recorder:
        recorder.raise_on("queue_get_or_create", NotFoundError("injected"))
        with pytest.raises(NotFoundError, match="injected"):
            modal.Queue.from_name("whatever", create_if_missing=True).hydrate()


def test_intercept_canned_response(client):
    with intercept(client) as recorder:
        recorder.override("app_list", lambda **kw: [{"app_id": "ap-fake", "state": "running",
                                                     "description": "canned", "name": None,
                                                     "created_at": 0}])
        from modal_amd._sync import synchronizer

        rows = synchronizer.run(client.svc.app_list())
        assert rows[0]["app_id"] == "ap-fake"
