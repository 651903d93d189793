from __future__ import annotations

import pytest

import modal_amd as modal
from modal_amd._sync import synchronizer


def _make_app(value):
    app = modal.App("versioned")

    @app.function(name="get_value")
    def get_value():
        return value

    return app


def test_deploy_history_and_rollback(client):
    _make_app("v1").deploy(name="versioned", client=client)
    _make_app("v2").deploy(name="versioned", client=client)

    fn = modal.Function.from_name("versioned", "get_value")
    assert fn.remote() == "v2"

    history = synchronizer.run(client.svc.app_history(name="versioned"))
    assert [h["version"] for h in history] == [1, 2]

    synchronizer.run(client.svc.app_rollback(name="versioned", version=0))
    fn1 = modal.Function.from_name("versioned", "get_value")
    assert fn1.remote() == "v1"


def test_billing_summary(client):
    app = modal.App("billed")

    @app.function()
    def spin():
        return 1

    with app.run(client=client):
        for _ in range(3):
            spin.remote()
    from modal_amd.billing import usage_summary

    rows = usage_summary()
    row = next(r for r in rows if r["function"] == "spin")
    assert row["inputs"] == 3
    assert row["runtime_seconds"] >= 0
