"""Durable spawned calls: detached work survives scheduler restarts
(SURVEY hard part 5 — the scheduler-death half of exactly-once-ish
accounting; worker death was covered in round 1)."""

from __future__ import annotations

import os
import time

import modal_amd as modal
from modal_amd._sync import synchronizer
from modal_amd.client import _Client
from modal_amd.scheduler.core import Scheduler


def _boot(run_dir, workers=0):
    async def make():
        s = Scheduler(run_dir=run_dir)
        await s.start()
        c = _Client(s, "client")
        _Client.set_default(c)
        for _ in range(workers):
            await s.pool.spawn_worker(gpu_index=None)
        return s, c

    return synchronizer.run(make())


def test_spawned_calls_survive_scheduler_restart(run_dir):
    # scheduler A: deploy an app, spawn work, go down before it can run
    s1, c1 = _boot(run_dir, workers=0)
    app = modal.App("wal-app")

    @app.function()
    def triple(x):
        return x * 3

    app.deploy(name="wal-app", client=c1)
    call_ids = [triple.spawn(i).object_id for i in range(5)]
    assert all(cid.startswith("fc-") for cid in call_ids)
    wal_files = os.listdir(os.path.join(run_dir, "wal"))
    assert len(wal_files) == 5, wal_files

    # hard stop: no graceful drain (the pool never even had workers)
    synchronizer.run(c1.close())
    _Client._singleton = None

    # scheduler B on the SAME run_dir: calls replay and execute
    s2, c2 = _boot(run_dir, workers=2)
    restored = [cid for cid in call_ids if cid in s2.calls]
    assert sorted(restored) == sorted(call_ids), "WAL did not restore all calls"
    for i, cid in enumerate(call_ids):
        fc = modal.FunctionCall.from_id(cid, client=c2)
        assert fc.get(timeout=60) == i * 3
    # journals drop once complete
    deadline = time.time() + 10
    while os.listdir(os.path.join(run_dir, "wal")) and time.time() < deadline:
        time.sleep(0.1)
    assert os.listdir(os.path.join(run_dir, "wal")) == []
    synchronizer.run(c2.close())


def test_partial_completion_no_reexecution(run_dir):
    """A crash mid-call: inputs that finished keep their journaled results
    (executed exactly once); blocked inputs re-run after restart."""
    counter_dir = os.path.join(run_dir, "exec-markers")
    os.makedirs(counter_dir, exist_ok=True)
    gate = os.path.join(run_dir, "unblock")
    os.environ["MARKER_DIR"] = counter_dir  # before workers spawn (inherited)
    os.environ["WAL_GATE"] = gate
    s1, c1 = _boot(run_dir, workers=1)
    app = modal.App("wal-app2")

    @app.function()
    def mark_fast(x):
        import os as _os

        open(_os.path.join(_os.environ["MARKER_DIR"], f"ran-{x}-{_os.getpid()}"), "a").write("1")
        return x + 100

    @app.function()
    def mark_gated(x):
        import os as _os
        import time as _t

        open(_os.path.join(_os.environ["MARKER_DIR"], f"ran-{x}-{_os.getpid()}"), "a").write("1")
        deadline = _t.time() + 120
        while not _os.path.exists(_os.environ["WAL_GATE"]) and _t.time() < deadline:
            _t.sleep(0.05)
        return x + 100

    app.deploy(name="wal-app2", client=c1)
    fcs = [mark_fast.spawn(0), mark_fast.spawn(1)]
    assert fcs[0].get(timeout=60) == 100
    assert fcs[1].get(timeout=60) == 101
    fcs += [mark_gated.spawn(2), mark_gated.spawn(3)]
    import time as _time

    _time.sleep(0.5)  # let the gated spawns journal + start
    # 2 and 3 are blocked on the gate; crash now
    synchronizer.run(c1.close())
    _Client._singleton = None

    open(gate, "w").write("go")  # unblock post-restart executions
    s2, c2 = _boot(run_dir, workers=2)
    for i in (2, 3):
        fc = modal.FunctionCall.from_id(fcs[i].object_id, client=c2)
        assert fc.get(timeout=60) == i + 100
    # completed-and-dropped calls are gone from the new scheduler (their
    # results were already delivered); each input executed exactly once
    # pre-crash, blocked ones may re-run (at-least-once on crash, parity
    # with the reference's INTERNAL_FAILURE redelivery semantics)
    ran0 = [f for f in os.listdir(counter_dir) if f.startswith("ran-0-")]
    ran1 = [f for f in os.listdir(counter_dir) if f.startswith("ran-1-")]
    assert len(ran0) == 1 and len(ran1) == 1
    synchronizer.run(c2.close())


def test_ephemeral_calls_not_journaled(client, run_dir):
    """Ephemeral apps' calls die with their client (reference semantics)."""
    app = modal.App("eph")

    @app.function()
    def f(x):
        return x

    with app.run(client=client):
        fc = f.spawn(1)
        assert fc.get(timeout=30) == 1
    wal = os.path.join(run_dir, "wal")
    assert not os.path.isdir(wal) or os.listdir(wal) == []
