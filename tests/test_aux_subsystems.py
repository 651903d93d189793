"""Schedules, autoscaler logic, telemetry, snapshot state machine, flash."""

from __future__ import annotations

import json
import os
import socket
import struct
import threading
import time
from datetime import datetime, timezone

import pytest

import modal_amd as modal


def test_cron_matcher():
    from modal_amd.scheduler.cron import cron_matches

    dt = datetime(2026, 9, 13, 14, 30, tzinfo=timezone.utc)  # Sunday
    assert cron_matches("30 14 * * *", dt)
    assert cron_matches("*/15 * * * *", dt)
    assert not cron_matches("31 14 * * *", dt)
    assert cron_matches("30 14 13 9 *", dt)
    assert not cron_matches("30 14 14 9 *", dt)
    assert cron_matches("* * * * 0", dt)  # sunday == 0
    assert not cron_matches("* * * * 1", dt)


def test_period_schedule_fires(client, run_dir):
    app = modal.App("sched-app")
    marker = f"{run_dir}/fired"

    @app.function(schedule=modal.Period(seconds=0.5))
    def tick():
        with open(marker, "a") as f:
            f.write("x")

    app.deploy(name="sched-deployed", client=client)
    deadline = time.time() + 15
    while time.time() < deadline:
        if os.path.exists(marker) and len(open(marker).read()) >= 1:
            break
        time.sleep(0.2)
    else:
        pytest.fail("schedule never fired")


def test_flash_autoscaler_decisions():
    from modal_amd.flash import FlashAutoscaler

    metric = {"v": 100.0}
    scaler = FlashAutoscaler(
        lambda: metric["v"], target_value=50.0, min_replicas=1, max_replicas=8,
        scale_up_stabilization=0.0, scale_down_stabilization=10.0,
    )
    # metric 2x target -> double replicas, immediately (no up window)
    assert scaler.decide(2, now=100.0) == 4
    # metric at target -> hold
    metric["v"] = 50.0
    assert scaler.decide(4, now=101.0) == 4
    # metric far below target -> scale down only after the window
    metric["v"] = 10.0
    assert scaler.decide(4, now=102.0) == 4  # pending
    assert scaler.decide(4, now=105.0) == 4  # still inside window
    assert scaler.decide(4, now=113.0) == 1  # window elapsed


def test_flash_manager_registry():
    from modal_amd.flash import FlashManager

    mgr = FlashManager()
    mgr.register("svc", "http://127.0.0.1:9000")
    assert mgr.list()[0].name == "svc"
    mgr.deregister("svc")
    assert mgr.list() == []


def test_import_telemetry_socket(tmp_path):
    from modal_amd.runtime.telemetry import instrument_imports, uninstrument_imports

    sock_path = str(tmp_path / "telemetry.sock")
    received = []
    server = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    server.bind(sock_path)
    server.listen(1)

    def acceptor():
        conn, _ = server.accept()
        buf = b""
        while True:
            try:
                data = conn.recv(4096)
            except OSError:
                return
            if not data:
                return
            buf += data
            while len(buf) >= 4:
                (n,) = struct.unpack("<I", buf[:4])
                if len(buf) < 4 + n:
                    break
                received.append(json.loads(buf[4 : 4 + n]))
                buf = buf[4 + n :]

    t = threading.Thread(target=acceptor, daemon=True)
    t.start()
    try:
        instrument_imports(sock_path)
        import wsgiref.handlers  # noqa: F401  - something not yet imported

        deadline = time.time() + 5
        while time.time() < deadline and not any(
            "wsgiref" in m.get("name", "") for m in received
        ):
            time.sleep(0.05)
        assert any("wsgiref" in m.get("name", "") for m in received)
        assert all(m["event"] == "module_load" for m in received)
    finally:
        uninstrument_imports()
        server.close()


def test_snapshot_state_machine_cpu():
    from modal_amd.runtime.gpu_snapshot import CudaCheckpointState, GPUMemorySnapshot

    snap = GPUMemorySnapshot()
    assert snap.state is CudaCheckpointState.RUNNING
    snap.checkpoint()  # no GPU here: transitions straight to CHECKPOINTED
    assert snap.state is CudaCheckpointState.CHECKPOINTED
    snap.restore()
    assert snap.state is CudaCheckpointState.RUNNING
    with pytest.raises(RuntimeError):
        snap.restore()


@pytest.mark.gpu
def test_snapshot_roundtrip_gpu():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from modal_amd.runtime.gpu_snapshot import GPUMemorySnapshot

    t1 = torch.arange(1000, device="cuda", dtype=torch.float32)
    t2 = torch.full((64, 64), 3.0, device="cuda")
    expect1, expect2 = t1.cpu().clone(), t2.cpu().clone()
    snap = GPUMemorySnapshot()
    snap.checkpoint()
    assert t1.device.type == "cpu" and t1.numel() == 0  # paged out
    snap.restore()
    assert t1.is_cuda and t2.is_cuda
    assert torch.equal(t1.cpu(), expect1)
    assert torch.equal(t2.cpu(), expect2)


def test_autoscaler_spawns_for_backlog(client):
    """Backlogged CPU function grows the pool beyond one worker."""
    app = modal.App("scale-app")

    @app.function()
    def slow(x):
        time.sleep(0.4)
        return x

    with app.run(client=client):
        out = sorted(slow.map(range(24), order_outputs=False))
        assert out == list(range(24))
        svc = client.svc
        assert len(svc.pool.workers) >= 2, "autoscaler never scaled up"


def test_environments_and_workspace():
    import modal_amd.environments as envs
    from modal_amd.workspace import Workspace

    envs.create_environment("staging")
    names = [e.name for e in envs.list_environments()]
    assert "staging" in names and "main" in names
    envs.delete_environment("staging")
    assert Workspace.current().name


def test_scheduler_placement_steers_dispatch(run_dir, monkeypatch):
    """SchedulerPlacement(gpu_index=N) pins work to that GPU's worker."""
    monkeypatch.setenv("MODAL_AMD_FAKE_GPUS", "3")
    monkeypatch.setenv("MODAL_AMD_WORKER_COUNT", "3")
    from modal_amd._sync import synchronizer
    from modal_amd.client import _Client
    from modal_amd.scheduler.core import Scheduler

    async def make():
        s = Scheduler(run_dir=run_dir)
        await s.start()
        c = _Client(s, "client")
        _Client.set_default(c)
        return s, c

    s, c = synchronizer.run(make())
    try:
        app = modal.App("placement-app")

        @app.function(placement=modal.SchedulerPlacement(gpu_index=2))
        def where():
            import os as _os

            return _os.environ.get("HIP_VISIBLE_DEVICES")

        with app.run(client=c):
            results = {where.remote() for _ in range(4)}
            assert results == {"2"}, results
    finally:
        synchronizer.run(c.close())
        _Client._singleton = None


def test_node_metrics_exposition(client):
    """scheduler-level items/sec + p50/p99 + worker gauges in
    Prometheus text format, over RPC and over the web gateway."""
    import urllib.request

    import modal_amd as modal
    from modal_amd._sync import synchronizer

    app = modal.App("metrics-app")

    @app.function()
    def ident(x):
        return x

    @app.function()
    @modal.fastapi_endpoint()
    def tiny():
        return {"ok": True}

    with app.run(client=client):
        assert ident.remote(5) == 5
        assert sorted(ident.map(range(40))) == list(range(40))
        text = synchronizer.run(client.svc.node_metrics())
        assert "modal_amd_inputs_total" in text
        counters = {
            line.split()[0]: float(line.split()[1])
            for line in text.splitlines()
            if line and not line.startswith("#") and "{" not in line
        }
        assert counters["modal_amd_inputs_total"] >= 41
        assert counters["modal_amd_outputs_total"] >= 41
        assert counters["modal_amd_failures_total"] == 0
        assert "modal_amd_unary_latency_seconds" in text  # p50/p99 present
        # gateway scrape target
        base = tiny.web_url.rsplit("/", 1)[0]
        with urllib.request.urlopen(base + "/_metrics", timeout=10) as resp:
            scraped = resp.read().decode()
        assert "modal_amd_workers" in scraped


def test_memory_snapshot_scaledown_pageout(client):
    """enable_memory_snapshot functions page GPU state to host on
    scaledown instead of losing the warm worker; dispatch restores
    (parity: memory-snapshot cold-start elimination)."""
    import time as _time

    import modal_amd as modal
    from modal_amd._sync import synchronizer

    app = modal.App("snap-scale-app")

    @app.function(enable_memory_snapshot=True, scaledown_window=0.1)
    def bump(x):
        return x + 1

    with app.run(client=client):
        assert bump.remote(1) == 2
        _time.sleep(0.3)  # idle past the window
        pool = client.svc.pool
        synchronizer.run(pool._scaledown_once())
        paged = [w for w in pool.workers.values() if w.paged]
        assert paged, "idle snapshot-enabled worker should page out, not die"
        assert all(w.alive for w in paged)
        # next dispatch restores transparently
        assert bump.remote(41) == 42
        assert not any(w.paged for w in pool.workers.values())


def test_enter_snap_hook_ordering(client):
    """@modal.enter(snap=True) hooks run before plain @modal.enter hooks
    (pre-snapshot vs post-restore; parity: _partial_function.py:589)."""
    import modal_amd as modal

    app = modal.App("snap-order-app")

    @app.cls()
    class Svc:
        @modal.enter(snap=True)
        def pre(self):
            self.order = ["pre"]

        @modal.enter()
        def post(self):
            self.order.append("post")

        @modal.method()
        def get_order(self):
            return self.order

    with app.run(client=client):
        assert Svc().get_order.remote() == ["pre", "post"]


@pytest.mark.gpu
def test_memory_snapshot_worker_gpu_roundtrip(client):
    """Page-out/restore through the scheduler keeps a worker's resident
    CUDA tensors bit-intact (framework-level cuda-checkpoint analog)."""
    import time as _time

    import modal_amd as modal
    from modal_amd._sync import synchronizer

    app = modal.App("snap-gpu-app")

    @app.function(gpu=1, enable_memory_snapshot=True, scaledown_window=0.1)
    def total():
        import torch

        t = getattr(torch, "_snap_cache", None)
        if t is None:
            t = torch.arange(4096, device="cuda", dtype=torch.float32)
            torch._snap_cache = t
        return float(t.sum().item())

    expect = 4095.0 * 4096.0 / 2.0
    with app.run(client=client):
        assert total.remote() == expect
        _time.sleep(0.3)
        pool = client.svc.pool
        synchronizer.run(pool._scaledown_once())
        assert any(w.paged for w in pool.workers.values())
        assert total.remote() == expect  # restored in place
        assert not any(w.paged for w in pool.workers.values())
