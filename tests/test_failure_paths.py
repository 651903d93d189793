"""Failure detection / recovery: worker death requeue, cancellation, caps.

Parity targets: INTERNAL_FAILURE requeue up to 8x without consuming user
retries (reference _functions.py:106), heartbeat/cancellation propagation
(container_io_manager.py:645-710), kill-switch semantics.
"""

from __future__ import annotations

import os
import time

import pytest

import modal_amd as modal
from modal_amd.functions import FunctionCallCancelledError


def test_worker_death_requeues_input(client, run_dir):
    """An input whose worker dies mid-flight reruns on another worker."""
    app = modal.App("death-app")
    marker = os.path.join(run_dir, "died-once")

    @app.function()
    def maybe_die(x, path):
        import os as _os

        if x == 3 and not _os.path.exists(path):
            with open(path, "w") as f:
                f.write("dying")
            _os._exit(1)  # hard-kill the whole worker process
        return x * 10

    with app.run(client=client):
        out = sorted(maybe_die.map(range(6), kwargs={"path": marker}, order_outputs=False))
        assert out == [0, 10, 20, 30, 40, 50]
        assert os.path.exists(marker), "the poisoned input never ran"


def test_repeated_internal_failure_finalizes(client, run_dir):
    """An input that always kills its worker eventually fails with
    InternalFailure after the 8-requeue cap (reference _functions.py:106)."""
    app = modal.App("death-cap-app")

    @app.function()
    def always_die():
        import os as _os

        _os._exit(1)

    from modal_amd.exception import InternalFailure
    import modal_amd.scheduler.calls as calls_mod

    # shrink the cap so the test is fast
    old_cap = calls_mod.MAX_INTERNAL_FAILURE_COUNT
    import modal_amd.scheduler.workerhost as wh
    import modal_amd.scheduler.core as core_mod

    wh.MAX_INTERNAL_FAILURE_COUNT = 2
    core_mod.MAX_INTERNAL_FAILURE_COUNT = 2
    try:
        with app.run(client=client):
            with pytest.raises(InternalFailure):
                always_die.remote()
    finally:
        wh.MAX_INTERNAL_FAILURE_COUNT = old_cap
        core_mod.MAX_INTERNAL_FAILURE_COUNT = old_cap


def test_cancel_function_call(client):
    app = modal.App("cancel-app")

    @app.function()
    def slow():
        time.sleep(30)
        return "done"

    with app.run(client=client):
        fc = slow.spawn()
        time.sleep(0.3)
        fc.cancel()
        with pytest.raises(FunctionCallCancelledError):
            fc.get(timeout=10)


def test_exception_has_clean_traceback(client):
    app = modal.App("tb-app")

    @app.function()
    def inner_fail():
        def deep():
            raise ValueError("deep failure")

        deep()

    with app.run(client=client):
        try:
            inner_fail.remote()
            pytest.fail("should have raised")
        except ValueError as exc:
            import traceback

            frames = traceback.extract_tb(exc.__traceback__)
            filenames = [f.filename for f in frames]
            # user frames survive the wire; asyncio/framework frames are cut
            assert any("test_failure_paths" in f for f in filenames)
            assert not any("asyncio" in f for f in filenames)


def test_map_with_failures_and_retries(client, run_dir):
    """User retry policy applies per input inside a map."""
    app = modal.App("map-retry-app")

    @app.function(retries=modal.Retries(max_retries=2, initial_delay=1.0))
    def flaky(x, base):
        import os as _os

        path = f"{base}/attempt-{x}"
        n = int(open(path).read()) if _os.path.exists(path) else 0
        with open(path, "w") as f:
            f.write(str(n + 1))
        if x % 3 == 0 and n == 0:
            raise RuntimeError(f"transient {x}")
        return x

    with app.run(client=client):
        out = sorted(flaky.map(range(6), kwargs={"base": run_dir}, order_outputs=False))
        assert out == list(range(6))


def test_cancel_with_terminate_recycles_worker(client):
    """terminate_containers=True kills the executing worker; the pool
    recovers for subsequent calls (parity: FunctionCall.cancel(
    terminate_containers=True))."""
    app = modal.App("cancel-term-app")

    @app.function()
    def hang():
        time.sleep(60)

    @app.function()
    def ping():
        return "alive"

    with app.run(client=client):
        fc = hang.spawn()
        time.sleep(0.5)
        fc.cancel(terminate_containers=True)
        with pytest.raises(FunctionCallCancelledError):
            fc.get(timeout=10)
        # pool recovers: a fresh worker serves the next call
        assert ping.remote() == "alive"


def test_function_call_from_id_inside_worker(client):
    """A FunctionCall handle passed between functions stays usable."""
    app = modal.App("fc-handoff")

    @app.function()
    def slow_value():
        time.sleep(0.3)
        return 99

    @app.function()
    def waiter(call_id: str):
        import modal_amd as modal

        fc = modal.FunctionCall.from_id(call_id)
        return fc.get(timeout=30) + 1

    with app.run(client=client):
        fc = slow_value.spawn()
        assert waiter.remote(fc.object_id) == 100


def test_map_early_break_cleans_up(client):
    """Breaking out of a map iterator mid-stream cancels the pump."""
    app = modal.App("early-break")

    @app.function()
    def ident(x):
        return x

    with app.run(client=client):
        seen = 0
        for _value in ident.map(range(10_000), order_outputs=False):
            seen += 1
            if seen >= 50:
                break
        assert seen == 50
        # the runtime stays healthy for the next call
        assert ident.remote(7) == 7


def test_chunked_map_mixed_failures_with_retries(client, run_dir):
    """Range-protocol stress: chunks with partial failures materialize
    per-item records that retry with the chunk payload still available."""
    app = modal.App("chunk-stress")

    @app.function(retries=modal.Retries(max_retries=2, initial_delay=1.0))
    def sometimes(x, base):
        import os as _os

        if x % 7 == 0:
            marker = f"{base}/m-{x}"
            if not _os.path.exists(marker):
                open(marker, "w").write("1")
                raise ValueError(f"first-attempt failure {x}")
        return x * 2

    with app.run(client=client):
        out = sorted(
            sometimes.map(range(200), kwargs={"base": run_dir}, order_outputs=False)
        )
        assert out == [x * 2 for x in range(200)]


def test_concurrent_maps_same_function(client):
    """Two maps over one function interleave without cross-talk."""
    import threading

    app = modal.App("concurrent-maps")

    @app.function()
    def tag(x, label):
        return f"{label}:{x}"

    results = {}

    with app.run(client=client):
        def run_map(label):
            results[label] = sorted(
                tag.map(range(300), kwargs={"label": label}, order_outputs=False)
            )

        t1 = threading.Thread(target=run_map, args=("a",))
        t2 = threading.Thread(target=run_map, args=("b",))
        t1.start(); t2.start(); t1.join(); t2.join()
        assert results["a"] == sorted(f"a:{x}" for x in range(300))
        assert results["b"] == sorted(f"b:{x}" for x in range(300))


def test_chunked_map_worker_death_requeues_chunk(client, run_dir):
    """A worker dying mid-chunk requeues the whole chunk elsewhere."""
    app = modal.App("chunk-death")
    marker = os.path.join(run_dir, "chunk-killed")

    @app.function()
    def kill_once(x, path):
        import os as _os

        if x == 70 and not _os.path.exists(path):
            open(path, "w").write("x")
            _os._exit(1)
        return x

    with app.run(client=client):
        out = sorted(kill_once.map(range(150), kwargs={"path": marker}, order_outputs=False))
        assert out == list(range(150))
        assert os.path.exists(marker)


def test_cancel_interrupts_sync_user_code(client):
    """Cancelling a running input stops pure-Python sync user code via
    async-exception injection WITHOUT killing the worker (SURVEY hard
    part 4: the thread-pool analog of SIGUSR1 cancellation)."""
    import time

    app = modal.App("cancel-sync-app")

    @app.function()
    def spinner():
        deadline = time.time() + 60
        x = 0
        while time.time() < deadline:  # pure-Python loop: injectable
            x += 1
        return x

    @app.function()
    def quick(v):
        return v * 2

    with app.run(client=client):
        fc = spinner.spawn()
        time.sleep(1.0)  # let it start spinning
        workers_before = {w.task_id for w in client.svc.pool.workers.values()}
        t0 = time.time()
        fc.cancel()
        with pytest.raises(Exception):
            fc.get(timeout=20)
        assert time.time() - t0 < 15, "cancel should interrupt the loop promptly"
        # the worker survived (no termination, no respawn needed)
        assert quick.remote(21) == 42
        workers_after = {w.task_id for w in client.svc.pool.workers.values()}
        assert workers_before & workers_after, "cancel should not kill workers"


def test_completed_call_records_are_garbage_collected(client):
    """Long-completed CallRecords drop after the retention window so a
    daemon does not grow unboundedly; recent calls survive for late
    .get()/gather."""
    import time

    app = modal.App("gc-app")

    @app.function()
    def f(x):
        return x

    with app.run(client=client):
        svc = client.svc
        assert f.remote(1) == 1
        assert sorted(f.map(range(5))) == list(range(5))
        n_before = len(svc.calls)
        assert n_before >= 2
        # age the finished records past retention, then run one GC pass
        for rec in svc.calls.values():
            if rec.finished_at is not None:
                rec.finished_at = time.time() - svc.CALL_RETENTION_SECONDS - 1
        dropped = svc._gc_calls_once()
        assert dropped >= 2
        # fresh calls still work and their records are retained
        fc = f.spawn(9)
        assert fc.get(timeout=30) == 9
        assert svc._gc_calls_once() == 0  # recent: kept


def test_map_spreads_across_workers(client):
    """Chunk dispatch load-balances across the pool (the 8-GPU scaling
    premise: no worker starves while others queue)."""
    from modal_amd._sync import synchronizer

    async def spawn_more():
        for _ in range(3):
            await client.svc.pool.spawn_worker()

    synchronizer.run(spawn_more())
    app = modal.App("spread-app")

    @app.function()
    def work(x):
        import os
        import time

        time.sleep(0.001)  # long enough that one worker cannot take all
        return os.getpid()

    with app.run(client=client):
        pids = set(work.map(range(2000)))
        assert len(pids) >= 3, f"work concentrated on {len(pids)} worker(s)"


def test_bigarg_xfer_survives_worker_death(client, run_dir):
    """A worker dying mid-chunk on the big-arg (xfer-spill) path: the chunk
    redelivers from the SAME spill file (it is unlinked only on chunk
    completion), payloads stay bit-exact, and no xfer files leak."""
    import hashlib

    app = modal.App("xfer-death")
    marker = os.path.join(run_dir, "xfer-killed")

    @app.function()
    def digest_or_die(blob, path):
        import hashlib as h
        import os as _os

        d = h.sha256(blob).hexdigest()
        # the odd-length payload kills its first worker mid-chunk
        if len(blob) % 2 == 1 and not _os.path.exists(path):
            open(path, "w").write("x")
            _os._exit(1)
        return d

    payloads = [os.urandom(1 << 20) for _ in range(24)]  # 24 x 1 MiB: spills
    payloads[10] = os.urandom((1 << 20) + 1)  # the deterministic killer
    expected = sorted(hashlib.sha256(p).hexdigest() for p in payloads)
    with app.run(client=client):
        out = sorted(
            digest_or_die.map(payloads, kwargs={"path": marker}, order_outputs=False)
        )
    assert out == expected
    assert os.path.exists(marker), "kill never triggered (fixture too lucky)"
    leftovers = os.listdir(os.path.join(run_dir, "xfer"))
    assert leftovers == [], f"xfer files leaked after worker death: {leftovers}"


def test_batched_chunk_worker_death_requeues(client, run_dir):
    """A worker killed mid-batched-chunk: the chunk group requeues and every
    item still arrives exactly once (the batched path shares the range
    protocol's redelivery)."""
    app = modal.App("batch-death")

    @app.function()
    @modal.batched(max_batch_size=16, wait_ms=1)
    def slowish(xs):
        import time as _t

        _t.sleep(0.02)
        return [x + 100 for x in xs]

    with app.run(client=client):
        import threading

        svc = client.svc

        def killer():
            time.sleep(0.15)
            for w in list(svc.pool.workers.values()):
                if w.alive and w.proc is not None:
                    try:
                        w.proc.kill()
                    except Exception:
                        pass
                    break

        t = threading.Thread(target=killer, daemon=True)
        t.start()
        got = sorted(slowish.map(range(400), order_outputs=False))
        assert got == [x + 100 for x in range(400)]
        t.join()
