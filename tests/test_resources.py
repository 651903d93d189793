"""Queue / Dict / Secret semantics (parity: reference queue_test, dict_test)."""

from __future__ import annotations

import threading
import time

import pytest

import modal_amd as modal
from modal_amd.exception import NotFoundError, QueueEmptyError


def test_queue_basic(client):
    with modal.Queue.ephemeral() as q:
        q.put(1)
        q.put_many([2, 3])
        assert q.len() == 3
        assert q.get() == 1
        assert q.get_many(2) == [2, 3]
        assert q.get(block=False) is None


def test_queue_partitions(client):
    with modal.Queue.ephemeral() as q:
        q.put(1)
        q.put(2, partition="other")
        assert q.len() == 1
        assert q.len(partition="other") == 1
        assert q.len(total=True) == 2
        assert q.get(partition="other") == 2


def test_queue_exceptions_are_stdlib_subclasses(client):
    """Parity: the reference raises stdlib queue.Empty/queue.Full; ported user
    code catching those must keep working (advisor finding, round 1)."""
    import queue as stdlib_queue

    with modal.Queue.ephemeral() as q:
        with pytest.raises(stdlib_queue.Empty):
            q.get(timeout=0.05)
        q.put_many(list(range(5000)))
        with pytest.raises(stdlib_queue.Full):
            q.put(5001, timeout=0.05)


def test_queue_blocking_timeout(client):
    with modal.Queue.ephemeral() as q:
        t0 = time.time()
        with pytest.raises(QueueEmptyError):
            q.get(timeout=0.2)
        assert 0.1 < time.time() - t0 < 5


def test_queue_producer_consumer_threads(client):
    with modal.Queue.ephemeral() as q:
        got = []

        def consumer():
            for _ in range(10):
                got.append(q.get(timeout=5))

        t = threading.Thread(target=consumer)
        t.start()
        for i in range(10):
            q.put(i)
        t.join()
        assert sorted(got) == list(range(10))


def test_queue_named_and_delete(client):
    q = modal.Queue.from_name("q-named", create_if_missing=True)
    q.put("x")
    q2 = modal.Queue.from_name("q-named")
    assert q2.get() == "x"
    modal.Queue.delete("q-named")
    with pytest.raises(NotFoundError):
        modal.Queue.from_name("q-named").hydrate()


def test_queue_iterate(client):
    with modal.Queue.ephemeral() as q:
        q.put_many([1, 2, 3])
        assert list(q.iterate()) == [1, 2, 3]
        assert q.len() == 3  # non-destructive


def test_dict_basic(client):
    with modal.Dict.ephemeral() as d:
        d.put("k", {"v": 1})
        assert d.get("k") == {"v": 1}
        assert d.get("missing", 5) == 5
        assert d.contains("k")
        assert d.len() == 1
        d.update({"a": 1, "b": 2})
        assert sorted([k for k in d.keys()]) == ["a", "b", "k"]
        assert d.pop("a") == 1
        with pytest.raises(KeyError):
            d.pop("a")
        d.clear()
        assert d.len() == 0


def test_dict_object_keys(client):
    with modal.Dict.ephemeral() as d:
        d.put((1, 2), "tuple-key")
        assert d.get((1, 2)) == "tuple-key"


def test_dict_skip_if_exists(client):
    with modal.Dict.ephemeral() as d:
        assert d.put("k", 1, skip_if_exists=True) is True
        assert d.put("k", 2, skip_if_exists=True) is False
        assert d.get("k") == 1


def test_secret_from_dict_and_name(client):
    s = modal.Secret.from_dict({"API_KEY": "abc"})
    s.hydrate()
    assert s.env() == {"API_KEY": "abc"}
    modal.Secret.create_deployed("my-secret", {"TOK": "t1"})
    s2 = modal.Secret.from_name("my-secret")
    s2.hydrate()
    assert s2.env() == {"TOK": "t1"}
    with pytest.raises(NotFoundError):
        modal.Secret.from_name("nope").hydrate()


def test_secret_reaches_worker_env(client):
    app = modal.App("test-secret-env")

    @app.function(secrets=[modal.Secret.from_dict({"MY_TEST_VAR": "hello-worker"})])
    def read_env():
        import os

        return os.environ.get("MY_TEST_VAR")

    with app.run(client=client):
        assert read_env.remote() == "hello-worker"


def test_queue_used_inside_worker(client):
    """Handles serialize as ids and rebind to the worker's scheduler proxy."""
    app = modal.App("test-q-worker")

    @app.function()
    def pusher(q, n):
        for i in range(n):
            q.put(i * 2)
        return q.len()

    with app.run(client=client):
        with modal.Queue.ephemeral() as q:
            count = pusher.remote(q, 5)
            assert count == 5
            assert [q.get() for _ in range(5)] == [0, 2, 4, 6, 8]


def test_aio_variants_for_resources(client):
    """Every resource method carries a working .aio twin."""
    import asyncio

    async def main():
        q = await modal.Queue.from_name("aio-q", create_if_missing=True).hydrate.aio()
        await q.put.aio("via-aio")
        assert await q.get.aio() == "via-aio"
        d = await modal.Dict.from_name("aio-d", create_if_missing=True).hydrate.aio()
        await d.put.aio("k", 1)
        assert await d.get.aio("k") == 1
        vol = await modal.Volume.from_name("aio-v", create_if_missing=True).hydrate.aio()
        entries = await vol.listdir.aio("/")
        assert entries == []

    asyncio.run(main())


def test_update_autoscaler_and_stats(client):
    """update_autoscaler / keep_warm adjust the live FunctionDef;
    get_current_stats reports backlog+runners (parity: reference
    _functions.py:1195-1292, :2021)."""
    import modal_amd as modal

    app = modal.App("scaler-app")

    @app.function()
    def f(x):
        return x

    with app.run(client=client):
        assert f.remote(1) == 1
        f.update_autoscaler(min_containers=2, max_containers=5, buffer_containers=1,
                            scaledown_window=120)
        fdef = client.svc.functions[f.object_id]
        assert (fdef.min_containers, fdef.max_containers) == (2, 5)
        assert fdef.buffer_containers == 1 and fdef.scaledown_window == 120
        f.keep_warm(3)
        assert client.svc.functions[f.object_id].min_containers == 3
        stats = f.get_current_stats()
        assert stats["backlog"] == 0
        assert "num_total_tasks" in stats
