"""End-to-end: App + @app.function + .remote/.spawn/.map on CPU workers.

Covers BASELINE.json config 1 (hello-world plumbing) behavior: serialization,
hydration, invocation FSM, exceptions, generators.
"""

from __future__ import annotations

import time

import pytest

import modal_amd as modal


def test_hello_remote(client):
    app = modal.App("test-hello")

    @app.function()
    def double(x):
        return x * 2

    with app.run(client=client):
        assert double.remote(21) == 42
        assert double.local(5) == 10


def test_remote_kwargs_and_objects(client):
    app = modal.App("test-kwargs")

    @app.function()
    def combine(a, b=1, *, c=2):
        return a + b + c

    with app.run(client=client):
        assert combine.remote(1) == 4
        assert combine.remote(1, 2, c=3) == 6
        assert combine.remote(1, b=10) == 13


def test_exception_propagates(client):
    app = modal.App("test-exc")

    @app.function()
    def boom():
        raise ValueError("kapow")

    with app.run(client=client):
        with pytest.raises(ValueError, match="kapow"):
            boom.remote()


def test_spawn_and_get(client):
    app = modal.App("test-spawn")

    @app.function()
    def slow_add(a, b):
        time.sleep(0.1)
        return a + b

    with app.run(client=client):
        fc = slow_add.spawn(3, 4)
        assert fc.get(timeout=30) == 7
        # cached second get
        assert fc.get() == 7


def test_map_ordered(client):
    app = modal.App("test-map")

    @app.function()
    def sq(x):
        return x * x

    with app.run(client=client):
        results = list(sq.map(range(50)))
        assert results == [x * x for x in range(50)]


def test_map_unordered_and_starmap(client):
    app = modal.App("test-map2")

    @app.function()
    def add(a, b):
        return a + b

    with app.run(client=client):
        out = sorted(add.starmap([(1, 2), (3, 4), (5, 6)]))
        assert out == [3, 7, 11]
        unordered = sorted(add.map(range(10), range(10), order_outputs=False))
        assert unordered == [2 * x for x in range(10)]


def test_map_return_exceptions(client):
    app = modal.App("test-map-exc")

    @app.function()
    def maybe_fail(x):
        if x == 3:
            raise RuntimeError("nope")
        return x

    with app.run(client=client):
        results = list(maybe_fail.map(range(5), return_exceptions=True))
        assert results[3].__class__ is RuntimeError
        ok = [r for i, r in enumerate(results) if i != 3]
        assert ok == [0, 1, 2, 4]
        with pytest.raises(RuntimeError, match="nope"):
            list(maybe_fail.map(range(5)))


def test_generator_streaming(client):
    app = modal.App("test-gen")

    @app.function()
    def counter(n):
        for i in range(n):
            yield i * 10

    with app.run(client=client):
        items = list(counter.remote_gen(5))
        assert items == [0, 10, 20, 30, 40]


def test_for_each_and_spawn_map(client):
    app = modal.App("test-feach")

    @app.function()
    def noop(x):
        return x

    with app.run(client=client):
        noop.for_each(range(10))
        fc = noop.spawn_map(range(20))
        assert fc is not None


def test_async_api(client):
    import asyncio

    app = modal.App("test-aio")

    @app.function()
    def triple(x):
        return 3 * x

    async def main():
        async with app.run(client=client):
            r = await triple.remote.aio(4)
            outs = [o async for o in triple.map.aio(range(5))]
            return r, outs

    r, outs = asyncio.run(main())
    assert r == 12
    assert outs == [0, 3, 6, 9, 12]


def test_function_timeout(client):
    app = modal.App("test-timeout")

    @app.function(timeout=1)
    def sleepy():
        time.sleep(10)

    with app.run(client=client):
        t0 = time.time()
        with pytest.raises(modal.exception.FunctionTimeoutError):
            sleepy.remote()
        assert time.time() - t0 < 8


def test_retries_with_state_file(client, run_dir):
    app = modal.App("test-retries")
    marker = f"{run_dir}/attempts.txt"

    @app.function(retries=modal.Retries(max_retries=3, initial_delay=1.0))
    def flaky(path):
        import os

        n = int(open(path).read()) if os.path.exists(path) else 0
        with open(path, "w") as f:
            f.write(str(n + 1))
        if n < 2:
            raise RuntimeError("transient")
        return "ok"

    with app.run(client=client):
        assert flaky.remote(marker) == "ok"
        assert int(open(marker).read()) == 3


def test_map_large_payloads(client):
    """Per-item payloads in the MiB range spill chunks to the CAS."""
    import os as _os

    app = modal.App("test-big-map")

    @app.function()
    def digest(blob):
        import hashlib

        return hashlib.sha256(blob).hexdigest()

    blobs = [_os.urandom(1 << 20) for _ in range(6)]  # 6 x 1 MiB
    with app.run(client=client):
        import hashlib

        out = list(digest.map(blobs))
        assert out == [hashlib.sha256(b).hexdigest() for b in blobs]


def test_map_large_outputs(client):
    app = modal.App("test-big-out")

    @app.function()
    def inflate(n):
        return bytes([n % 256]) * (3 * 1024 * 1024)  # 3 MiB result

    with app.run(client=client):
        out = list(inflate.map(range(3)))
        for i, blob in enumerate(out):
            assert blob == bytes([i % 256]) * (3 * 1024 * 1024)


def test_nested_invocations_inside_worker(client):
    """Functions calling other functions (unary, map, remote_gen) from
    inside a worker — the thread-local blocking path end-to-end."""
    app = modal.App("nested-app")

    @app.function()
    def leaf(x):
        return x + 1

    @app.function()
    def streamer(n):
        for i in range(n):
            yield i * 10

    @app.function()
    def orchestrator(n):
        unary = leaf.remote(n)                      # nested unary
        mapped = sorted(leaf.map(range(3)))         # nested map
        streamed = list(streamer.remote_gen(3))     # nested generator
        return {"unary": unary, "mapped": mapped, "streamed": streamed}

    with app.run(client=client):
        out = orchestrator.remote(41)
    assert out == {"unary": 42, "mapped": [1, 2, 3], "streamed": [0, 10, 20]}


def test_async_function_uses_handles_aio(client):
    """Async user functions use .aio handle APIs on the worker's own
    loop (per-loop connection path)."""
    app = modal.App("async-handles-app")

    @app.function()
    def helper(x):
        return x * 3

    @app.function()
    async def async_orchestrator(n):
        d = modal.Dict.from_name("async-d", create_if_missing=True)
        await d.put.aio("k", n)
        v = await d.get.aio("k")
        r = await helper.remote.aio(v)
        total = 0
        async for out in helper.map.aio(range(3)):
            total += out
        return {"dict": v, "unary": r, "map_sum": total}

    with app.run(client=client):
        out = async_orchestrator.remote(7)
    assert out == {"dict": 7, "unary": 21, "map_sum": 0 + 3 + 6}


def test_nested_big_args_spill_to_cas(client):
    """A worker passing a >2 MiB argument to another function spills it
    through the shared CAS instead of crashing on the proxy client
    (client.blob_store must be a real store inside workers)."""
    import hashlib

    app = modal.App("bigarg-app")

    @app.function()
    def digest(blob: bytes) -> str:
        import hashlib as h

        return h.sha256(blob).hexdigest()

    @app.function()
    def orchestrate(n: int) -> str:
        import os as _os

        blob = _os.urandom(n)
        import hashlib as h

        assert digest.remote(blob) == h.sha256(blob).hexdigest()
        return "ok"

    with app.run(client=client):
        assert orchestrate.remote(3 * 1024 * 1024) == "ok"


def test_big_args_xfer_path(client, run_dir):
    """>2 MiB chunk payloads ride the one-shot file handoff (no hashing,
    no compression) and are unlinked when the chunk completes."""
    import hashlib
    import os

    app = modal.App("bigargs")

    @app.function()
    def digest(blob):
        import hashlib as h

        return h.sha256(blob).hexdigest()

    payloads = [os.urandom(1024 * 1024) for _ in range(12)]
    expected = sorted(hashlib.sha256(p).hexdigest() for p in payloads)
    with app.run(client=client):
        outs = sorted(digest.map(payloads, order_outputs=False))
    assert outs == expected
    xfer_dir = os.path.join(run_dir, "xfer")
    leftovers = os.listdir(xfer_dir) if os.path.isdir(xfer_dir) else []
    assert leftovers == [], f"xfer files leaked: {leftovers}"
