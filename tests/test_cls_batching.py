"""Class services, @batched dynamic batching, @concurrent input slots."""

from __future__ import annotations

import time

import pytest

import modal_amd as modal


def test_cls_method_remote(client):
    app = modal.App("cls-app")

    @app.cls()
    class Counter:
        base = modal.parameter(default=100)

        @modal.enter()
        def setup(self):
            self.offset = 7

        @modal.method()
        def add(self, x):
            return self.base + self.offset + x

    with app.run(client=client):
        c = Counter()
        assert c.add.remote(1) == 108


def test_cls_parametrized(client):
    app = modal.App("cls-param")

    @app.cls()
    class Scaler:
        factor = modal.parameter(default=2)

        @modal.method()
        def scale(self, x):
            return x * self.factor

    with app.run(client=client):
        assert Scaler().scale.remote(10) == 20
        assert Scaler(factor=5).scale.remote(10) == 50


def test_cls_lifecycle_exit_hook(client, run_dir):
    app = modal.App("cls-exit")
    marker = f"{run_dir}/exit-marker"

    @app.cls()
    class Svc:
        @modal.enter()
        def up(self):
            self.path = None

        @modal.method()
        def ping(self, path):
            self.path = path
            type(self)._path = path
            return "pong"

        @modal.exit()
        def down(self):
            with open(type(self)._path, "w") as f:
                f.write("exited")

    with app.run(client=client):
        assert Svc().ping.remote(marker) == "pong"
    # worker shutdown runs @exit hooks; give the pool a moment
    deadline = time.time() + 10
    while time.time() < deadline:
        try:
            assert open(marker).read() == "exited"
            break
        except FileNotFoundError:
            time.sleep(0.2)
    else:
        pytest.fail("exit hook never ran")


def test_cls_generator_method(client):
    app = modal.App("cls-gen")

    @app.cls()
    class Gen:
        @modal.method()
        def items(self, n):
            for i in range(n):
                yield i * 2

    with app.run(client=client):
        assert list(Gen().items.remote_gen(3)) == [0, 2, 4]


def test_batched_function(client):
    app = modal.App("batched-app")

    @app.function()
    @modal.batched(max_batch_size=4, wait_ms=300)
    def batch_double(xs):
        # xs arrives as a list; return a list of the same length
        assert isinstance(xs, list)
        return [x * 2 for x in xs]

    with app.run(client=client):
        results = list(batch_double.map(range(10)))
        assert results == [x * 2 for x in range(10)]


def test_batched_error_propagates_per_item(client):
    app = modal.App("batched-err")

    @app.function()
    @modal.batched(max_batch_size=8, wait_ms=100)
    def bad_batch(xs):
        raise ValueError("batch failed")

    with app.run(client=client):
        out = list(bad_batch.map(range(4), return_exceptions=True))
        assert all(isinstance(o, ValueError) for o in out)


def test_concurrent_overlap(client):
    app = modal.App("concurrent-app")

    @app.function()
    @modal.concurrent(max_inputs=8)
    def sleeper(x):
        time.sleep(0.3)
        return x

    with app.run(client=client):
        t0 = time.time()
        out = sorted(sleeper.map(range(8), order_outputs=False))
        elapsed = time.time() - t0
        assert out == list(range(8))
        # 8 x 0.3 s sequential would be 2.4 s+; concurrency should crush that
        assert elapsed < 1.8, f"no overlap: {elapsed:.2f}s"


def test_cls_with_options_and_from_name(client):
    app = modal.App("cls-deploy")

    @app.cls()
    class Echo:
        @modal.method()
        def say(self, v):
            return f"echo:{v}"

    app.deploy(name="cls-deployed", client=client)
    remote_cls = modal.Cls.from_name("cls-deployed", "Echo")
    obj = remote_cls()
    assert obj.say.remote("hi") == "echo:hi"


def test_batched_actually_groups(client):
    """The runtime accumulates multiple logical calls into one execution."""
    app = modal.App("batch-grouping")

    @app.function()
    @modal.batched(max_batch_size=32, wait_ms=150)
    def sizes(xs):
        return [len(xs)] * len(xs)

    with app.run(client=client):
        out = list(sizes.map(range(64), order_outputs=False))
        assert len(out) == 64
        assert max(out) > 1, "no batching happened"


def test_app_server_lifecycle(client):
    """@app.server: @enter starts an HTTP process; start() probes the port
    inside the worker; the URL serves; @exit tears the process down
    (round-1 review: Server was a bare Cls alias)."""
    import urllib.request

    import modal_amd as modal

    app = modal.App("server-app")

    @app.server(port=18431, startup_timeout=20)
    class Httpd:
        @modal.enter()
        def boot(self):
            import subprocess
            import sys

            self.proc = subprocess.Popen(
                [sys.executable, "-c",
                 "import http.server;"
                 "h=type('H',(http.server.BaseHTTPRequestHandler,),"
                 "{'do_GET':lambda s:(s.send_response(200),s.end_headers(),"
                 "s.wfile.write(b'served-by-modal-amd')),"
                 "'log_message':lambda s,*a:None});"
                 "http.server.HTTPServer(('127.0.0.1',18431),h).serve_forever()"],
            )

        @modal.exit()
        def shutdown(self):
            self.proc.terminate()

    with app.run(client=client):
        server = Httpd.start()
        assert server.url == "http://127.0.0.1:18431"
        with urllib.request.urlopen(server.url, timeout=10) as resp:
            assert resp.read() == b"served-by-modal-amd"
        server.stop()


def test_server_config_validation():
    import modal_amd as modal
    from modal_amd.exception import InvalidError

    app = modal.App("server-val")
    import pytest as _pytest

    with _pytest.raises(InvalidError):
        @app.server(port=0)
        class Bad1:
            pass
    with _pytest.raises(InvalidError):
        @app.server(port=80, startup_timeout=0)
        class Bad2:
            pass


def test_batched_map_rides_chunks(client):
    """@modal.batched over Function.map uses chunk intake: ordered results,
    per-batch error isolation, common-kwargs transposition (round-2 perf
    path — 365k items/s/worker CPU vs ~15k on the per-item road)."""
    app = modal.App("batch-chunks")

    @app.function()
    @modal.batched(max_batch_size=64, wait_ms=1)
    def double(xs):
        assert isinstance(xs, list) and len(xs) <= 64
        return [x * 2 for x in xs]

    @app.function()
    @modal.batched(max_batch_size=32, wait_ms=1)
    def scale(xs, k):
        return [x * kk for x, kk in zip(xs, k)]

    @app.function()
    @modal.batched(max_batch_size=16, wait_ms=1)
    def picky(xs):
        if 40 in xs:
            raise ValueError("boom")
        return [x + 1 for x in xs]

    with app.run(client=client):
        assert list(double.map(range(1000))) == [x * 2 for x in range(1000)]
        assert list(scale.map(range(100), kwargs={"k": 3})) == [x * 3 for x in range(100)]
        res = list(picky.map(range(64), return_exceptions=True))
        bad = [r for r in res if isinstance(r, Exception)]
        assert len(bad) == 16  # only the batch containing 40 fails
        assert sum(1 for r in res if not isinstance(r, Exception)) == 48
        assert double.remote(5) == 10  # unary path still batches singles


@pytest.mark.parametrize(
    "batch,chunk",
    [(3, 8), (8, 8), (13, 8), (64, 16), (7, 64), (100, 64), (64, 256)],
)
def test_batched_chunk_geometry(client, monkeypatch, batch, chunk):
    """Protocol geometry sweep: batch size vs chunk size in every relation
    (batch < chunk, =, >, non-divisible) keeps map exactly-once + ordered."""
    monkeypatch.setenv("MODAL_AMD_CHUNK_ITEMS", str(chunk))
    app = modal.App(f"geo-{batch}-{chunk}")

    @app.function()
    @modal.batched(max_batch_size=batch, wait_ms=1)
    def f(xs):
        assert len(xs) <= batch
        return [x * 3 for x in xs]

    with app.run(client=client):
        n = 311  # prime: never divides evenly into batches or chunks
        assert list(f.map(range(n))) == [x * 3 for x in range(n)]
