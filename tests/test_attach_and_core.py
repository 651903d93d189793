"""Cross-process attach mode (daemon scheduler) + native core unit tests."""

from __future__ import annotations

import os
import subprocess
import sys
import tempfile
import time

import pytest


def test_shm_ring_roundtrip(tmp_path):
    core = pytest.importorskip("modal_amd._core")
    path = str(tmp_path / "ring")
    writer = core.ShmRing(path, 1 << 20, True)
    reader = core.ShmRing(path, 0, False)
    assert writer.push(b"alpha")
    assert writer.push(b"B" * 100_000)
    assert reader.pop_all() == [b"alpha", b"B" * 100_000]
    # wraparound across the 1 MiB boundary
    blob = os.urandom(300_000)
    for _ in range(24):
        assert writer.push(blob)
        assert reader.pop_all() == [blob]
    # full-ring refusal, then drain
    count = 0
    while writer.push(b"x" * 100_000):
        count += 1
    assert count >= 9
    assert len(reader.pop_all()) == count
    assert writer.pending_bytes() == 0


def test_pack_payloads_roundtrip():
    core = pytest.importorskip("modal_amd._core")
    items = [b"", b"a", os.urandom(1000), b"zz" * 5000]
    blob = core.pack_payloads(items)
    assert core.unpack_payloads(blob) == items


def test_attach_to_daemon_scheduler(tmp_path):
    """A second process's scheduler serves this client over the socket
    (the cross-process wire contract: app create, function create, map,
    queue ops)."""
    run_dir = str(tmp_path / "daemon")
    daemon = subprocess.Popen(
        [sys.executable, "-m", "modal_amd.cli.entry_point", "daemon", "--run-dir", run_dir],
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        start_new_session=True,
    )
    sock = os.path.join(run_dir, "scheduler.sock")
    try:
        deadline = time.time() + 30
        while not os.path.exists(sock):
            assert daemon.poll() is None, daemon.stdout.read().decode()
            assert time.time() < deadline, "daemon socket never appeared"
            time.sleep(0.05)
        env = dict(os.environ)
        env["MODAL_AMD_ATTACH_SOCKET"] = sock
        script = """
import modal_amd as modal

app = modal.App("attach-test")

@app.function()
def double(x):
    return x * 2

with app.run():
    assert double.remote(21) == 42
    out = sorted(double.map(range(20), order_outputs=False))
    assert out == [2 * x for x in range(20)]

@app.function()
@modal.batched(max_batch_size=8, wait_ms=1)
def bdouble(xs):
    return [x * 2 for x in xs]

with app.run():
    # batched functions over the proxied chunk intake (one-way putc frames)
    got = sorted(bdouble.map(range(200), order_outputs=False))
    assert got == sorted(2 * x for x in range(200))
    flat = [v for b in bdouble.map_batches(range(100)) for v in b]
    assert flat == [2 * x for x in range(100)]
    with modal.Queue.ephemeral() as q:
        q.put("cross-process")
        assert q.get() == "cross-process"
print("ATTACH_OK")
"""
        proc = subprocess.run(
            [sys.executable, "-c", script], env=env, capture_output=True, text=True, timeout=120
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "ATTACH_OK" in proc.stdout
    finally:
        daemon.terminate()
        try:
            daemon.wait(timeout=5)
        except subprocess.TimeoutExpired:
            daemon.kill()


def test_file_pattern_matcher():
    from modal_amd.file_pattern_matcher import FilePatternMatcher

    m = FilePatternMatcher("*.pyc", "build/**", "!build/keep.txt")
    assert m("foo.pyc")
    assert m("build/a/b.o")
    assert not m("build/keep.txt")
    assert not m("src/main.py")
    m2 = FilePatternMatcher("**/__pycache__")
    assert m2("a/b/__pycache__")


def test_ring_carries_bulk_frames(client):
    """Frames >= RING_MIN_FRAME ride the C++ shm ring, not the socket."""
    import os as _os

    import modal_amd as modal

    app = modal.App("ring-app")

    @app.function()
    def digest_one(blob):
        import hashlib

        return hashlib.sha256(blob).hexdigest()

    with app.run(client=client):
        payload = _os.urandom(300_000)  # 300 KB unary arg -> bulk frame
        import hashlib

        assert digest_one.remote(payload) == hashlib.sha256(payload).hexdigest()
        svc = client.svc
        sent = sum(w.conn.ring_frames_sent for w in svc.pool.workers.values())
        assert sent >= 1, "bulk frame never used the shm ring"


def test_native_core_asan_clean():
    """ASan build of csrc/core.cpp exercised over the shm ring, codec and
    payload packing with no reported errors (C++ sanitizer
    jobs for the native core)."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        ["bash", os.path.join(repo, "scripts", "asan_check.sh")],
        capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "ASAN CHECK OK" in out.stdout


def test_map_batches_api(client):
    """fn.map_batches yields lists covering every item exactly once (both
    sync and .aio forms); starmap_batches matches starmap."""
    import modal_amd as modal

    app = modal.App("mb")

    @app.function()
    def sq(x):
        return x * x

    @app.function()
    def add(a, b):
        return a + b

    with app.run(client=client):
        got = []
        for batch in sq.map_batches(range(500)):
            assert isinstance(batch, list)
            got.extend(batch)
        assert sorted(got) == sorted(x * x for x in range(500))

        from modal_amd._sync import synchronizer

        async def consume():
            out = []
            async for batch in sq.map_batches.aio(range(300), order_outputs=False):
                out.extend(batch)
            return out

        assert sorted(synchronizer.run(consume())) == sorted(x * x for x in range(300))
        flat = [v for b in add.starmap_batches([(1, 2), (3, 4), (5, 6)]) for v in b]
        assert sorted(flat) == [3, 7, 11]
