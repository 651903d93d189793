"""Sandbox: create/exec/stdio/wait/FS (BASELINE config 4 behavior, CPU side)."""

from __future__ import annotations

import time

import pytest

import modal_amd as modal
from modal_amd.exception import SandboxTimeoutError


def test_sandbox_create_exec_wait(client):
    sb = modal.Sandbox.create("bash", "-c", "echo hello-out; echo hello-err >&2; exit 3")
    rc = sb.wait(raise_on_termination=False)
    assert rc == 3
    assert sb.stdout.read().strip() == "hello-out"
    assert sb.stderr.read().strip() == "hello-err"


def test_sandbox_exec_process(client):
    sb = modal.Sandbox.create("sleep", "60")
    try:
        p = sb.exec("bash", "-c", "echo from-exec; read line; echo got:$line")
        p.stdin.write("ping\n")
        p.stdin.write_eof()
        p.stdin.drain()
        assert p.wait() == 0
        out = p.stdout.read()
        assert "from-exec" in out
        assert "got:ping" in out
    finally:
        sb.terminate()


def test_sandbox_stdio_lines_and_offsets(client):
    sb = modal.Sandbox.create("bash", "-c", "for i in 1 2 3; do echo line$i; done")
    sb.wait(raise_on_termination=False)
    lines = [ln.strip() for ln in sb.stdout]
    assert lines == ["line1", "line2", "line3"]


def test_sandbox_timeout(client):
    sb = modal.Sandbox.create("sleep", "60", timeout=0.5)
    with pytest.raises(SandboxTimeoutError):
        sb.wait()


def test_sandbox_poll_and_terminate(client):
    sb = modal.Sandbox.create("sleep", "60")
    assert sb.poll() is None
    sb.terminate()
    rc = sb.wait(raise_on_termination=False)
    assert rc != 0


def test_sandbox_filesystem(client):
    sb = modal.Sandbox.create("sleep", "60")
    try:
        f = sb.open("hello.txt", "w")
        f.write("alpha\nbeta\n")
        f.close()
        f2 = sb.open("hello.txt", "r")
        assert f2.read() == "alpha\nbeta\n"
        f2.close()
        assert "hello.txt" in sb.ls(".")
        sb.mkdir("subdir")
        assert sb.exists("subdir")
        sb.rm("hello.txt")
        assert not sb.exists("hello.txt")
    finally:
        sb.terminate()


def test_sandbox_env_and_secrets(client):
    sb = modal.Sandbox.create(
        "bash", "-c", "echo v=$MYVAR s=$MYSECRET",
        env={"MYVAR": "direct"},
        secrets=[modal.Secret.from_dict({"MYSECRET": "fromsecret"})],
    )
    sb.wait(raise_on_termination=False)
    assert sb.stdout.read().strip() == "v=direct s=fromsecret"


def test_sandbox_volume_mount(client):
    with modal.Volume.ephemeral() as vol:
        with vol.batch_upload() as batch:
            import io

            batch.put_file(io.BytesIO(b"volume-data"), "data.txt")
        sb = modal.Sandbox.create(
            "bash", "-c", "cat vol/data.txt", volumes={"vol": vol}
        )
        sb.wait(raise_on_termination=False)
        assert sb.stdout.read() == "volume-data"


def test_sandbox_tags_and_list(client):
    sb = modal.Sandbox.create("sleep", "60", name="tagged-sb")
    try:
        sb.set_tags({"team": "infra"})
        found = modal.Sandbox.list(tags={"team": "infra"})
        assert any(s.object_id == sb.object_id for s in found)
        by_name = modal.Sandbox.from_name("tagged-sb")
        assert by_name.object_id == sb.object_id
    finally:
        sb.terminate()


def test_sandbox_snapshot_fs(client):
    sb = modal.Sandbox.create("bash", "-c", "echo snapshot-me > file.txt; sleep 60")
    try:
        time.sleep(0.3)
        img = sb.snapshot_filesystem()
        assert img.object_id.startswith("im-")
    finally:
        sb.terminate()


def test_sandbox_snapshot_and_restore(client):
    """fs snapshot -> new sandbox restored from it (parity: sandbox
    _experimental_snapshot / from_snapshot flow)."""
    sb = modal.Sandbox.create("bash", "-c", "echo snapshot-state > state.txt; sleep 60")
    try:
        time.sleep(0.4)
        img = sb.snapshot_filesystem()
    finally:
        sb.terminate()
    sb2 = modal.Sandbox.create("bash", "-c", "cat state.txt", image=img)
    sb2.wait(raise_on_termination=False)
    assert sb2.stdout.read().strip() == "snapshot-state"


def test_sandbox_pty_exec(client):
    """PTY-backed exec: the command sees a real controlling terminal,
    stdout+stderr are merged, and resize propagates TIOCSWINSZ."""
    sb = modal.Sandbox.create("sleep", "60")
    try:
        p = sb.exec(
            "bash", "-c", "tty; echo err-msg >&2; stty size; read x; echo got:$x",
            pty_info={"rows": 31, "cols": 97},
        )
        p.resize(40, 120)
        p.stdin.write("ping\n")
        p.stdin.drain()
        assert p.wait() == 0
        out = p.stdout.read()
        assert "/dev/pts/" in out          # a real PTY, controlling tty works
        assert "err-msg" in out            # stderr merged into the PTY stream
        # window size came through TIOCSWINSZ (initial 31x97, or 40x120 if
        # the resize() landed before the shell ran stty)
        assert "31 97" in out or "40 120" in out
        assert "got:ping" in out           # interactive stdin through the master
    finally:
        sb.terminate()


def test_sandbox_memory_limit_enforced(client):
    """Sandbox memory= caps the process address space (RLIMIT_AS):
    a 200 MiB allocation under a 128 MiB cap dies; an 8 MiB one lives."""
    code = "x = bytearray(200 * 1024 * 1024); print('allocated')"
    sb = modal.Sandbox.create("python3", "-c", code, memory=128)
    rc = sb.wait(raise_on_termination=False)
    assert rc != 0, sb.stdout.read()
    err = sb.stderr.read()
    assert "MemoryError" in err or rc != 0

    sb2 = modal.Sandbox.create(
        "python3", "-c", "x = bytearray(8 * 1024 * 1024); print('ok')", memory=512
    )
    assert sb2.wait(raise_on_termination=False) == 0
    assert "ok" in sb2.stdout.read()


def test_sandbox_cpu_affinity(client):
    import os as _os

    if len(_os.sched_getaffinity(0)) < 2:
        import pytest as _pytest

        _pytest.skip("needs >=2 CPUs")
    sb = modal.Sandbox.create(
        "python3", "-c", "import os; print(len(os.sched_getaffinity(0)))", cpu=1
    )
    assert sb.wait(raise_on_termination=False) == 0
    assert sb.stdout.read().strip() == "1"
