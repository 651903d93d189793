"""Streaming logs: offset-resumable long-poll, incl. over the proxy
transport (round-1 review Missing #5: a daemon-attached CLI couldn't tail
logs it didn't host)."""

from __future__ import annotations

import os
import subprocess
import sys
import threading
import time

import modal_amd as modal
from modal_amd._sync import synchronizer


def test_app_get_logs_offset_resume(client):
    app = modal.App("logs-app")

    @app.function()
    def chatty(i):
        print(f"line-{i}")
        return i

    with app.run(client=client):
        list(chatty.map(range(5), order_outputs=False))
        app_id = app.app_id
        svc = client.svc

        async def drain():
            # wait until all 5 lines arrived (stdout forwarding is async)
            deadline = time.time() + 20
            seen: list = []
            offset = 0
            while len(seen) < 5 and time.time() < deadline:
                resp = await svc.app_get_logs(app_id=app_id, offset=offset, timeout=2.0)
                seen += [e["data"] for e in resp["entries"]]
                assert resp["next_offset"] >= offset
                offset = resp["next_offset"]
            return seen, offset

        seen, offset = synchronizer.run(drain())
        joined = "".join(seen)
        for i in range(5):
            assert f"line-{i}" in joined
        # resume from the end: nothing new within the timeout
        resp = synchronizer.run(
            svc.app_get_logs(app_id=app_id, offset=offset, timeout=0.2)
        )
        assert resp["entries"] == []


def test_daemon_attached_tail_live(tmp_path):
    """`modal-amd app logs -f` against a daemon-hosted app: the tail sees
    lines produced AFTER it started (live, not a snapshot)."""
    run_dir = str(tmp_path / "daemon")
    daemon = subprocess.Popen(
        [sys.executable, "-m", "modal_amd.cli.entry_point", "daemon", "--run-dir", run_dir],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, start_new_session=True,
    )
    sock = os.path.join(run_dir, "scheduler.sock")
    try:
        deadline = time.time() + 30
        while not os.path.exists(sock):
            assert daemon.poll() is None
            assert time.time() < deadline
            time.sleep(0.05)

        env = dict(os.environ)
        env["MODAL_AMD_ATTACH_SOCKET"] = sock
        script = r"""
import sys, threading, time
import modal_amd as modal
from modal_amd.logs_manager import tail_app_logs

app = modal.App("tail-test")

@app.function()
def speak(i):
    print(f"spoken-{i}", flush=True)
    return i

with app.run():
    got = []
    def tail():
        for entry in tail_app_logs(app.app_id, timeout=25):
            got.append(entry["data"])
            if sum("spoken-" in d for d in got) >= 6:
                break
    t = threading.Thread(target=tail, daemon=True)
    t.start()
    # produce lines AFTER the tail started
    for i in range(6):
        speak.remote(i)
        time.sleep(0.05)
    t.join(timeout=30)
    text = "".join(got)
    missing = [i for i in range(6) if f"spoken-{i}" not in text]
    assert not missing, f"tail missed {missing}: {text!r}"
print("TAIL_OK")
"""
        proc = subprocess.run(
            [sys.executable, "-c", script], env=env, capture_output=True, text=True,
            timeout=120,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "TAIL_OK" in proc.stdout
    finally:
        daemon.terminate()
        try:
            daemon.wait(timeout=5)
        except subprocess.TimeoutExpired:
            daemon.kill()


def test_daemon_attached_app_run_streams_logs(tmp_path):
    """app.run() with enable_output against a DAEMON scheduler streams the
    app's worker prints back to the attached terminal (round-1: the proxy
    path returned nothing)."""
    run_dir = str(tmp_path / "daemon")
    daemon = subprocess.Popen(
        [sys.executable, "-m", "modal_amd.cli.entry_point", "daemon", "--run-dir", run_dir],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, start_new_session=True,
    )
    sock = os.path.join(run_dir, "scheduler.sock")
    try:
        deadline = time.time() + 30
        while not os.path.exists(sock):
            assert daemon.poll() is None
            assert time.time() < deadline
            time.sleep(0.05)
        env = dict(os.environ)
        env["MODAL_AMD_ATTACH_SOCKET"] = sock
        script = r"""
import time
import modal_amd as modal

app = modal.App("stream-app")

@app.function()
def talk(i):
    print(f"streamed-line-{i}", flush=True)
    return i

with modal.enable_output():
    with app.run():
        for i in range(4):
            talk.remote(i)
        time.sleep(2.0)  # let the long-poll round-trip deliver
print("RUN_DONE")
"""
        proc = subprocess.run(
            [sys.executable, "-c", script], env=env, capture_output=True, text=True,
            timeout=120,
        )
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "RUN_DONE" in proc.stdout
        missing = [i for i in range(4) if f"streamed-line-{i}" not in proc.stdout]
        assert not missing, f"attached app.run missed logs {missing}: {proc.stdout!r}"
    finally:
        daemon.terminate()
        try:
            daemon.wait(timeout=5)
        except subprocess.TimeoutExpired:
            daemon.kill()
