"""Sandbox/image isolation: cgroups, namespaces, overlay roots.

Round-1 review Missing #3: sandboxes were process groups with rlimits.
These tests verify the container-grade path where the node allows it
(root + unshare + overlayfs + cgroup controllers) and skip where not.
"""

from __future__ import annotations

import os
import time

import pytest

import modal_amd as modal
from modal_amd.scheduler.isolation import capabilities


def _caps():
    return capabilities()


needs_pidns = pytest.mark.skipif(
    not _caps().get("pidns"), reason="no PID-namespace privileges on this node"
)
needs_overlay = pytest.mark.skipif(
    not _caps().get("overlay"), reason="no overlayfs privileges on this node"
)
needs_cg_memory = pytest.mark.skipif(
    not (_caps().get("cgv2") or _caps().get("cgv1_memory")),
    reason="no writable memory cgroup controller",
)
needs_cg_pids = pytest.mark.skipif(
    not (_caps().get("cgv2") or _caps().get("cgv1_pids")),
    reason="no writable pids cgroup controller",
)


@needs_pidns
def test_sandbox_cannot_see_host_pids(client):
    sb = modal.Sandbox.create(
        "sh", "-c", "ls /proc | grep -c '^[0-9][0-9]*$'", client=client
    )
    sb.wait()
    n = int(sb.stdout.read().strip())
    assert n <= 3, f"sandbox saw {n} PIDs — host pid namespace is leaking"
    assert sb.returncode == 0


@needs_pidns
@needs_overlay
def test_sandbox_absolute_writes_stay_in_diff(client, run_dir):
    marker = f"/isolation-test-{os.getpid()}"
    sb = modal.Sandbox.create(
        "sh", "-c", f"echo contained > {marker} && cat {marker}", client=client
    )
    sb.wait()
    assert sb.returncode == 0
    assert "contained" in sb.stdout.read()
    # the write never reached the host root
    assert not os.path.exists(marker)
    # ... but landed in the sandbox's fs diff
    sandboxes_root = os.path.join(run_dir, "sandboxes")
    diffs = [
        os.path.join(sandboxes_root, d, "fsdiff", marker.lstrip("/"))
        for d in os.listdir(sandboxes_root)
    ]
    assert any(os.path.exists(p) for p in diffs)


@needs_pidns
@needs_overlay
def test_sandbox_workdir_writes_reach_host(client, run_dir):
    """The run_dir (volumes, workdir) is bind-mounted through: that's the
    data plane, and it must NOT be swallowed by the overlay."""
    sb = modal.Sandbox.create("sh", "-c", "echo through > fromsandbox.txt", client=client)
    sb.wait()
    assert sb.returncode == 0
    sandboxes_root = os.path.join(run_dir, "sandboxes")
    found = [
        d for d in os.listdir(sandboxes_root)
        if os.path.exists(os.path.join(sandboxes_root, d, "fromsandbox.txt"))
    ]
    assert found, "workdir write did not reach the host-side sandbox dir"


@needs_cg_memory
def test_sandbox_memory_limit_enforced(client):
    # allocate ~300 MiB under a 128 MiB memory.max: the allocator dies
    sb = modal.Sandbox.create(
        "python3", "-c", "x = bytearray(300 * 1024 * 1024); print('survived')",
        memory=128, client=client,
    )
    sb.wait(raise_on_termination=False)
    assert sb.returncode != 0
    assert "survived" not in sb.stdout.read()


@needs_pidns
def test_exec_joins_sandbox_namespaces(client):
    sb = modal.Sandbox.create("sleep", "30", client=client)
    try:
        p = sb.exec("sh", "-c", "ls /proc | grep -c '^[0-9][0-9]*$'")
        p.wait()
        n = int(p.stdout.read().strip())
        assert n <= 5, f"exec saw {n} PIDs — not inside the sandbox namespace"
    finally:
        sb.terminate()


@needs_pidns
@needs_overlay
def test_image_run_commands_inside_image_root(client, run_dir):
    """Image.run_commands absolute-path writes become image content
    (the fsdiff layer), not host mutations."""
    image = modal.Image.debian_slim().run_commands(
        "echo layercontent > /etc/modal-image-marker"
    )
    app = modal.App("iso-image")

    @app.function(image=image)
    def noop():
        return 1

    with app.run(client=client):
        assert noop.remote() == 1
    assert not os.path.exists("/etc/modal-image-marker")
    images_root = os.path.join(run_dir, "images")
    hits = []
    for d in os.listdir(images_root):
        p = os.path.join(images_root, d, "fsdiff", "etc", "modal-image-marker")
        if os.path.exists(p):
            hits.append(p)
    assert hits, "run_commands write did not land in the image fs layer"
    assert open(hits[0]).read().strip() == "layercontent"


@needs_pidns
@needs_overlay
def test_sandbox_sees_image_layer(client):
    """A sandbox created from a built image sees the image's fs layer at
    its absolute path (overlay lower stacking)."""
    image = modal.Image.debian_slim().run_commands(
        "echo from-image > /etc/modal-image-file"
    )
    sb = modal.Sandbox.create(
        "cat", "/etc/modal-image-file", image=image, client=client
    )
    sb.wait()
    assert sb.returncode == 0, sb.stderr.read()
    assert "from-image" in sb.stdout.read()


def test_cxx_supervisor_spawn_unit():
    """The native clone3 spawner: stdio pipes, exit codes, signals, setsid."""
    import asyncio

    from modal_amd.scheduler import supervisor

    if not supervisor.available():
        pytest.skip("native core without spawn_supervised")

    async def main():
        p = await supervisor.spawn(["bash", "-c", "read l; echo ok:$l; exit 5"])
        p.stdin.write(b"x\n")
        await p.stdin.drain()
        p.stdin.close()
        assert await p.stdout.read() == b"ok:x\n"
        assert await p.wait() == 5
        p2 = await supervisor.spawn(["sleep", "30"])
        assert os.getpgid(p2.pid) == p2.pid  # setsid: own process group
        p2.terminate()
        assert await p2.wait() == -15

    asyncio.run(main())


def test_sandbox_uses_cxx_spawner(client):
    """Sandboxes created through the native path behave identically
    (stdio capture, returncode, termination)."""
    import modal_amd as modal

    sb = modal.Sandbox.create("bash", "-c", "echo native-spawn; exit 7", client=client)
    rc = sb.wait(raise_on_termination=False)
    assert rc == 7
    assert "native-spawn" in sb.stdout.read()


def test_cxx_supervisor_cgroup_attach(client):
    """With delegated cgroup v2, the child lands in its box atomically."""
    from modal_amd.scheduler.isolation import CgroupBox, capabilities

    if not capabilities().get("cgv2"):
        pytest.skip("no cgroup v2 delegation")
    cg = CgroupBox("cxx-attach-test", memory_mib=256)
    if not cg.create() or cg.v2_dir is None:
        pytest.skip("cgroup v2 box creation not permitted")
    import asyncio

    from modal_amd.scheduler import supervisor

    async def main():
        p = await supervisor.spawn(
            ["bash", "-c", "cat /proc/self/cgroup; sleep 0.1"],
            cgroup_dir=cg.v2_dir,
        )
        out = await p.stdout.read()
        await p.wait()
        return out.decode()

    try:
        out = asyncio.run(main())
        assert "cxx-attach-test" in out, out
    finally:
        cg.cleanup()


def test_sandbox_python_spawn_fallback(client, monkeypatch):
    """MODAL_AMD_PY_SPAWN=1 keeps the asyncio+preexec road working (the
    cgroup-v1 / no-clone3 fallback must not rot)."""
    import modal_amd as modal

    monkeypatch.setenv("MODAL_AMD_PY_SPAWN", "1")
    sb = modal.Sandbox.create("bash", "-c", "echo py-road; exit 4", client=client)
    assert sb.wait(raise_on_termination=False) == 4
    assert "py-road" in sb.stdout.read()
    from modal_amd.scheduler.supervisor import SupervisedProcess

    st = client.svc.sandbox_service._get(sb.object_id)
    assert not isinstance(st.main.proc, SupervisedProcess)
