"""Sandbox service: on-demand supervised processes with exec/stdio/FS.

Single-node re-implementation of the reference's Sandbox control plane +
task command router (/root/reference/py/modal/sandbox.py:370,
task_command_router_client.py:211). A sandbox is a process group rooted in a
private workdir under the run dir; ``exec`` spawns additional processes in
the same workdir/env (the container-exec analog). Stdio is captured into
offset-addressable buffers so reads are resumable from any byte offset —
the same contract as the reference's offset-resumable stdio streams
(task_command_router_client.py:31,523-614).
"""

from __future__ import annotations

import asyncio
import os
import signal
import time
from typing import Any, Optional

from ..exception import NotFoundError, SandboxTimeoutError
from ..utils.ids import new_id

STDIN_CHUNK = 256 * 1024  # parity: reference task_command_router_client.py:31



def _resource_preexec(cpu: "Optional[float]", memory_mib: "Optional[int]"):
    """Unprivileged resource enforcement for sandbox processes (parity:
    the reference's container cpu/memory reservations, sandbox.py:551):
    RLIMIT_AS caps the address space at ``memory`` MiB; ``cpu`` cores are
    approximated by pinning the process to that many CPUs. Runs in the
    child between fork and exec."""
    if cpu is None and memory_mib is None:
        return None

    def _apply() -> None:
        os.setsid()
        if memory_mib:
            import resource

            limit = int(memory_mib) * 1024 * 1024
            try:
                resource.setrlimit(resource.RLIMIT_AS, (limit, limit))
            except (ValueError, OSError):
                pass
        if cpu:
            try:
                import math

                avail = sorted(os.sched_getaffinity(0))
                take = max(1, math.ceil(cpu))
                os.sched_setaffinity(0, set(avail[:take]))
            except OSError:
                pass

    return _apply


class _ProcState:
    """One supervised process: the sandbox entrypoint or an exec."""

    def __init__(self, proc_id: str):
        self.proc_id = proc_id
        self.proc: Optional[asyncio.subprocess.Process] = None
        self.stdout = bytearray()
        self.stderr = bytearray()
        self.stdout_eof = False
        self.stderr_eof = False
        self.cond = asyncio.Condition()
        self.returncode: Optional[int] = None
        self.stdin_offset = 0  # bytes accepted so far (resume dedup)
        self._readers: list[asyncio.Task] = []
        self.pty_master: Optional[int] = None  # PTY-backed exec (shell)

    async def attach(self, proc: asyncio.subprocess.Process) -> None:
        self.proc = proc
        loop = asyncio.get_running_loop()
        if proc.stdout is not None:
            self._readers.append(loop.create_task(self._pump(proc.stdout, 1)))
        if proc.stderr is not None:
            self._readers.append(loop.create_task(self._pump(proc.stderr, 2)))
        self._readers.append(loop.create_task(self._wait()))

    async def attach_pty(self, proc: asyncio.subprocess.Process, master: int) -> None:
        """PTY-backed process: output (stdout+stderr merged, PTY semantics)
        comes from the master fd; stdin writes go to it."""
        self.proc = proc
        self.pty_master = master
        loop = asyncio.get_running_loop()
        self._readers.append(loop.create_task(self._pump_pty()))
        self._readers.append(loop.create_task(self._wait()))

    async def _pump_pty(self) -> None:
        loop = asyncio.get_running_loop()
        while True:
            try:
                chunk = await loop.run_in_executor(None, os.read, self.pty_master, 65536)
            except OSError:
                chunk = b""  # EIO when the last slave fd closes = EOF
            async with self.cond:
                if not chunk:
                    self.stdout_eof = True
                    self.stderr_eof = True
                    self.cond.notify_all()
                    try:
                        os.close(self.pty_master)
                    except OSError:
                        pass
                    self.pty_master = None
                    return
                self.stdout.extend(chunk)
                self.cond.notify_all()

    def resize(self, rows: int, cols: int) -> None:
        if self.pty_master is None:
            raise NotFoundError(f"{self.proc_id} has no PTY")
        import fcntl
        import struct
        import termios

        fcntl.ioctl(self.pty_master, termios.TIOCSWINSZ, struct.pack("HHHH", rows, cols, 0, 0))

    async def _pump(self, stream: asyncio.StreamReader, fd: int) -> None:
        buf = self.stdout if fd == 1 else self.stderr
        while True:
            chunk = await stream.read(64 * 1024)
            async with self.cond:
                if not chunk:
                    if fd == 1:
                        self.stdout_eof = True
                    else:
                        self.stderr_eof = True
                    self.cond.notify_all()
                    return
                buf.extend(chunk)
                self.cond.notify_all()

    async def _wait(self) -> None:
        rc = await self.proc.wait()
        async with self.cond:
            self.returncode = rc
            self.stdout_eof = True
            self.stderr_eof = True
            self.cond.notify_all()

    async def read(self, fd: int, offset: int, max_bytes: int, timeout: float) -> dict:
        deadline = time.monotonic() + timeout
        async with self.cond:
            while True:
                buf = self.stdout if fd == 1 else self.stderr
                eof = self.stdout_eof if fd == 1 else self.stderr_eof
                if offset < len(buf):
                    data = bytes(buf[offset : offset + max_bytes])
                    return {"data": data, "eof": eof and offset + len(data) >= len(buf),
                            "next_offset": offset + len(data)}
                if eof:
                    return {"data": b"", "eof": True, "next_offset": offset}
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    return {"data": b"", "eof": False, "next_offset": offset}
                try:
                    await asyncio.wait_for(self.cond.wait(), remaining)
                except asyncio.TimeoutError:
                    return {"data": b"", "eof": False, "next_offset": offset}

    async def write_stdin(self, offset: int, data: bytes, eof: bool) -> int:
        """Offset-resumable stdin: bytes before stdin_offset are dedup'd
        (parity: resumable stdin stream, reference :523-614)."""
        if self.pty_master is not None:
            if offset < self.stdin_offset:
                data = data[self.stdin_offset - offset :]
            if data:
                loop = asyncio.get_running_loop()
                await loop.run_in_executor(None, os.write, self.pty_master, bytes(data))
                self.stdin_offset += len(data)
            # PTY has no half-close; EOF is signalled in-band (^D) by clients
            return self.stdin_offset
        if self.proc is None or self.proc.stdin is None:
            raise NotFoundError("stdin not available")
        if offset < self.stdin_offset:
            data = data[self.stdin_offset - offset :]
        if data:
            self.proc.stdin.write(data)
            self.stdin_offset += len(data)
            await self.proc.stdin.drain()
        if eof:
            self.proc.stdin.close()
        return self.stdin_offset

    async def wait(self, timeout: Optional[float] = None) -> int:
        async with self.cond:
            deadline = None if timeout is None else time.monotonic() + timeout
            while self.returncode is None:
                remaining = None if deadline is None else deadline - time.monotonic()
                if remaining is not None and remaining <= 0:
                    raise asyncio.TimeoutError
                await asyncio.wait_for(self.cond.wait(), remaining)
            return self.returncode

    def kill(self, sig: int = signal.SIGKILL) -> None:
        if self.proc is not None and self.returncode is None:
            try:
                os.killpg(self.proc.pid, sig)
            except (ProcessLookupError, PermissionError):
                try:
                    self.proc.kill()
                except ProcessLookupError:
                    pass


class SandboxState:
    def __init__(self, sandbox_id: str, workdir: str, env: dict, app_id: str):
        self.sandbox_id = sandbox_id
        self.task_id = new_id("task")
        self.workdir = workdir
        self.env = env
        self.app_id = app_id
        self.main = _ProcState(sandbox_id)
        self.execs: dict[str, _ProcState] = {}
        self.created_at = time.time()
        self.timeout: Optional[float] = None
        self.timeout_task: Optional[asyncio.Task] = None
        self.timed_out = False
        self.name: Optional[str] = None
        self.tags: dict[str, str] = {}
        self.gpu_index: Optional[int] = None
        # isolation state (scheduler/isolation.py): set when the sandbox
        # runs in its own PID+mount namespaces / cgroup / overlay root
        self.cgroup: Any = None
        self.ns_root: Optional[str] = None  # overlay mountpoint for chroot
        self.ns_ready: Optional[str] = None  # host-visible mount sentinel
        self.isolated = False

    def ns_target_pid(self) -> Optional[int]:
        """PID of the process inside the namespaces (unshare's child) —
        the nsenter target for the exec path. Polls briefly: an exec issued
        right after create can race unshare's fork."""
        if not self.isolated or self.main.proc is None:
            return None
        pid = self.main.proc.pid
        deadline = time.time() + 2.0
        while time.time() < deadline:
            try:
                with open(f"/proc/{pid}/task/{pid}/children") as f:
                    kids = f.read().split()
            except OSError:
                return None  # main exited
            if kids:
                try:
                    return int(kids[0])
                except ValueError:
                    return None
            if self.main.returncode is not None:
                return None
            time.sleep(0.01)
        return None


class SandboxService:
    def __init__(self, run_dir: str):
        self.run_dir = run_dir
        self.root = os.path.join(run_dir, "sandboxes")
        os.makedirs(self.root, exist_ok=True)
        self.sandboxes: dict[str, SandboxState] = {}
        self.by_name: dict[tuple[str, str], str] = {}

    def _get(self, sandbox_id: str) -> SandboxState:
        sb = self.sandboxes.get(sandbox_id)
        if sb is None:
            raise NotFoundError(f"Sandbox {sandbox_id} not found")
        return sb

    def _proc(self, target_id: str) -> _ProcState:
        if target_id in self.sandboxes:
            return self.sandboxes[target_id].main
        for sb in self.sandboxes.values():
            if target_id in sb.execs:
                return sb.execs[target_id]
        raise NotFoundError(f"No process {target_id}")

    async def create(
        self,
        entrypoint_args: list[str],
        env: Optional[dict] = None,
        workdir: Optional[str] = None,
        timeout: Optional[float] = None,
        gpu: Optional[int] = None,
        app_id: str = "",
        name: Optional[str] = None,
        environment: str = "main",
        volume_paths: Optional[dict] = None,
        cpu: Optional[float] = None,
        memory: Optional[int] = None,
        pids_max: Optional[int] = None,
        image_fsroot: Optional[str] = None,
        restore_blob: Optional[str] = None,
        blob_store: Any = None,
    ) -> dict:
        sandbox_id = new_id("sandbox")
        sb_dir = os.path.join(self.root, sandbox_id)
        os.makedirs(sb_dir, exist_ok=True)
        if restore_blob and blob_store is not None:
            await self.restore_fs(sb_dir, blob_store, restore_blob)
        full_env = dict(os.environ)
        full_env.update(env or {})
        full_env["MODAL_AMD_SANDBOX_ID"] = sandbox_id
        if gpu is not None:
            full_env["HIP_VISIBLE_DEVICES"] = str(gpu)
            full_env["CUDA_VISIBLE_DEVICES"] = str(gpu)
        # volume mounts: symlink shared trees into the sandbox workdir space
        for mount_path, vol_dir in (volume_paths or {}).items():
            os.makedirs(vol_dir, exist_ok=True)
            link = mount_path if os.path.isabs(mount_path) else os.path.join(sb_dir, mount_path)
            try:
                os.makedirs(os.path.dirname(link), exist_ok=True)
                if not os.path.exists(link):
                    os.symlink(vol_dir, link)
            except OSError:
                pass
        state = SandboxState(sandbox_id, workdir or sb_dir, full_env, app_id)
        state.gpu_index = gpu
        if name:
            self.by_name[(environment, name)] = sandbox_id
            state.name = name
        self.sandboxes[sandbox_id] = state
        args = entrypoint_args or ["sleep", "infinity"]

        # real isolation where the node allows it: cgroup limits +
        # PID/mount namespaces + overlay root (scheduler/isolation.py);
        # rlimit+affinity fallback otherwise (round-1 behavior)
        from .isolation import CgroupBox, isolation_argv

        cg = None
        if cpu or memory or pids_max:
            cg = CgroupBox(sandbox_id, memory_mib=memory, cpu=cpu, pids_max=pids_max)
            if not cg.create():
                cg = None
        state.cgroup = cg
        wrapped = isolation_argv(
            args,
            sandbox_dir=sb_dir,
            run_dir=self.run_dir,
            workdir=state.workdir,
            image_fsroot=image_fsroot,
        )
        if wrapped is not None:
            args, ns_meta = wrapped
            state.isolated = True
            state.ns_root = ns_meta["mnt"]
            state.ns_ready = ns_meta["ready"]

        # native spawner first: clone3(CLONE_INTO_CGROUP) places the child in
        # its cgroup v2 ATOMICALLY, and the child runs only C between clone
        # and exec (no Python preexec_fn in a threaded process). Python
        # fallback covers cgroup v1 boxes and MODAL_AMD_PY_SPAWN=1.
        from . import supervisor as _sup

        # the native spawner pays a full fork of this (torch-loaded) process
        # (~1 ms of page-table copy; CPython's no-preexec road uses vfork),
        # so take it only when its atomic cgroup placement matters
        cxx_ok = (
            os.environ.get("MODAL_AMD_PY_SPAWN") != "1"
            and _sup.available()
            and cg is not None
            and cg.v2_dir is not None
        )
        proc = None
        if cxx_ok:
            try:
                proc = await _sup.spawn(
                    list(args),
                    cwd=state.workdir,
                    env=full_env,
                    cgroup_dir=(cg.v2_dir if cg is not None else "") or "",
                    rlimit_as_mib=int(memory) if (memory and cg is None) else 0,
                    cpu=cpu,
                )
            except Exception:
                proc = None  # e.g. clone3 blocked by seccomp: take the Python road
        if proc is None:
            rlimit_fallback = _resource_preexec(cpu, None if cg is not None else memory)

            def preexec() -> None:
                if rlimit_fallback is not None:
                    rlimit_fallback()  # includes setsid
                else:
                    os.setsid()
                if cg is not None:
                    cg.attach_pid_in_child()  # inherited by the whole subtree

            proc = await asyncio.create_subprocess_exec(
                *args,
                cwd=state.workdir,
                env=full_env,
                stdin=asyncio.subprocess.PIPE,
                stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.PIPE,
                preexec_fn=preexec,
            )
        await state.main.attach(proc)
        if timeout:
            state.timeout = timeout

            async def _timeout_kill() -> None:
                await asyncio.sleep(timeout)
                state.timed_out = True
                state.main.kill()

            state.timeout_task = asyncio.get_running_loop().create_task(_timeout_kill())
        return {"sandbox_id": sandbox_id, "task_id": state.task_id}

    async def exec(
        self,
        sandbox_id: str,
        cmd: list[str],
        env: Optional[dict] = None,
        workdir: Optional[str] = None,
        timeout: Optional[float] = None,
        exec_id: Optional[str] = None,
        pty: bool = False,
        rows: int = 24,
        cols: int = 80,
    ) -> dict:
        sb = self._get(sandbox_id)
        # exec_id idempotency (parity: command-router exec_id semantics)
        if exec_id and exec_id in sb.execs:
            return {"exec_id": exec_id}
        exec_id = exec_id or new_id("task")
        state = _ProcState(exec_id)
        sb.execs[exec_id] = state
        full_env = dict(sb.env)
        full_env.update(env or {})
        if pty:
            # interactive exec (modal-amd shell): run on a pseudo-terminal,
            # stdout+stderr merged, child gets it as controlling tty
            import fcntl
            import pty as _pty
            import struct
            import termios

            master, slave = _pty.openpty()
            fcntl.ioctl(slave, termios.TIOCSWINSZ, struct.pack("HHHH", rows, cols, 0, 0))
            full_env.setdefault("TERM", "xterm-256color")

            def _make_ctty() -> None:
                os.setsid()
                fcntl.ioctl(0, termios.TIOCSCTTY, 0)

            try:
                proc = await asyncio.create_subprocess_exec(
                    *cmd,
                    cwd=workdir or sb.workdir,
                    env=full_env,
                    stdin=slave,
                    stdout=slave,
                    stderr=slave,
                    preexec_fn=_make_ctty,
                )
            except BaseException:
                os.close(master)
                raise
            finally:
                os.close(slave)
            await state.attach_pty(proc, master)
        else:
            argv = list(cmd)
            cwd = workdir or sb.workdir
            target = sb.ns_target_pid()
            if target is not None and sb.ns_ready:
                # don't nsenter mid-setup: wait for the mount sentinel
                deadline = time.time() + 5.0
                while not os.path.exists(sb.ns_ready) and time.time() < deadline:
                    if sb.main.returncode is not None:
                        break
                    await asyncio.sleep(0.01)
            if target is not None:
                # join the sandbox's PID+mount namespaces (+chroot into its
                # overlay root): the exec sees exactly what the sandbox sees
                # (parity: command-router exec runs inside the container)
                from .isolation import nsenter_argv

                argv = nsenter_argv(target, argv, cwd, sb.ns_root)
                cwd = None
            # execs join the sandbox's cgroup (parity: container execs are
            # subject to the container's limits) via the native spawner
            from . import supervisor as _sup

            cgdir = ""
            if sb.cgroup is not None and getattr(sb.cgroup, "v2_dir", None):
                cgdir = sb.cgroup.v2_dir
            proc = None
            # native only when joining the sandbox cgroup (else vfork wins)
            if cgdir and os.environ.get("MODAL_AMD_PY_SPAWN") != "1" and _sup.available():
                try:
                    proc = await _sup.spawn(
                        list(argv), cwd=cwd or "", env=full_env, cgroup_dir=cgdir
                    )
                except Exception:
                    proc = None
            if proc is None:
                proc = await asyncio.create_subprocess_exec(
                    *argv,
                    **({"cwd": cwd} if cwd else {}),
                    env=full_env,
                    stdin=asyncio.subprocess.PIPE,
                    stdout=asyncio.subprocess.PIPE,
                    stderr=asyncio.subprocess.PIPE,
                    start_new_session=True,
                )
            await state.attach(proc)
        if timeout:

            async def _timeout_kill() -> None:
                await asyncio.sleep(timeout)
                state.kill()

            asyncio.get_running_loop().create_task(_timeout_kill())
        return {"exec_id": exec_id}

    async def stdio_read(
        self, target_id: str, fd: int, offset: int, max_bytes: int = 1 << 20, timeout: float = 55.0
    ) -> dict:
        return await self._proc(target_id).read(fd, offset, max_bytes, timeout)

    async def stdin_write(self, target_id: str, offset: int, data: bytes, eof: bool = False) -> int:
        return await self._proc(target_id).write_stdin(offset, data, eof)

    async def resize(self, target_id: str, rows: int, cols: int) -> None:
        """Propagate a terminal resize to a PTY-backed exec (SIGWINCH)."""
        self._proc(target_id).resize(rows, cols)

    async def wait(self, target_id: str, timeout: Optional[float] = None, raise_on_timeout: bool = True) -> dict:
        state = self._proc(target_id)
        sb = self.sandboxes.get(target_id)
        try:
            rc = await state.wait(timeout)
        except asyncio.TimeoutError:
            if raise_on_timeout:
                raise SandboxTimeoutError(f"{target_id} still running after {timeout}s") from None
            return {"returncode": None, "running": True}
        timed_out = bool(sb.timed_out) if sb is not None else False
        return {"returncode": rc, "running": False, "timed_out": timed_out}

    async def poll(self, target_id: str) -> dict:
        state = self._proc(target_id)
        sb = self.sandboxes.get(target_id)
        return {
            "returncode": state.returncode,
            "running": state.returncode is None,
            "timed_out": bool(sb.timed_out) if sb is not None else False,
        }

    async def terminate(self, sandbox_id: str) -> None:
        sb = self._get(sandbox_id)
        if sb.timeout_task is not None:
            sb.timeout_task.cancel()
        for ex in sb.execs.values():
            ex.kill()
        sb.main.kill()
        if sb.cgroup is not None:
            # members die asynchronously; try now, the stale reaper gets
            # whatever is left
            await asyncio.sleep(0)
            sb.cgroup.cleanup()
        # /dev/shm fsdiff fallback (isolation_argv when the sandbox dir's
        # fs can't host an overlay upper): reap it with the sandbox
        shm = os.path.join("/dev/shm", "modal-amd-fsdiff", sandbox_id)
        if os.path.isdir(shm):
            import shutil

            shutil.rmtree(shm, ignore_errors=True)

    async def list(self, app_id: Optional[str] = None, tags: Optional[dict] = None) -> list[dict]:
        out = []
        for sb in self.sandboxes.values():
            if app_id and sb.app_id != app_id:
                continue
            if tags and any(sb.tags.get(k) != v for k, v in tags.items()):
                continue
            out.append(
                {
                    "sandbox_id": sb.sandbox_id,
                    "task_id": sb.task_id,
                    "created_at": sb.created_at,
                    "returncode": sb.main.returncode,
                    "name": sb.name,
                    "tags": sb.tags,
                }
            )
        return out

    async def set_tags(self, sandbox_id: str, tags: dict) -> None:
        self._get(sandbox_id).tags.update(tags)

    async def from_name(self, name: str, environment: str = "main") -> str:
        sid = self.by_name.get((environment, name))
        if sid is None:
            raise NotFoundError(f"Sandbox '{name}' not found")
        return sid

    async def restore_fs(self, workdir: str, blob_store: Any, blob_id: str) -> None:
        """Materialize a filesystem snapshot (tar.gz in the CAS) into a fresh
        sandbox workdir (parity: restore via _experimental_from_snapshot,
        reference sandbox.py:2210-2337)."""
        import io
        import tarfile

        data = blob_store.get(blob_id)
        with tarfile.open(fileobj=io.BytesIO(data), mode="r:gz") as tar:
            tar.extractall(workdir)

    async def snapshot_fs(self, sandbox_id: str, blob_store: Any) -> dict:
        """Tar the sandbox workdir into the CAS (parity: SandboxSnapshotFs)."""
        sb = self._get(sandbox_id)
        import io
        import tarfile

        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w:gz") as tar:
            tar.add(sb.workdir, arcname=".")
        digest = blob_store.put(buf.getvalue())
        return {"image_id": new_id("image"), "blob_id": digest}

    async def shutdown(self) -> None:
        for sb in list(self.sandboxes.values()):
            try:
                await self.terminate(sb.sandbox_id)
            except Exception:
                pass

    def cleanup_dirs(self) -> None:
        import shutil

        shutil.rmtree(self.root, ignore_errors=True)
