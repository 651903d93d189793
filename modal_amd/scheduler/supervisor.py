"""C++ clone3 sandbox spawner: atomic cgroup placement, no preexec_fn.

The asyncio road runs a Python ``preexec_fn`` between fork and exec — unsafe
in a threaded process (only async-signal-safe code is legal there) and racy
for cgroups (the child runs before any parent-side ``cgroup.procs`` write;
the child-side write still leaves a pre-attach window). The native spawner
(csrc/core.cpp ``spawn_supervised``) uses ``clone3(CLONE_INTO_CGROUP)``:
the kernel creates the child DIRECTLY inside the target cgroup v2, then the
child does only setsid/dup2/chdir/execve in C.

Parity target: the reference ships its sandbox supervisor as native code
(Go); this is the Linux-native C++ equivalent for the pieces
where nativeness matters (process creation), with asyncio pipes on top.
"""

from __future__ import annotations

import asyncio
import os
import signal
from typing import Any, Optional


def available() -> bool:
    try:
        from .. import _core

        return hasattr(_core, "spawn_supervised")
    except Exception:
        return False


class SupervisedProcess:
    """asyncio.subprocess.Process-shaped handle over a clone3 child."""

    def __init__(
        self,
        pid: int,
        stdin: Any,
        stdout: asyncio.StreamReader,
        stderr: asyncio.StreamReader,
    ):
        self.pid = pid
        self.stdin = stdin
        self.stdout = stdout
        self.stderr = stderr
        self.returncode: Optional[int] = None
        self._wait_fut: Optional[asyncio.Future] = None

    async def wait(self) -> int:
        if self.returncode is not None:
            return self.returncode
        loop = asyncio.get_running_loop()
        if self._wait_fut is None:

            def _reap() -> int:
                _pid, status = os.waitpid(self.pid, 0)
                if os.WIFSIGNALED(status):
                    return -os.WTERMSIG(status)
                return os.WEXITSTATUS(status)

            self._wait_fut = loop.run_in_executor(None, _reap)
        self.returncode = await asyncio.shield(self._wait_fut)
        return self.returncode

    def send_signal(self, sig: int) -> None:
        if self.returncode is None:
            os.kill(self.pid, sig)

    def kill(self) -> None:
        self.send_signal(signal.SIGKILL)

    def terminate(self) -> None:
        self.send_signal(signal.SIGTERM)


async def spawn(
    argv: list[str],
    *,
    cwd: str = "",
    env: Optional[dict[str, str]] = None,
    cgroup_dir: str = "",
    rlimit_as_mib: int = 0,
    cpu: Optional[float] = None,
) -> SupervisedProcess:
    """Spawn ``argv`` through the native supervisor with asyncio stdio pipes."""
    import shutil

    from .. import _core

    exe = argv[0]
    if "/" not in exe:
        resolved = shutil.which(exe)
        if resolved is None:
            raise FileNotFoundError(exe)
        argv = [resolved] + list(argv[1:])

    loop = asyncio.get_running_loop()
    stdin_r, stdin_w = os.pipe()
    stdout_r, stdout_w = os.pipe()
    stderr_r, stderr_w = os.pipe()
    try:
        pid = _core.spawn_supervised(
            list(argv),
            cwd or "",
            [f"{k}={v}" for k, v in (env or os.environ).items()],
            cgroup_dir,
            int(rlimit_as_mib or 0),
            stdin_r,
            stdout_w,
            stderr_w,
        )
    finally:
        os.close(stdin_r)
        os.close(stdout_w)
        os.close(stderr_w)
    if cpu:
        # affinity is advisory; by-pid assignment avoids preexec entirely
        try:
            import math

            avail = sorted(os.sched_getaffinity(0))
            os.sched_setaffinity(pid, set(avail[: max(1, math.ceil(cpu))]))
        except OSError:
            pass

    stdout = asyncio.StreamReader(loop=loop)
    await loop.connect_read_pipe(
        lambda: asyncio.StreamReaderProtocol(stdout, loop=loop),
        os.fdopen(stdout_r, "rb", 0),
    )
    stderr = asyncio.StreamReader(loop=loop)
    await loop.connect_read_pipe(
        lambda: asyncio.StreamReaderProtocol(stderr, loop=loop),
        os.fdopen(stderr_r, "rb", 0),
    )
    w_transport, w_protocol = await loop.connect_write_pipe(
        lambda: asyncio.streams.FlowControlMixin(loop=loop),
        os.fdopen(stdin_w, "wb", 0),
    )
    stdin = asyncio.StreamWriter(w_transport, w_protocol, None, loop)
    return SupervisedProcess(pid, stdin, stdout, stderr)
