"""Sandbox: on-demand supervised processes with exec/stdio/FS/snapshots.

Parity: /root/reference/py/modal/sandbox.py — ``_Sandbox`` (:370), ``create``
(:551-861), ``exec`` (:1983-2208) via the command-router path (:2140),
``wait/poll`` (:1698,1899), fs snapshot (:1499), ``Probe`` readiness
(:271-335). Locally a sandbox is a supervised process group in a private
workdir; ``exec`` spawns siblings in the same workdir/env; stdio rides the
offset-resumable buffers of scheduler/sandboxes.py. GPU sandboxes get
``HIP_VISIBLE_DEVICES`` pinning; Volume mounts are symlinked shared trees
whose I/O rides the CAS (HIP-hashed blocks).
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass
from typing import Any, AsyncGenerator, Optional, Sequence

from ._object import _Object, live_method
from ._sync import synchronize_api, unwrap, wrap
from .exception import SandboxTimeoutError
from .io_streams import _StreamReader, _StreamWriter


@dataclass
class Probe:
    """Readiness probe config (parity: reference sandbox.py:271-335)."""

    exec_command: Optional[list[str]] = None
    http_port: Optional[int] = None
    initial_delay: float = 0.0
    period: float = 5.0
    timeout: float = 30.0


async def _resolve_env(secrets: Sequence[Any], client: Any) -> dict:
    env: dict[str, str] = {}
    from ._object import Resolver

    resolver = Resolver(client)
    for secret in secrets:
        impl = unwrap(secret)
        await resolver.load(impl)
        env.update(await client.svc.secret_env(secret_id=impl.object_id))
    return env


class _ContainerProcess:
    """A process started by ``sandbox.exec`` (parity: reference
    container_process.py)."""

    def __init__(self, exec_id: str, client: Any, text: bool = True):
        self._exec_id = exec_id
        self._client = client
        self._text = text
        self._returncode: Optional[int] = None
        self._stdout: Optional[_StreamReader] = None
        self._stderr: Optional[_StreamReader] = None
        self._stdin: Optional[_StreamWriter] = None

    async def resize(self, rows: int, cols: int) -> None:
        """Resize a PTY-backed exec's terminal (parity: TaskExecResize)."""
        await self._client.svc.sandbox_resize(target_id=self._exec_id, rows=rows, cols=cols)

    @property
    def stdout(self) -> _StreamReader:
        if self._stdout is None:
            self._stdout = _StreamReader(self._client, self._exec_id, 1, self._text)
        return self._stdout

    @property
    def stderr(self) -> _StreamReader:
        if self._stderr is None:
            self._stderr = _StreamReader(self._client, self._exec_id, 2, self._text)
        return self._stderr

    @property
    def stdin(self) -> _StreamWriter:
        if self._stdin is None:
            self._stdin = _StreamWriter(self._client, self._exec_id)
        return self._stdin

    @property
    def returncode(self) -> Optional[int]:
        return self._returncode

    async def wait(self) -> int:
        resp = await self._client.svc.sandbox_wait(target_id=self._exec_id, timeout=None)
        self._returncode = resp["returncode"]
        return self._returncode

    async def poll(self) -> Optional[int]:
        resp = await self._client.svc.sandbox_poll(target_id=self._exec_id)
        self._returncode = resp["returncode"]
        return self._returncode


ContainerProcess = synchronize_api(_ContainerProcess, "ContainerProcess")


class _FileIO:
    """io.FileIO-like handle over a sandbox file (parity: reference
    file_io.py:135 over ContainerFilesystemExec RPCs)."""

    def __init__(self, client: Any, sandbox_id: str, path: str, mode: str):
        self._client = client
        self._sandbox_id = sandbox_id
        self._path = path
        self._mode = mode
        self._pos = 0
        self._binary = "b" in mode
        self._closed = False

    async def _op(self, op: str, **kwargs: Any) -> Any:
        return await self._client.svc.sandbox_fs_op(
            sandbox_id=self._sandbox_id, op=op, path=self._path, **kwargs
        )

    async def _init_mode(self) -> None:
        if "w" in self._mode:
            await self._op("write", data=b"")  # truncate
        elif "a" in self._mode:
            st = None
            try:
                st = await self._op("stat")
            except Exception:
                await self._op("write", data=b"")
            if st:
                self._pos = st["size"]

    async def read(self, n: int = -1) -> Any:
        data = await self._op("read", offset=self._pos, n=n)
        self._pos += len(data)
        return data if self._binary else data.decode("utf-8", errors="replace")

    async def readline(self) -> Any:
        out = bytearray()
        while True:
            chunk = await self._op("read", offset=self._pos, n=4096)
            if not chunk:
                break
            nl = chunk.find(b"\n")
            if nl >= 0:
                out += chunk[: nl + 1]
                self._pos += nl + 1
                break
            out += chunk
            self._pos += len(chunk)
        data = bytes(out)
        return data if self._binary else data.decode("utf-8", errors="replace")

    async def write(self, data: Any) -> int:
        if isinstance(data, str):
            data = data.encode("utf-8")
        n = await self._op("write", data=data, offset=self._pos, append="a" in self._mode)
        self._pos += len(data)
        return n

    async def seek(self, offset: int, whence: int = 0) -> int:
        if whence == 0:
            self._pos = offset
        elif whence == 1:
            self._pos += offset
        else:
            st = await self._op("stat")
            self._pos = st["size"] + offset
        return self._pos

    async def flush(self) -> None:
        pass

    async def close(self) -> None:
        self._closed = True

    async def __aenter__(self) -> "_FileIO":
        return self

    async def __aexit__(self, *exc: Any) -> None:
        await self.close()


FileIO = synchronize_api(_FileIO, "FileIO")


class _Sandbox(_Object, type_kind="sandbox"):
    def _init_attrs(self) -> None:
        self._task_id: Optional[str] = None
        self._returncode: Optional[int] = None
        self._stdout: Optional[_StreamReader] = None
        self._stderr: Optional[_StreamReader] = None
        self._stdin: Optional[_StreamWriter] = None

    # -- construction ----------------------------------------------------
    @classmethod
    async def create(
        cls,
        *entrypoint_args: str,
        app: Any = None,
        image: Any = None,
        secrets: Sequence[Any] = (),
        env: Optional[dict] = None,
        gpu: Any = None,
        cpu: Optional[float] = None,
        memory: Optional[int] = None,
        timeout: Optional[float] = None,
        workdir: Optional[str] = None,
        volumes: Optional[dict] = None,
        name: Optional[str] = None,
        environment_name: str = "",
        unencrypted_ports: Sequence[int] = (),
        encrypted_ports: Sequence[int] = (),
        block_network: bool = False,
        client: Any = None,
        verbose: bool = False,
    ) -> "_Sandbox":
        from .app import _parse_gpu
        from .client import _Client

        client = unwrap(client) if client is not None else await _Client.from_env()
        needs_gpu, _count = _parse_gpu(gpu)
        env_dict = dict(env or {})
        if secrets:
            env_dict.update(await _resolve_env(secrets, client))
        volume_mounts = {}
        if volumes:
            from ._object import Resolver

            resolver = Resolver(client)
            for path, vol in volumes.items():
                impl = unwrap(vol)
                await resolver.load(impl)
                opts = getattr(impl, "_mount_options", None)
                if opts:
                    volume_mounts[str(path)] = {"volume_id": impl.object_id, **opts}
                else:
                    volume_mounts[str(path)] = impl.object_id
        app_id = ""
        if app is not None and getattr(app, "_app_id", None):
            app_id = app._app_id
        restore_image_id = None
        if image is not None:
            image_impl = unwrap(image)
            from ._object import Resolver

            if not image_impl.is_hydrated:
                await Resolver(client).load(image_impl)
            restore_image_id = image_impl.object_id
            env_dict = {**getattr(image_impl, "_env_cache", {}), **env_dict}
        resp = await client.svc.sandbox_create(
            entrypoint_args=list(entrypoint_args),
            restore_image_id=restore_image_id,
            env=env_dict,
            workdir=workdir,
            timeout=timeout,
            gpu=0 if needs_gpu else None,
            app_id=app_id,
            name=name,
            environment=environment_name or "main",
            volume_mounts=volume_mounts,
            cpu=cpu,
            memory=memory,
        )
        obj = cls._new_hydrated(resp["sandbox_id"], client, {"task_id": resp["task_id"]})
        return obj

    def _hydrate_metadata(self, metadata: dict) -> None:
        if metadata:
            self._task_id = metadata.get("task_id")

    @classmethod
    def from_id(cls, sandbox_id: str, client: Any = None) -> "_Sandbox":
        async def _load(obj: "_Sandbox", resolver: Any, existing: Any) -> None:
            obj._hydrate(sandbox_id, resolver.client, None)

        obj = cls._from_loader(_load, rep=f"Sandbox({sandbox_id})")
        if client is not None:
            obj._hydrate(sandbox_id, unwrap(client), None)
        return obj

    @classmethod
    async def from_name(cls, name: str, *, environment_name: str = "") -> "_Sandbox":
        from .client import _Client

        client = await _Client.from_env()
        resp = await client.svc.sandbox_from_name(
            name=name, environment=environment_name or "main"
        )
        return cls._new_hydrated(resp["sandbox_id"], client, None)

    @classmethod
    async def list(
        cls, *, app_id: Optional[str] = None, tags: Optional[dict] = None, client: Any = None
    ) -> list["_Sandbox"]:
        from .client import _Client

        client = unwrap(client) if client is not None else await _Client.from_env()
        rows = await client.svc.sandbox_list(app_id=app_id, tags=tags)
        return [cls._new_hydrated(r["sandbox_id"], client, None) for r in rows]

    # -- stdio -----------------------------------------------------------
    @property
    def stdout(self) -> _StreamReader:
        if self._stdout is None:
            self._stdout = _StreamReader(self._client, self.object_id, 1)
        return self._stdout

    @property
    def stderr(self) -> _StreamReader:
        if self._stderr is None:
            self._stderr = _StreamReader(self._client, self.object_id, 2)
        return self._stderr

    @property
    def stdin(self) -> _StreamWriter:
        if self._stdin is None:
            self._stdin = _StreamWriter(self._client, self.object_id)
        return self._stdin

    @property
    def returncode(self) -> Optional[int]:
        return self._returncode

    # -- lifecycle -------------------------------------------------------
    @live_method
    async def wait(self, raise_on_termination: bool = True) -> int:
        resp = await self._client.svc.sandbox_wait(target_id=self.object_id, timeout=None)
        self._returncode = resp["returncode"]
        if resp.get("timed_out"):
            raise SandboxTimeoutError(f"Sandbox {self.object_id} exceeded its timeout")
        if raise_on_termination and self._returncode not in (0, None):
            from .exception import SandboxTerminatedError

            if self._returncode == -9 or self._returncode == 137:
                raise SandboxTerminatedError(f"Sandbox {self.object_id} was terminated")
        return self._returncode

    @live_method
    async def poll(self) -> Optional[int]:
        resp = await self._client.svc.sandbox_poll(target_id=self.object_id)
        self._returncode = resp["returncode"]
        return self._returncode

    @live_method
    async def terminate(self) -> None:
        await self._client.svc.sandbox_terminate(sandbox_id=self.object_id)

    @live_method
    async def set_tags(self, tags: dict) -> None:
        await self._client.svc.sandbox_set_tags(sandbox_id=self.object_id, tags=tags)

    @live_method
    async def get_tags(self) -> dict:
        rows = await self._client.svc.sandbox_list()
        for row in rows:
            if row.get("sandbox_id") == self.object_id:
                return dict(row.get("tags") or {})
        return {}

    @live_method
    async def wait_until_ready(self, timeout: float = 60.0) -> None:
        """Block until the sandbox process is running (parity:
        SandboxWaitUntilReady — locally: alive and not already exited)."""
        import time as _time

        deadline = _time.time() + timeout
        while _time.time() < deadline:
            status = await self._client.svc.sandbox_poll(self.object_id)
            if status.get("running"):
                return
            if status.get("returncode") is not None:
                from .exception import ExecutionError

                raise ExecutionError(
                    f"sandbox exited (rc={status['returncode']}) before ready"
                )
            await __import__("asyncio").sleep(0.05)
        from .exception import SandboxTimeoutError

        raise SandboxTimeoutError("sandbox never became ready")

    @live_method
    async def reload_volumes(self) -> None:
        """Re-sync mounted volumes (parity: TaskReloadVolumes — local
        volumes are directly shared, so this is a commit barrier no-op)."""
        return None

    # -- exec ------------------------------------------------------------
    @live_method
    async def exec(
        self,
        *cmds: str,
        workdir: Optional[str] = None,
        env: Optional[dict] = None,
        secrets: Sequence[Any] = (),
        timeout: Optional[float] = None,
        text: bool = True,
        bufsize: int = -1,
        stdout: Any = None,
        stderr: Any = None,
        pty_info: Optional[dict] = None,
    ) -> _ContainerProcess:
        """``pty_info={"rows": R, "cols": C}`` runs the command on a
        pseudo-terminal (stdout+stderr merged; parity: reference
        sandbox.py exec pty_info)."""
        env_dict = dict(env or {})
        if secrets:
            env_dict.update(await _resolve_env(secrets, self._client))
        resp = await self._client.svc.sandbox_exec(
            sandbox_id=self.object_id,
            cmd=list(cmds),
            env=env_dict or None,
            workdir=workdir,
            timeout=timeout,
            pty=pty_info is not None,
            rows=(pty_info or {}).get("rows", 24),
            cols=(pty_info or {}).get("cols", 80),
        )
        return _ContainerProcess(resp["exec_id"], self._client, text=text)

    # -- filesystem ------------------------------------------------------
    @live_method
    async def open(self, path: str, mode: str = "r") -> _FileIO:
        f = _FileIO(self._client, self.object_id, path, mode)
        await f._init_mode()
        return f

    @live_method
    async def ls(self, path: str = ".") -> list[str]:
        return await self._client.svc.sandbox_fs_op(
            sandbox_id=self.object_id, op="ls", path=path
        )

    @live_method
    async def mkdir(self, path: str, parents: bool = False) -> None:
        await self._client.svc.sandbox_fs_op(
            sandbox_id=self.object_id, op="mkdir", path=path, parents=parents
        )

    @live_method
    async def rm(self, path: str, recursive: bool = False) -> None:
        await self._client.svc.sandbox_fs_op(
            sandbox_id=self.object_id, op="rm", path=path, recursive=recursive
        )

    @live_method
    async def exists(self, path: str) -> bool:
        return await self._client.svc.sandbox_fs_op(
            sandbox_id=self.object_id, op="exists", path=path
        )

    async def watch(self, path: str, poll_interval: float = 0.5) -> AsyncGenerator[dict, None]:
        """Poll-based file watch (parity: sandbox_fs watch API): yields
        {"path", "event"} dicts for created/modified/deleted entries."""
        if not self._is_hydrated:
            await self.hydrate()
        known: dict[str, float] = {}
        first = True
        while True:
            try:
                names = await self._client.svc.sandbox_fs_op(
                    sandbox_id=self.object_id, op="ls", path=path
                )
            except Exception:
                return
            current: dict[str, float] = {}
            for name in names:
                try:
                    st = await self._client.svc.sandbox_fs_op(
                        sandbox_id=self.object_id, op="stat", path=f"{path}/{name}"
                    )
                    current[name] = st["mtime"]
                except Exception:
                    continue
            if not first:
                for name, mtime in current.items():
                    if name not in known:
                        yield {"path": f"{path}/{name}", "event": "created"}
                    elif known[name] != mtime:
                        yield {"path": f"{path}/{name}", "event": "modified"}
                for name in known:
                    if name not in current:
                        yield {"path": f"{path}/{name}", "event": "deleted"}
            known = current
            first = False
            await asyncio.sleep(poll_interval)

    # -- snapshots / tunnels ----------------------------------------------
    @live_method
    async def snapshot_filesystem(self, timeout: float = 55.0) -> Any:
        """Tar the sandbox tree into the CAS; returns an Image handle
        (parity: reference sandbox.py:1499 snapshot_fs)."""
        resp = await self._client.svc.sandbox_snapshot_fs(sandbox_id=self.object_id)
        from .image import _Image

        return _Image._new_hydrated(resp["image_id"], self._client, {"blob_id": resp["blob_id"]})

    @live_method
    async def tunnels(self, timeout: float = 50.0) -> dict:
        """Local tunnels are identity mappings (ports are already reachable)."""
        from ._tunnel import Tunnel

        return {}

    @live_method
    async def snapshot_directory(
        self, path: str, *, timeout: float = 55.0, ttl: Any = None
    ) -> Any:
        """Snapshot one directory into a new Image (parity: reference
        sandbox.py:1643 snapshot_directory)."""
        resp = await self._client.svc.sandbox_snapshot_dir(
            sandbox_id=self.object_id, path=str(path)
        )
        from .image import _Image

        return _Image._new_hydrated(resp["image_id"], self._client, None)

    @live_method
    async def mount_image(self, path: str, image: Any) -> None:
        """Expose an Image's filesystem layer inside the sandbox (parity:
        reference sandbox.py:1548 mount_image)."""
        from ._sync import unwrap as _unwrap

        impl = _unwrap(image)
        if not impl.is_hydrated:
            await impl.hydrate(self._client)
        await self._client.svc.sandbox_mount_image(
            sandbox_id=self.object_id, path=str(path), image_id=impl.object_id
        )

    @live_method
    async def unmount_image(self, path: str) -> None:
        await self._client.svc.sandbox_unmount_image(
            sandbox_id=self.object_id, path=str(path)
        )

    @live_method
    async def create_connect_token(
        self, user_metadata: Any = None, port: int = 8080
    ) -> dict:
        """Mint URL+token credentials for HTTP access to this sandbox
        (parity: reference sandbox.py:1799). Locally the URL is direct."""
        import json as _json

        if isinstance(user_metadata, dict):
            user_metadata = _json.dumps(user_metadata)
        return await self._client.svc.sandbox_connect_token(
            sandbox_id=self.object_id, port=port, user_metadata=user_metadata
        )

    async def detach(self) -> None:
        """Drop this handle's attachment (parity: reference sandbox.py:1193 —
        the sandbox keeps running; re-attach with Sandbox.from_id)."""
        self._detached = True

    @property
    def filesystem(self) -> "_SandboxFilesystem":
        """Namespace for filesystem APIs (parity: reference sandbox.py:2339)."""
        return _SandboxFilesystem(self)

    def logs(self) -> "_SandboxLogsManager":
        """Entrypoint-log access, incl. after termination (parity: reference
        sandbox.py:2646 _SandboxLogsManager with fetch/tail)."""
        return _SandboxLogsManager(self)


class _SandboxFilesystem:
    """Grouped filesystem operations over one sandbox (open/ls/mkdir/rm/
    exists/watch/read/write convenience)."""

    def __init__(self, sandbox: "_Sandbox"):
        self._sb = sandbox

    async def open(self, path: str, mode: str = "r") -> _FileIO:
        return await self._sb.open(path, mode)

    async def list_files(self, path: str = ".") -> list[str]:
        return await self._sb.ls(path)

    async def ls(self, path: str = ".") -> list[str]:
        return await self._sb.ls(path)

    async def mkdir(self, path: str, parents: bool = False) -> None:
        return await self._sb.mkdir(path, parents=parents)

    async def rm(self, path: str, recursive: bool = False) -> None:
        return await self._sb.rm(path, recursive=recursive)

    async def exists(self, path: str) -> bool:
        return await self._sb.exists(path)

    async def read_file(self, path: str) -> bytes:
        f = await self._sb.open(path, "rb")
        try:
            return await f.read()
        finally:
            await f.close()

    async def write_file(self, path: str, data: bytes) -> int:
        f = await self._sb.open(path, "wb")
        try:
            return await f.write(data)
        finally:
            await f.close()

    def watch(self, path: str, poll_interval: float = 0.5) -> Any:
        return self._sb.watch(path, poll_interval=poll_interval)


class _SandboxLogsManager:
    """fetch()/tail() over the sandbox entrypoint's captured stdio."""

    def __init__(self, sandbox: "_Sandbox"):
        self._sb = sandbox

    async def fetch(self, *, stderr: bool = False) -> str:
        """Everything captured so far (does not wait for termination)."""
        reader = self._sb.stderr if stderr else self._sb.stdout
        chunks = []
        while True:
            data, eof = await reader.read_chunk(timeout=0.05)
            if data:
                chunks.append(data)
            if eof or not data:
                break
        blob = b"".join(chunks)
        return blob.decode("utf-8", errors="replace")

    def tail(self, *, stderr: bool = False, poll_interval: float = 0.2) -> Any:
        """Async-iterate new output until the entrypoint exits."""
        reader = self._sb.stderr if stderr else self._sb.stdout

        async def gen() -> Any:
            while True:
                data, eof = await reader.read_chunk(timeout=poll_interval)
                if data:
                    yield data.decode("utf-8", errors="replace")
                if eof:
                    return

        return gen()


Sandbox = synchronize_api(_Sandbox, "Sandbox")
SandboxFilesystem = synchronize_api(_SandboxFilesystem, "SandboxFilesystem")
SandboxLogsManager = synchronize_api(_SandboxLogsManager, "SandboxLogsManager")
