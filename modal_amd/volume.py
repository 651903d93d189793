"""Volume: committed shared filesystem handles.

Parity: /root/reference/py/modal/volume.py — ``_Volume`` (:349),
commit/reload (:752,770), streamed ``read_file`` (:837-963),
``batch_upload`` (:1027), v2 per-8 MiB-block upload with missing-block
negotiation (:1401-1525). Blocks are hashed with the HIP sha256 kernel when
large (ops/hashing) and deduplicated through the content-addressed store.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, AsyncGenerator, BinaryIO, Optional, Union

from ._object import _Object, live_method
from ._sync import synchronize_api, synchronizer, wrap
from .exception import InvalidError
from .scheduler.blobs import BLOCK_SIZE


@dataclass
class FileEntry:
    """Directory entry (parity: modal.volume.FileEntry)."""

    path: str
    type: str  # "file" | "dir"
    size: int
    mtime: float

    @property
    def is_dir(self) -> bool:
        return self.type == "dir"


class _Volume(_Object, type_kind="volume"):
    @classmethod
    def from_name(
        cls,
        name: str,
        *,
        environment_name: str = "",
        create_if_missing: bool = False,
        version: Any = None,
    ) -> "_Volume":
        async def _load(obj: "_Volume", resolver: Any, existing: Any) -> None:
            resp = await resolver.client.svc.volume_get_or_create(
                name=name,
                environment=environment_name or "main",
                create_if_missing=create_if_missing,
                ephemeral=False,
            )
            obj._hydrate(
                resp["volume_id"], resolver.client,
                {"version": resp.get("version"), "name": name},
            )

        return cls._from_loader(_load, rep=f"Volume.from_name({name!r})")

    @classmethod
    def from_id(cls, object_id: str, client: Any = None) -> "_Volume":
        async def _load(obj: "_Volume", resolver: Any, existing: Any) -> None:
            obj._hydrate(object_id, resolver.client, None)

        obj = cls._from_loader(_load, rep=f"Volume.from_id({object_id!r})")
        if client is not None:
            obj._hydrate(object_id, client, None)
        return obj

    @property
    def name(self) -> Any:
        return (getattr(self, "_metadata", None) or {}).get("name")

    # -- mount options (parity: reference volume.py:419 read_only,
    #    :450 with_mount_options) -----------------------------------------
    def with_mount_options(
        self, *, read_only: Any = None, sub_path: Any = None
    ) -> "_Volume":
        """A derived handle with per-mount options. `read_only` blocks
        client-side writes AND makes function/sandbox mounts skip the
        exit-time commit; `sub_path` scopes the mount to a subdirectory."""
        base = self
        opts = dict(getattr(self, "_mount_options", None) or {})
        if read_only is not None:
            opts["read_only"] = bool(read_only)
        if sub_path is not None:
            opts["sub_path"] = str(sub_path).strip("/")

        async def _load(obj: "_Volume", resolver: Any, existing: Any) -> None:
            await resolver.load(base)
            obj._hydrate(base._object_id, base._client, getattr(base, "_metadata", None))
            obj._mount_options = opts

        derived = _Volume._from_loader(
            _load, rep=f"{self._rep}.with_mount_options({opts})", deps=lambda: [base]
        )
        if self._is_hydrated:
            derived._hydrate(self._object_id, self._client, getattr(self, "_metadata", None))
        derived._mount_options = opts
        return derived

    def read_only(self) -> "_Volume":
        return self.with_mount_options(read_only=True)

    @property
    def is_read_only(self) -> bool:
        return bool((getattr(self, "_mount_options", None) or {}).get("read_only"))

    def _check_writable(self) -> None:
        if self.is_read_only:
            raise InvalidError("Volume handle is read-only (with_mount_options)")

    @live_method
    async def info(self) -> dict:
        """Name + file count (parity: reference info())."""
        return await self._client.svc.volume_info(volume_id=self.object_id)

    @classmethod
    async def lookup(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_Volume":
        obj = cls.from_name(
            name, environment_name=environment_name, create_if_missing=create_if_missing
        )
        return await obj.hydrate()

    @classmethod
    def ephemeral(cls, *, environment_name: str = "") -> "_EphemeralVolume":
        return _EphemeralVolume(environment_name)

    @classmethod
    async def delete(cls, name: str, *, environment_name: str = "") -> None:
        from .client import _Client

        client = await _Client.from_env()
        resp = await client.svc.volume_get_or_create(
            name=name, environment=environment_name or "main", create_if_missing=False, ephemeral=False
        )
        await client.svc.volume_delete(volume_id=resp["volume_id"])

    @classmethod
    async def rename(cls, old_name: str, new_name: str, *, environment_name: str = "") -> None:
        from .client import _Client

        client = await _Client.from_env()
        resp = await client.svc.volume_get_or_create(
            name=old_name, environment=environment_name or "main", create_if_missing=False, ephemeral=False
        )
        await client.svc.volume_rename(
            volume_id=resp["volume_id"], new_name=new_name, environment=environment_name or "main"
        )

    # -- reads -----------------------------------------------------------
    async def read_file(self, path: str) -> AsyncGenerator[bytes, None]:
        """Stream a file's contents in 8 MiB blocks (parity: reference
        read_file block streaming, volume.py:837-963).

        Same-node fast path: the volume tree lives on this filesystem, so
        read straight from it (file-cache speed) instead of shuttling every
        block through the control socket; the RPC loop remains the
        fallback for any client without filesystem access."""
        if not self._is_hydrated:
            await self.hydrate()
        try:
            vol_dir = await self._client.svc.volume_dir(volume_id=self.object_id)
        except Exception:
            vol_dir = None
        if vol_dir and os.path.isdir(vol_dir):
            full = os.path.normpath(os.path.join(vol_dir, path.lstrip("/")))
            if full.startswith(os.path.abspath(vol_dir)) and os.path.isfile(full):
                import asyncio as _asyncio

                loop = _asyncio.get_running_loop()
                read_size = 4 * BLOCK_SIZE  # local reads: fewer executor hops
                with open(full, "rb") as f:
                    while True:
                        chunk = await loop.run_in_executor(None, f.read, read_size)
                        if not chunk:
                            return
                        yield chunk
                        if len(chunk) < read_size:
                            return
        offset = 0
        while True:
            chunk = await self._client.svc.volume_get_file(
                volume_id=self.object_id, rel_path=path, offset=offset, n_bytes=BLOCK_SIZE
            )
            if not chunk:
                if offset == 0:
                    # distinguish empty file from missing (service raises on missing)
                    await self._client.svc.volume_get_file(
                        volume_id=self.object_id, rel_path=path, offset=0, n_bytes=1
                    )
                return
            yield chunk
            if len(chunk) < BLOCK_SIZE:
                return
            offset += len(chunk)

    async def read_file_into_fileobj(self, path: str, fileobj: BinaryIO) -> int:
        """Parity alias (reference read_file_into_fileobj)."""
        return await self.read_file_into(path, fileobj)

    @live_method
    async def read_file_into(self, path: str, fileobj: BinaryIO) -> int:
        """Same-node fast path: one executor call does the whole
        readinto-loop (single buffer reused, one copy into fileobj) instead
        of bridging every 32 MiB chunk through the event loop — ~2x the
        streaming generator for large files."""
        try:
            vol_dir = await self._client.svc.volume_dir(volume_id=self.object_id)
        except Exception:
            vol_dir = None
        if vol_dir and os.path.isdir(vol_dir):
            full = os.path.normpath(os.path.join(vol_dir, path.lstrip("/")))
            if full.startswith(os.path.abspath(vol_dir)) and os.path.isfile(full):
                import asyncio as _asyncio

                def _drain() -> int:
                    size = os.stat(full).st_size
                    # zero-copy into a real file: kernel-side sendfile
                    fd = None
                    if hasattr(fileobj, "fileno"):
                        try:
                            fd = fileobj.fileno()
                        except (OSError, ValueError, AttributeError):
                            fd = None
                    if fd is not None:
                        sent = 0
                        with open(full, "rb", buffering=0) as f:
                            fileobj.flush()
                            src = f.fileno()
                            while sent < size:
                                n = os.sendfile(fd, src, sent, size - sent)
                                if n == 0:
                                    break
                                sent += n
                        fileobj.seek(0, os.SEEK_END)
                        return sent
                    # single-copy into a BytesIO: read straight into its buffer
                    if hasattr(fileobj, "getbuffer") and hasattr(fileobj, "seek") and size:
                        start = fileobj.tell()
                        # extend the BytesIO to its final size in one step
                        fileobj.seek(start + size - 1)
                        fileobj.write(b"\0")
                        if fileobj.getbuffer().nbytes >= start + size:
                            view = fileobj.getbuffer()[start : start + size]
                            got = 0
                            with open(full, "rb", buffering=0) as f:
                                while got < size:
                                    n = f.readinto(view[got:])
                                    if not n:
                                        break
                                    got += n
                            view.release()
                            fileobj.seek(start + got)
                            return got
                    total = 0
                    buf = bytearray(64 * 1024 * 1024)
                    view = memoryview(buf)
                    with open(full, "rb", buffering=0) as f:
                        while True:
                            n = f.readinto(buf)
                            if not n:
                                return total
                            fileobj.write(view[:n])
                            total += n

                return await _asyncio.get_running_loop().run_in_executor(None, _drain)
        total = 0
        async for chunk in self.read_file(path):
            fileobj.write(chunk)
            total += len(chunk)
        return total

    @live_method
    async def listdir(self, path: str = "/", *, recursive: bool = False) -> list[FileEntry]:
        entries = await self._client.svc.volume_list_files(
            volume_id=self.object_id, rel_path=path, recursive=recursive
        )
        return [FileEntry(e["path"], e["type"], e["size"], e["mtime"]) for e in entries]

    async def iterdir(self, path: str = "/", *, recursive: bool = True) -> AsyncGenerator[FileEntry, None]:
        if not self._is_hydrated:
            await self.hydrate()
        for entry in await self.listdir(path, recursive=recursive):
            yield entry

    # -- writes ----------------------------------------------------------
    @live_method
    async def remove_file(self, path: str, recursive: bool = False) -> None:
        self._check_writable()
        await self._client.svc.volume_remove_file(
            volume_id=self.object_id, rel_path=path, recursive=recursive
        )

    @live_method
    async def copy_files(self, src_paths: list[str], dst_path: str) -> None:
        await self._client.svc.volume_copy_files(
            volume_id=self.object_id, src_paths=list(src_paths), dst_path=dst_path
        )

    @live_method
    async def commit(self) -> None:
        await self._client.svc.volume_commit(volume_id=self.object_id)

    @live_method
    async def reload(self) -> None:
        await self._client.svc.volume_reload(volume_id=self.object_id)

    def batch_upload(self, force: bool = False) -> "_VolumeUploadContextManager":
        self._check_writable()
        return _VolumeUploadContextManager(self, force=force)

    async def _put_data(self, data: bytes, remote_path: str) -> None:
        """Block-wise upload through the CAS: hash each 8 MiB block (HIP
        kernel above the crossover), store, then commit the manifest."""
        import asyncio

        store = self._client.blob_store
        # memoryview slices: no second copy of the whole payload (every
        # consumer — np.frombuffer staging, hashlib, file writes — takes
        # buffer objects)
        mv = memoryview(data)
        blocks = [
            mv[off : off + BLOCK_SIZE] for off in range(0, max(len(data), 1), BLOCK_SIZE)
        ]
        loop = asyncio.get_running_loop()
        # same-node fast path: stage the raw file next to the volume tree so
        # the service renames it into place instead of re-reading every CAS
        # block (and GPU-decompressing the compressed ones); the disk write
        # overlaps the GPU hash+compress pass below
        stage_task = None
        run_dir = getattr(self._client, "run_dir", None) or getattr(
            self._client.svc, "run_dir", None
        )
        if run_dir and os.path.isdir(run_dir):

            def _stage() -> str:
                import tempfile as _tf

                fd, tmp = _tf.mkstemp(dir=run_dir, prefix=".volstage-")
                with os.fdopen(fd, "wb") as f:
                    f.write(data)
                return tmp

            stage_task = loop.run_in_executor(None, _stage)
        # hash (one batched GPU dispatch) + batched compression + CAS writes
        digests = await loop.run_in_executor(None, store.put_many, blocks)
        content_tmp = await stage_task if stage_task is not None else None
        resp = await self._client.svc.volume_put_file_blocks(
            volume_id=self.object_id,
            rel_path=remote_path,
            block_digests=digests,
            size=len(data),
            content_tmp=content_tmp,
        )
        if resp.get("missing_blocks"):
            raise InvalidError(f"blocks missing after upload: {resp['missing_blocks']}")


class _VolumeUploadContextManager:
    """``with vol.batch_upload() as b: b.put_file(...)`` (parity: reference
    _VolumeUploadContextManager)."""

    def __init__(self, volume: _Volume, force: bool = False):
        self._volume = volume
        self._force = force
        self._jobs: list[tuple[str, str]] = []  # (local, remote) files
        self._data_jobs: list[tuple[bytes, str]] = []

    def put_file(self, local_file: Union[str, os.PathLike, BinaryIO], remote_path: str) -> None:
        if hasattr(local_file, "read"):
            self._data_jobs.append((local_file.read(), str(remote_path)))
        else:
            self._jobs.append((str(local_file), str(remote_path)))

    def put_directory(self, local_path: Union[str, os.PathLike], remote_path: str, recursive: bool = True) -> None:
        local_path = str(local_path)
        for dirpath, _dirnames, filenames in os.walk(local_path):
            for fn in filenames:
                full = os.path.join(dirpath, fn)
                rel = os.path.relpath(full, local_path)
                self._jobs.append((full, os.path.join(str(remote_path), rel)))
            if not recursive:
                break

    async def _commit(self) -> None:
        if not self._volume._is_hydrated:
            await self._volume.hydrate()
        for local, remote in self._jobs:
            with open(local, "rb") as f:
                await self._volume._put_data(f.read(), remote)
        for data, remote in self._data_jobs:
            await self._volume._put_data(data, remote)
        await self._volume._client.svc.volume_commit(volume_id=self._volume.object_id)

    def __enter__(self) -> "_VolumeUploadContextManager":
        return self

    def __exit__(self, exc_type: Any, *exc: Any) -> None:
        if exc_type is None:
            synchronizer.run(self._commit())

    async def __aenter__(self) -> "_VolumeUploadContextManager":
        return self

    async def __aexit__(self, exc_type: Any, *exc: Any) -> None:
        if exc_type is None:
            await synchronizer.run_async(self._commit())


class _EphemeralVolume:
    def __init__(self, environment_name: str):
        self.environment_name = environment_name
        self._impl: Optional[_Volume] = None

    async def _create(self) -> _Volume:
        from .client import _Client

        client = await _Client.from_env()
        resp = await client.svc.volume_get_or_create(
            name=None, environment=self.environment_name or "main", create_if_missing=True, ephemeral=True
        )
        impl = _Volume._new_hydrated(resp["volume_id"], client, None)
        self._impl = impl
        return impl

    async def _cleanup(self) -> None:
        if self._impl is not None:
            await self._impl._client.svc.volume_delete(volume_id=self._impl.object_id)

    def __enter__(self) -> Any:
        return wrap(synchronizer.run(self._create()))

    def __exit__(self, *exc: Any) -> None:
        synchronizer.run(self._cleanup())

    async def __aenter__(self) -> Any:
        return wrap(await synchronizer.run_async(self._create()))

    async def __aexit__(self, *exc: Any) -> None:
        await synchronizer.run_async(self._cleanup())


Volume = synchronize_api(_Volume, "Volume")

from .object_manager import install as _install_manager  # noqa: E402

_install_manager(_Volume, Volume, "volume")
