"""NetworkFileSystem (legacy): aliased onto the Volume service.

Parity: /root/reference/py/modal/network_file_system.py:361 — the reference
keeps NFS for backward compatibility; SURVEY.md §2 row 30 allows aliasing it
to Volume. Same API names (from_name/ephemeral/write_file/read_file/listdir/
remove_file) over the volume backend.
"""

from __future__ import annotations

from typing import Any, BinaryIO

from ._object import live_method
from ._sync import synchronize_api
from .volume import _Volume


class _NetworkFileSystem(_Volume, type_kind="nfs"):
    @classmethod
    def from_name(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_NetworkFileSystem":
        async def _load(obj: "_NetworkFileSystem", resolver: Any, existing: Any) -> None:
            resp = await resolver.client.svc.volume_get_or_create(
                name=f"nfs/{name}",
                environment=environment_name or "main",
                create_if_missing=create_if_missing,
                ephemeral=False,
            )
            obj._hydrate(resp["volume_id"], resolver.client, None)

        return cls._from_loader(_load, rep=f"NetworkFileSystem.from_name({name!r})")

    @classmethod
    async def create_deployed(
        cls, deployment_name: str, *, environment_name: str = ""
    ) -> str:
        """Eagerly create a named NFS; returns its id (parity: reference
        network_file_system.py:197 create_deployed)."""
        from .client import _Client

        client = await _Client.from_env()
        resp = await client.svc.volume_get_or_create(
            name=f"nfs/{deployment_name}",
            environment=environment_name or "main",
            create_if_missing=True,
            ephemeral=False,
        )
        return resp["volume_id"]

    @live_method
    async def write_file(self, remote_path: str, fp: BinaryIO) -> int:
        data = fp.read()
        await self._put_data(data, remote_path)
        return len(data)

    @live_method
    async def add_local_file(
        self, local_path: "Any", remote_path: "Any" = None
    ) -> None:
        """Upload one local file (parity: reference NFS.add_local_file)."""
        import os as _os

        local_path = _os.fspath(local_path)
        if remote_path is None:
            remote_path = "/" + _os.path.basename(local_path)
        with open(local_path, "rb") as f:
            await self._put_data(f.read(), str(remote_path))

    @live_method
    async def add_local_dir(
        self, local_path: "Any", remote_path: "Any" = None
    ) -> None:
        """Upload a directory tree (parity: reference NFS.add_local_dir)."""
        import os as _os

        local_path = _os.fspath(local_path)
        if remote_path is None:
            remote_path = "/" + _os.path.basename(local_path.rstrip("/"))
        for root, _dirs, files in _os.walk(local_path):
            for fname in files:
                src = _os.path.join(root, fname)
                rel = _os.path.relpath(src, local_path)
                with open(src, "rb") as f:
                    await self._put_data(f.read(), str(remote_path).rstrip("/") + "/" + rel)


NetworkFileSystem = synchronize_api(_NetworkFileSystem, "NetworkFileSystem")
