"""Mount: content-deduplicated file bundles shipped to workers.

Parity: /root/reference/py/modal/mount.py — ``_Mount`` (:290), entry types
(:89-283), content-dedup key (:324), MountPutFile/MountGetOrCreate. Files
hash into the CAS (sharing the HIP sha256 path for large files) and
materialize once per content set; python-source mounts land on workers'
``sys.path``.
"""

from __future__ import annotations

import os
from typing import Any, Callable, Optional, Union

from ._object import _Object
from ._sync import synchronize_api
from .exception import InvalidError, ModuleNotMountable


class _Mount(_Object, type_kind="mount"):
    _entries: list

    def _init_attrs(self) -> None:
        self._entries = []  # (local_path, remote_path)
        self._dir: Optional[str] = None
        self._is_python_source = False

    @classmethod
    def _from_entries(cls, entries: list, rep: str, is_python_source: bool = False) -> "_Mount":
        async def _load(obj: "_Mount", resolver: Any, existing: Any) -> None:
            manifest = []
            for local, remote in obj._entries:
                with open(local, "rb") as f:
                    put = await resolver.client.svc.blob_put(data=f.read())
                manifest.append([remote, put["blob_id"], 0o644])
            resp = await resolver.client.svc.mount_get_or_create(manifest=manifest)
            obj._dir = resp["dir"]
            obj._hydrate(resp["mount_id"], resolver.client, {"dir": resp["dir"]})

        obj = cls._from_loader(_load, rep=rep)
        obj._entries = entries
        obj._is_python_source = is_python_source
        return obj

    def _hydrate_metadata(self, metadata: dict) -> None:
        if metadata:
            self._dir = metadata.get("dir")

    @classmethod
    def from_local_file(
        cls, local_path: Union[str, os.PathLike], remote_path: Optional[str] = None
    ) -> "_Mount":
        local_path = str(local_path)
        remote_path = remote_path or f"/root/{os.path.basename(local_path)}"
        return cls._from_entries(
            [(local_path, remote_path)], rep=f"Mount.from_local_file({local_path!r})"
        )

    @classmethod
    def from_local_dir(
        cls,
        local_path: Union[str, os.PathLike],
        *,
        remote_path: Optional[str] = None,
        condition: Optional[Callable[[str], bool]] = None,
        recursive: bool = True,
    ) -> "_Mount":
        local_path = str(local_path)
        if not os.path.isdir(local_path):
            raise InvalidError(f"{local_path} is not a directory")
        remote_path = remote_path or f"/root/{os.path.basename(local_path.rstrip('/'))}"
        entries = []
        for dirpath, _dirnames, filenames in os.walk(local_path):
            for fn in filenames:
                full = os.path.join(dirpath, fn)
                if condition is not None and not condition(full):
                    continue
                rel = os.path.relpath(full, local_path)
                entries.append((full, os.path.join(remote_path, rel)))
            if not recursive:
                break
        return cls._from_entries(entries, rep=f"Mount.from_local_dir({local_path!r})")

    @classmethod
    def from_local_python_packages(
        cls,
        *module_names: str,
        condition: Optional[Callable[[str], bool]] = None,
        ignore: Any = None,
    ) -> "_Mount":
        import importlib.util

        entries = []
        for mod in module_names:
            try:
                spec = importlib.util.find_spec(mod)
            except ModuleNotFoundError:
                spec = None
            if spec is None or not spec.origin:
                raise ModuleNotMountable(f"Module {mod!r} not found locally")
            if spec.submodule_search_locations:
                pkg_dir = os.path.dirname(spec.origin)
                for dirpath, _dn, filenames in os.walk(pkg_dir):
                    if "__pycache__" in dirpath:
                        continue
                    for fn in filenames:
                        if fn.endswith(".pyc"):
                            continue
                        full = os.path.join(dirpath, fn)
                        rel = os.path.relpath(full, os.path.dirname(pkg_dir))
                        entries.append((full, f"/pysource/{rel}"))
            else:
                entries.append((spec.origin, f"/pysource/{mod}.py"))
        return cls._from_entries(
            entries, rep=f"Mount.from_local_python_packages{module_names}", is_python_source=True
        )

    @property
    def entries(self) -> list:
        return list(self._entries)


Mount = synchronize_api(_Mount, "Mount")
