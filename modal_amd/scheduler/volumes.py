"""Volume service: committed shared filesystems with a block-hashed manifest.

Single-node re-implementation of the reference's Volume v2
(/root/reference/py/modal/volume.py:349; v2 block path :1401-1525,
blob_utils.py:534-721): files are written through the content-addressed
store — per-8 MiB-block SHA-256 (the HIP kernel for large blocks via
ops/hashing) with a manifest per file — then materialized into the volume
tree with hard links when possible. Missing-block negotiation collapses to
CAS presence checks (same filesystem, no second phase needed). Trailing-zero
awareness: all-zero blocks share one CAS entry automatically by content
address.
"""

from __future__ import annotations

import os
import shutil
import time
from typing import Optional

from ..exception import AlreadyExistsError, InvalidError, NotFoundError
from ..utils.ids import new_id
from .blobs import BlobStore


class VolumeState:
    def __init__(self, volume_id: str, name: Optional[str], root: str):
        self.volume_id = volume_id
        self.name = name
        self.root = root
        self.version = 1
        self.commit_count = 0
        self.last_commit = 0.0
        # manifest: rel_path -> {"size": int, "blocks": [sha256 hex, ...], "mtime": float}
        self.manifest: dict[str, dict] = {}


class VolumeService:
    def __init__(self, run_dir: str, blob_store: BlobStore):
        self.run_dir = run_dir
        self.root = os.path.join(run_dir, "volumes")
        os.makedirs(self.root, exist_ok=True)
        self.blob_store = blob_store
        self.volumes: dict[str, VolumeState] = {}
        self.by_name: dict[tuple[str, str], str] = {}

    def _get(self, volume_id: str) -> VolumeState:
        v = self.volumes.get(volume_id)
        if v is None:
            raise NotFoundError(f"Volume {volume_id} not found")
        return v

    def _safe_path(self, vol: VolumeState, rel: str) -> str:
        rel = rel.lstrip("/")
        path = os.path.normpath(os.path.join(vol.root, rel))
        if not path.startswith(os.path.abspath(vol.root)):
            raise InvalidError(f"Path escapes volume: {rel!r}")
        return path

    # -- lifecycle -------------------------------------------------------
    async def get_or_create(
        self, name: Optional[str], environment: str, create_if_missing: bool, ephemeral: bool
    ) -> dict:
        if name and not ephemeral:
            existing = self.by_name.get((environment, name))
            if existing:
                return {"volume_id": existing, "version": self.volumes[existing].version}
            if not create_if_missing:
                raise NotFoundError(f"Volume '{name}' not found")
        vid = new_id("volume")
        root = os.path.join(self.root, vid)
        os.makedirs(root, exist_ok=True)
        self.volumes[vid] = VolumeState(vid, name, root)
        if name and not ephemeral:
            self.by_name[(environment, name)] = vid
        return {"volume_id": vid, "version": 1}

    async def delete(self, volume_id: str) -> None:
        vol = self._get(volume_id)
        shutil.rmtree(vol.root, ignore_errors=True)
        self.volumes.pop(volume_id, None)
        self.by_name = {k: v for k, v in self.by_name.items() if v != volume_id}

    async def rename(self, volume_id: str, new_name: str, environment: str = "main") -> None:
        vol = self._get(volume_id)
        if (environment, new_name) in self.by_name:
            raise AlreadyExistsError(f"Volume '{new_name}' already exists")
        self.by_name = {k: v for k, v in self.by_name.items() if v != volume_id}
        self.by_name[(environment, new_name)] = volume_id
        vol.name = new_name

    # -- writes ----------------------------------------------------------
    async def put_file_blocks(
        self,
        volume_id: str,
        rel_path: str,
        block_digests: list[str],
        size: int,
        mode: int = 0o644,
        content_tmp: Optional[str] = None,
    ) -> dict:
        """Phase-2 commit of a file whose blocks are already in the CAS
        (parity: VolumePutFiles2 missing-block protocol, volume.py:1401-1445).
        Returns any blocks NOT yet in the store (client must upload & retry).
        ``content_tmp``: same-node fast path — a staged raw copy the client
        already wrote under the run dir; renamed into place instead of
        re-assembling from CAS blocks."""
        vol = self._get(volume_id)
        missing = [d for d in block_digests if not self.blob_store.has(d)]
        if missing:
            if content_tmp:
                try:
                    os.unlink(content_tmp)
                except OSError:
                    pass
            return {"missing_blocks": missing}
        dest = self._safe_path(vol, rel_path)
        os.makedirs(os.path.dirname(dest), exist_ok=True)
        staged = False
        if content_tmp and os.path.isfile(content_tmp) and os.path.getsize(content_tmp) == size:
            try:
                os.replace(content_tmp, dest)
                staged = True
            except OSError:
                pass  # cross-device etc.: fall through to block assembly
        if staged:
            pass
        elif len(block_digests) == 1:
            # single-block files come straight out of the CAS (hard link when
            # stored raw, decompressed write when LZ4-compressed)
            self.blob_store.materialize(block_digests[0], dest)
        else:
            # multi-block concat OFF the event loop, zero-copy for raw
            # blocks (this was 85% of upload wall time when done with
            # read()+write() on the loop)
            import asyncio

            def _concat() -> None:
                with open(dest, "wb") as f:
                    for digest in block_digests:
                        # open_path always yields the RAW form (decompressing
                        # a .z-stored blob once, cached); sendfile from there
                        src = self.blob_store.open_path(digest)
                        with open(src, "rb") as bf:
                            n = os.fstat(bf.fileno()).st_size
                            off = 0
                            while off < n:
                                sent = os.sendfile(f.fileno(), bf.fileno(), off, n - off)
                                if sent == 0:
                                    break
                                off += sent

            await asyncio.get_running_loop().run_in_executor(None, _concat)
        vol.manifest[rel_path.lstrip("/")] = {
            "size": size,
            "blocks": block_digests,
            "mtime": time.time(),
        }
        return {"missing_blocks": []}

    async def remove_file(self, volume_id: str, rel_path: str, recursive: bool = False) -> None:
        vol = self._get(volume_id)
        path = self._safe_path(vol, rel_path)
        if os.path.isdir(path) and not os.path.islink(path):
            if not recursive:
                raise InvalidError(f"{rel_path} is a directory (use recursive=True)")
            shutil.rmtree(path)
            prefix = rel_path.strip("/") + "/"
            vol.manifest = {k: v for k, v in vol.manifest.items() if not k.startswith(prefix)}
        elif os.path.exists(path) or os.path.islink(path):
            os.unlink(path)
            vol.manifest.pop(rel_path.lstrip("/"), None)
        else:
            raise NotFoundError(f"{rel_path} not in volume")

    async def copy_files(self, volume_id: str, src_paths: list[str], dst_path: str) -> None:
        vol = self._get(volume_id)
        dst = self._safe_path(vol, dst_path)
        for src_rel in src_paths:
            src = self._safe_path(vol, src_rel)
            if not os.path.exists(src):
                raise NotFoundError(f"{src_rel} not in volume")
            if os.path.isdir(src):
                shutil.copytree(src, os.path.join(dst, os.path.basename(src)), dirs_exist_ok=True)
            else:
                target = dst
                if os.path.isdir(dst) or dst_path.endswith("/") or len(src_paths) > 1:
                    os.makedirs(dst, exist_ok=True)
                    target = os.path.join(dst, os.path.basename(src))
                else:
                    os.makedirs(os.path.dirname(target) or vol.root, exist_ok=True)
                shutil.copyfile(src, target)

    async def commit(self, volume_id: str) -> dict:
        vol = self._get(volume_id)
        vol.commit_count += 1
        vol.last_commit = time.time()
        return {"commit_count": vol.commit_count}

    async def reload(self, volume_id: str) -> None:
        self._get(volume_id)  # shared filesystem: nothing to fetch

    # -- reads -----------------------------------------------------------
    async def get_file(
        self, volume_id: str, rel_path: str, offset: int = 0, n_bytes: int = -1
    ) -> bytes:
        vol = self._get(volume_id)
        path = self._safe_path(vol, rel_path)
        if not os.path.isfile(path):
            raise NotFoundError(f"{rel_path} not in volume")
        import asyncio

        def _read() -> bytes:
            with open(path, "rb") as f:
                f.seek(offset)
                return f.read(n_bytes if n_bytes >= 0 else -1)

        return await asyncio.get_running_loop().run_in_executor(None, _read)

    async def list_files(self, volume_id: str, rel_path: str = "/", recursive: bool = True) -> list[dict]:
        vol = self._get(volume_id)
        base = self._safe_path(vol, rel_path)
        out: list[dict] = []
        if os.path.isfile(base):
            st = os.stat(base)
            return [{"path": rel_path.lstrip("/"), "size": st.st_size, "mtime": st.st_mtime, "type": "file"}]
        if not os.path.isdir(base):
            raise NotFoundError(f"{rel_path} not in volume")
        if recursive:
            for dirpath, dirnames, filenames in os.walk(base):
                for d in dirnames:
                    full = os.path.join(dirpath, d)
                    out.append({"path": os.path.relpath(full, vol.root), "size": 0,
                                "mtime": os.stat(full).st_mtime, "type": "dir"})
                for fn in filenames:
                    full = os.path.join(dirpath, fn)
                    st = os.stat(full)
                    out.append({"path": os.path.relpath(full, vol.root), "size": st.st_size,
                                "mtime": st.st_mtime, "type": "file"})
        else:
            for entry in os.scandir(base):
                st = entry.stat()
                out.append({"path": os.path.relpath(entry.path, vol.root), "size": st.st_size,
                            "mtime": st.st_mtime, "type": "dir" if entry.is_dir() else "file"})
        out.sort(key=lambda e: e["path"])
        return out

    def volume_dir(self, volume_id: str) -> str:
        return self._get(volume_id).root
