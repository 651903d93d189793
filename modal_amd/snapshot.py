"""SandboxSnapshot handle (parity: /root/reference/py/modal/snapshot.py:17)."""

from __future__ import annotations

from typing import Any

from ._object import _Object
from ._sync import synchronize_api


class _SandboxSnapshot(_Object, type_kind="sandbox_snapshot"):
    @classmethod
    def from_id(cls, snapshot_id: str, client: Any = None) -> "_SandboxSnapshot":
        async def _load(obj: "_SandboxSnapshot", resolver: Any, existing: Any) -> None:
            obj._hydrate(snapshot_id, resolver.client, None)

        obj = cls._from_loader(_load, rep=f"SandboxSnapshot({snapshot_id})")
        if client is not None:
            obj._hydrate(snapshot_id, client, None)
        return obj


SandboxSnapshot = synchronize_api(_SandboxSnapshot, "SandboxSnapshot")
