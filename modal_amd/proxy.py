"""Proxy: static outbound-IP proxy handle (parity: /root/reference/py/modal/proxy.py:57).

No egress exists on this node; the handle is kept for API compatibility and
resolves to a no-op configuration object.
"""

from __future__ import annotations

from typing import Any

from ._object import _Object
from ._sync import synchronize_api
from .utils.ids import new_id


class _Proxy(_Object, type_kind="tunnel"):
    @classmethod
    def from_name(cls, name: str, *, environment_name: str = "") -> "_Proxy":
        async def _load(obj: "_Proxy", resolver: Any, existing: Any) -> None:
            obj._hydrate(new_id("tunnel"), resolver.client, {"name": name})

        return cls._from_loader(_load, rep=f"Proxy.from_name({name!r})")


Proxy = synchronize_api(_Proxy, "Proxy")
