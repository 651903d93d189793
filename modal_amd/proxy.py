"""Proxy: egress-proxy handle (parity: /root/reference/py/modal/proxy.py:57).

The reference routes a function's outbound traffic through a managed proxy
host (static IP). Locally this is a REAL forward proxy on 127.0.0.1
(scheduler/localproxy.py): functions declared with ``proxy=`` run with
HTTP(S)_PROXY pointed at it, so their HTTP traffic takes the same shape as
in the reference — one shared egress point.
"""

from __future__ import annotations

from typing import Any

from ._object import _Object
from ._sync import synchronize_api


class _Proxy(_Object, type_kind="tunnel"):
    @classmethod
    def from_name(cls, name: str, *, environment_name: str = "") -> "_Proxy":
        async def _load(obj: "_Proxy", resolver: Any, existing: Any) -> None:
            resp = await resolver.client.svc.proxy_get_or_create(
                name=name, environment=environment_name
            )
            obj._hydrate(
                resp["proxy_id"], resolver.client,
                {"name": name, "url": resp["url"], "port": resp["port"]},
            )

        return cls._from_loader(_load, rep=f"Proxy.from_name({name!r})")

    @property
    def url(self) -> str:
        return (getattr(self, "_metadata", None) or {}).get("url", "")


Proxy = synchronize_api(_Proxy, "Proxy")
