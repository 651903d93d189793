"""Namespace managers for named objects: ``Queue.objects``, ``Dict.objects``,
``Secret.objects``, ``Volume.objects``.

Parity: the reference's per-type managers (reference queue.py:36
``_QueueManager`` with create/list/delete; same shape on Dict/Secret/Volume).
Blocking methods with ``.aio`` twins, like the rest of the public surface.
"""

from __future__ import annotations

from typing import Any, Optional

from ._sync import synchronizer
from .exception import AlreadyExistsError


async def _client_of(client: Any) -> Any:
    if client is not None:
        return getattr(client, "_impl", client)
    from .client import _Client

    return await _Client.from_env()


class ObjectManager:
    """Workspace-level operations on one kind of named object."""

    def __init__(self, kind: str):
        self._kind = kind

    async def _create(
        self,
        name: str,
        *,
        allow_existing: bool = False,
        environment_name: str = "",
        client: Any = None,
    ) -> None:
        c = await _client_of(client)
        env = environment_name or "main"
        rows = await c.svc.named_objects_list(kind=self._kind, environment=env)
        if any(r["name"] == name for r in rows):
            if allow_existing:
                return
            raise AlreadyExistsError(f"{self._kind} '{name}' already exists")
        if self._kind == "queue":
            await c.svc.queue_get_or_create(
                name=name, environment=env, create_if_missing=True, ephemeral=False
            )
        elif self._kind == "dict":
            await c.svc.dict_get_or_create(
                name=name, environment=env, create_if_missing=True, ephemeral=False
            )
        elif self._kind == "secret":
            await c.svc.secret_get_or_create(name=name, environment=env, env={})
        elif self._kind == "volume":
            await c.svc.volume_get_or_create(
                name=name, environment=env, create_if_missing=True
            )

    async def _list(
        self, *, environment_name: str = "", client: Any = None
    ) -> list[dict]:
        c = await _client_of(client)
        return await c.svc.named_objects_list(
            kind=self._kind, environment=environment_name or "main"
        )

    async def _delete(
        self, name: str, *, environment_name: str = "", client: Any = None
    ) -> None:
        c = await _client_of(client)
        await c.svc.named_object_delete(
            kind=self._kind, name=name, environment=environment_name or "main"
        )

    # blocking surface + .aio twins -------------------------------------
    def create(self, name: str, **kwargs: Any) -> None:
        return synchronizer.run(self._create(name, **kwargs))

    def list(self, **kwargs: Any) -> list[dict]:
        return synchronizer.run(self._list(**kwargs))

    def delete(self, name: str, **kwargs: Any) -> None:
        return synchronizer.run(self._delete(name, **kwargs))

    def __repr__(self) -> str:
        return f"<ObjectManager kind={self._kind}>"


def install(impl_cls: type, wrapper_cls: Optional[type], kind: str) -> None:
    """Attach a shared manager instance as ``.objects`` on both the impl and
    the public wrapper class (plain attributes survive neither MRO copy)."""
    mgr = ObjectManager(kind)
    impl_cls.objects = mgr  # type: ignore[attr-defined]
    if wrapper_cls is not None:
        wrapper_cls.objects = mgr  # type: ignore[attr-defined]
