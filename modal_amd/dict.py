"""Dict: distributed key-value store on the in-process scheduler.

Parity: /root/reference/py/modal/dict.py — ``_Dict`` (:253), get/put/len/items
(:494-639), per-entry serde hooks (:48-68). Keys and values are cloudpickled;
any pickleable object is a valid key (matching the reference's behavior of
hashing serialized keys).
"""

from __future__ import annotations

from typing import Any, AsyncGenerator, Optional

from ._object import _Object, live_method
from ._serialization import deserialize, serialize
from ._sync import synchronize_api, synchronizer, wrap


class _Dict(_Object, type_kind="dict"):
    @classmethod
    def from_name(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_Dict":
        async def _load(obj: "_Dict", resolver: Any, existing: Any) -> None:
            did = await resolver.client.svc.dict_get_or_create(
                name=name,
                environment=environment_name or "main",
                create_if_missing=create_if_missing,
                ephemeral=False,
            )
            obj._hydrate(did, resolver.client, {"name": name})

        return cls._from_loader(_load, rep=f"Dict.from_name({name!r})")

    @classmethod
    def from_id(cls, object_id: str, client: Any = None) -> "_Dict":
        async def _load(obj: "_Dict", resolver: Any, existing: Any) -> None:
            obj._hydrate(object_id, resolver.client, None)

        obj = cls._from_loader(_load, rep=f"Dict.from_id({object_id!r})")
        if client is not None:
            obj._hydrate(object_id, client, None)
        return obj

    @property
    def name(self) -> Any:
        return (getattr(self, "_metadata", None) or {}).get("name")

    @live_method
    async def info(self) -> dict:
        return await self._client.svc.object_info(object_id=self.object_id)

    @classmethod
    async def lookup(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_Dict":
        obj = cls.from_name(name, environment_name=environment_name, create_if_missing=create_if_missing)
        return await obj.hydrate()

    @classmethod
    def ephemeral(cls, *, environment_name: str = "") -> "_EphemeralDict":
        return _EphemeralDict(environment_name)

    @classmethod
    async def delete(cls, name: str, *, environment_name: str = "") -> None:
        from .client import _Client

        client = await _Client.from_env()
        did = await client.svc.dict_get_or_create(
            name=name, environment=environment_name or "main", create_if_missing=False, ephemeral=False
        )
        await client.svc.dict_delete(dict_id=did)

    # -- operations ------------------------------------------------------
    @live_method
    async def get(self, key: Any, default: Any = None) -> Any:
        raw = await self._client.svc.dict_get(dict_id=self.object_id, key=serialize(key))
        if raw is None:
            return default
        return deserialize(raw)

    @live_method
    async def put(self, key: Any, value: Any, *, skip_if_exists: bool = False) -> bool:
        return await self._client.svc.dict_update(
            dict_id=self.object_id,
            updates={serialize(key): serialize(value)},
            if_not_exists=skip_if_exists,
        )

    @live_method
    async def update(self, other: Optional[dict] = None, /, **kwargs: Any) -> None:
        updates = dict(other or {})
        updates.update(kwargs)
        await self._client.svc.dict_update(
            dict_id=self.object_id,
            updates={serialize(k): serialize(v) for k, v in updates.items()},
        )

    @live_method
    async def pop(self, key: Any) -> Any:
        found, raw = await self._client.svc.dict_pop(dict_id=self.object_id, key=serialize(key))
        if not found:
            raise KeyError(key)
        return deserialize(raw)

    @live_method
    async def contains(self, key: Any) -> bool:
        return await self._client.svc.dict_contains(dict_id=self.object_id, key=serialize(key))

    # dict-style sugar (parity: reference dict.py __getitem__/__setitem__)
    async def __getitem__(self, key: Any) -> Any:
        sentinel = object()
        value = await self.get(key, sentinel)
        if value is sentinel:
            raise KeyError(key)
        return value

    async def __setitem__(self, key: Any, value: Any) -> None:
        await self.put(key, value)

    async def __delitem__(self, key: Any) -> None:
        await self.pop(key)

    async def __contains__(self, key: Any) -> bool:
        return await self.contains(key)

    async def __len__(self) -> int:
        return await self.len()

    @live_method
    async def len(self) -> int:
        return await self._client.svc.dict_len(dict_id=self.object_id)

    @live_method
    async def clear(self) -> None:
        await self._client.svc.dict_clear(dict_id=self.object_id)

    async def keys(self) -> AsyncGenerator[Any, None]:
        if not self._is_hydrated:
            await self.hydrate()
        for k, _ in await self._client.svc.dict_items(dict_id=self.object_id):
            yield deserialize(k)

    async def values(self) -> AsyncGenerator[Any, None]:
        if not self._is_hydrated:
            await self.hydrate()
        for _, v in await self._client.svc.dict_items(dict_id=self.object_id):
            yield deserialize(v)

    async def items(self) -> AsyncGenerator[tuple, None]:
        if not self._is_hydrated:
            await self.hydrate()
        for k, v in await self._client.svc.dict_items(dict_id=self.object_id):
            yield (deserialize(k), deserialize(v))


class _EphemeralDict:
    def __init__(self, environment_name: str):
        self.environment_name = environment_name
        self._impl: Optional[_Dict] = None

    async def _create(self) -> _Dict:
        from .client import _Client

        client = await _Client.from_env()
        did = await client.svc.dict_get_or_create(
            name=None, environment=self.environment_name or "main", create_if_missing=True, ephemeral=True
        )
        impl = _Dict._new_hydrated(did, client, None)
        self._impl = impl
        return impl

    async def _cleanup(self) -> None:
        if self._impl is not None:
            await self._impl._client.svc.dict_delete(dict_id=self._impl.object_id)

    def __enter__(self) -> Any:
        return wrap(synchronizer.run(self._create()))

    def __exit__(self, *exc: Any) -> None:
        synchronizer.run(self._cleanup())

    async def __aenter__(self) -> Any:
        return wrap(await synchronizer.run_async(self._create()))

    async def __aexit__(self, *exc: Any) -> None:
        await synchronizer.run_async(self._cleanup())


Dict = synchronize_api(_Dict, "Dict")

from .object_manager import install as _install_manager  # noqa: E402

_install_manager(_Dict, Dict, "dict")
