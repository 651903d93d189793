"""Queue: distributed-FIFO semantics on the in-process scheduler.

Parity: /root/reference/py/modal/queue.py — ``_Queue`` (:218), partition keys
(:318), blocking put/get with deadline polling (:491-520,667-697), iterate
(:740), limits (100k partitions x 5,000 items, :286). Values are cloudpickled
by the client layer, so workers on other GPUs see identical items; GPU-tensor
payloads ride the tensor-aware serializer and move over xGMI-backed shared
memory rather than through pickle copies.
"""

from __future__ import annotations

import time
from typing import Any, AsyncGenerator, Optional

from ._object import _Object, live_method
from ._serialization import deserialize, serialize
from ._sync import synchronize_api, synchronizer, wrap
from .exception import InvalidError, QueueEmptyError, QueueFullError


def _partition_key(partition: Optional[str]) -> Optional[bytes]:
    if partition is None:
        return None
    if not isinstance(partition, str) or not (1 <= len(partition.encode()) <= 64):
        raise InvalidError("Queue partition keys must be strings of 1-64 bytes")
    return partition.encode()


class _Queue(_Object, type_kind="queue"):
    @classmethod
    def from_name(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_Queue":
        async def _load(obj: "_Queue", resolver: Any, existing: Any) -> None:
            qid = await resolver.client.svc.queue_get_or_create(
                name=name,
                environment=environment_name or "main",
                create_if_missing=create_if_missing,
                ephemeral=False,
            )
            obj._hydrate(qid, resolver.client, {"name": name})

        return cls._from_loader(_load, rep=f"Queue.from_name({name!r})")

    @classmethod
    def from_id(cls, object_id: str, client: Any = None) -> "_Queue":
        """Handle from a raw ``qu-`` id (parity: reference from_id)."""
        async def _load(obj: "_Queue", resolver: Any, existing: Any) -> None:
            obj._hydrate(object_id, resolver.client, None)

        obj = cls._from_loader(_load, rep=f"Queue.from_id({object_id!r})")
        if client is not None:
            obj._hydrate(object_id, client, None)
        return obj

    @property
    def name(self) -> 'Any':
        """Deployment name (None for ephemeral queues)."""
        return (getattr(self, "_metadata", None) or {}).get("name")

    @live_method
    async def info(self) -> dict:
        """Name + partition stats (parity: reference info())."""
        return await self._client.svc.object_info(object_id=self.object_id)

    @staticmethod
    def validate_partition_key(partition: 'Any') -> bytes:
        """Parity: reference queue.py validate_partition_key."""
        from .exception import InvalidError

        if partition is None:
            return b""
        key = partition.encode() if isinstance(partition, str) else bytes(partition)
        if len(key) == 0 or len(key) > 64:
            raise InvalidError("Queue partition key must be 1-64 bytes")
        return key

    @classmethod
    async def lookup(
        cls, name: str, *, environment_name: str = "", create_if_missing: bool = False
    ) -> "_Queue":
        obj = cls.from_name(
            name, environment_name=environment_name, create_if_missing=create_if_missing
        )
        return await obj.hydrate()

    @classmethod
    def ephemeral(cls, *, environment_name: str = "") -> "_EphemeralQueue":
        return _EphemeralQueue(environment_name)

    @classmethod
    async def delete(cls, name: str, *, environment_name: str = "") -> None:
        from .client import _Client

        client = await _Client.from_env()
        qid = await client.svc.queue_get_or_create(
            name=name, environment=environment_name or "main", create_if_missing=False, ephemeral=False
        )
        await client.svc.queue_delete(queue_id=qid)

    # -- operations ------------------------------------------------------
    @live_method
    async def put(
        self,
        v: Any,
        block: bool = True,
        timeout: Optional[float] = None,
        *,
        partition: Optional[str] = None,
        partition_ttl: int = 86400,
    ) -> None:
        await self._put_many([v], block, timeout, partition)

    @live_method
    async def put_many(
        self,
        vs: list,
        block: bool = True,
        timeout: Optional[float] = None,
        *,
        partition: Optional[str] = None,
        partition_ttl: int = 86400,
    ) -> None:
        await self._put_many(vs, block, timeout, partition)

    async def _put_many(
        self, vs: list, block: bool, timeout: Optional[float], partition: Optional[str]
    ) -> None:
        values = [serialize(v) for v in vs]
        deadline = None if timeout is None else time.time() + timeout
        try:
            await self._client.svc.queue_put(
                queue_id=self.object_id,
                values=values,
                partition=_partition_key(partition),
                block=block,
                deadline=deadline,
            )
        except QueueFullError:
            raise
        except Exception as exc:
            if type(exc).__name__ == "QueueFullError":
                raise QueueFullError(str(exc)) from None
            raise

    @live_method
    async def get(
        self,
        block: bool = True,
        timeout: Optional[float] = None,
        *,
        partition: Optional[str] = None,
    ) -> Any:
        values = await self._get_many(1, block, timeout, partition)
        if not values:
            if block:
                raise QueueEmptyError("Timed out waiting for item")
            return None
        return values[0]

    @live_method
    async def get_many(
        self,
        n_values: int,
        block: bool = True,
        timeout: Optional[float] = None,
        *,
        partition: Optional[str] = None,
    ) -> list:
        return await self._get_many(n_values, block, timeout, partition)

    async def _get_many(
        self, n_values: int, block: bool, timeout: Optional[float], partition: Optional[str]
    ) -> list:
        deadline = None if timeout is None else time.time() + timeout
        raw = await self._client.svc.queue_get(
            queue_id=self.object_id,
            partition=_partition_key(partition),
            n_values=n_values,
            block=block,
            deadline=deadline,
        )
        return [deserialize(v) for v in raw]

    @live_method
    async def len(self, *, partition: Optional[str] = None, total: bool = False) -> int:
        return await self._client.svc.queue_len(
            queue_id=self.object_id, partition=_partition_key(partition), total=total
        )

    @live_method
    async def clear(self, *, partition: Optional[str] = None, all: bool = False) -> None:  # noqa: A002
        await self._client.svc.queue_clear(
            queue_id=self.object_id, partition=_partition_key(partition), all_partitions=all
        )

    async def iterate(
        self, *, partition: Optional[str] = None, item_poll_timeout: float = 0.0
    ) -> AsyncGenerator[Any, None]:
        """Non-destructive streaming over current items (parity: reference
        queue.py:740): yields existing items, then keeps polling for new ones
        until ``item_poll_timeout`` elapses with nothing new."""
        if not self._is_hydrated:
            await self.hydrate()
        import asyncio

        seen = 0
        deadline = time.time() + item_poll_timeout
        while True:
            items = await self._client.svc.queue_peek(
                queue_id=self.object_id, partition=_partition_key(partition), n=seen + 100
            )
            new = items[seen:]
            if new:
                deadline = time.time() + item_poll_timeout
                for raw in new:
                    yield deserialize(raw)
                seen += len(new)
            elif time.time() > deadline:
                return
            else:
                await asyncio.sleep(0.05)


class _EphemeralQueue:
    """Context manager for a nameless, lifetime-scoped queue
    (parity: Queue.ephemeral, reference queue.py)."""

    def __init__(self, environment_name: str):
        self.environment_name = environment_name
        self._impl: Optional[_Queue] = None

    async def _create(self) -> _Queue:
        from .client import _Client

        client = await _Client.from_env()
        qid = await client.svc.queue_get_or_create(
            name=None, environment=self.environment_name or "main", create_if_missing=True, ephemeral=True
        )
        impl = _Queue._new_hydrated(qid, client, None)
        self._impl = impl
        return impl

    async def _cleanup(self) -> None:
        if self._impl is not None:
            await self._impl._client.svc.queue_delete(queue_id=self._impl.object_id)

    def __enter__(self) -> Any:
        return wrap(synchronizer.run(self._create()))

    def __exit__(self, *exc: Any) -> None:
        synchronizer.run(self._cleanup())

    async def __aenter__(self) -> Any:
        return wrap(await synchronizer.run_async(self._create()))

    async def __aexit__(self, *exc: Any) -> None:
        await synchronizer.run_async(self._cleanup())


Queue = synchronize_api(_Queue, "Queue")

from .object_manager import install as _install_manager  # noqa: E402

_install_manager(_Queue, Queue, "queue")
