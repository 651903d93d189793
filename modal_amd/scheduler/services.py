"""In-memory resource services: Queue, Dict, Secret.

These are the single-node re-implementations of the server-side state behind
the reference's Queue*/Dict*/Secret* RPC groups
(/root/reference/modal_proto/api.proto; client behavior queue.py:218,
dict.py:253, secret.py:234). Values are opaque serialized bytes — the client
layer owns serde — so workers and the client process see identical semantics.
Queues support partitions (parity: queue.py:318) and blocking put/get with
deadlines; limits match the reference's documented 100k partitions x 5,000
items (queue.py:286).
"""

from __future__ import annotations

import asyncio
import time
from collections import deque
from typing import Optional

from ..exception import AlreadyExistsError, NotFoundError, QueueFullError
from ..utils.ids import new_id

QUEUE_MAX_LEN = 5_000  # per partition; parity: queue.py docstring :286
QUEUE_MAX_PARTITIONS = 100_000


class _Partition:
    def __init__(self) -> None:
        self.items: deque[bytes] = deque()
        self.not_empty = asyncio.Event()
        self.not_full = asyncio.Event()
        self.not_full.set()

    def _update(self) -> None:
        if self.items:
            self.not_empty.set()
        else:
            self.not_empty.clear()
        if len(self.items) < QUEUE_MAX_LEN:
            self.not_full.set()
        else:
            self.not_full.clear()


class QueueState:
    def __init__(self, queue_id: str, name: Optional[str]):
        self.queue_id = queue_id
        self.name = name
        self.partitions: dict[bytes, _Partition] = {}

    def partition(self, key: Optional[bytes]) -> _Partition:
        k = key or b""
        part = self.partitions.get(k)
        if part is None:
            if len(self.partitions) >= QUEUE_MAX_PARTITIONS:
                raise RuntimeError("too many queue partitions")
            part = _Partition()
            self.partitions[k] = part
        return part


class DictState:
    def __init__(self, dict_id: str, name: Optional[str]):
        self.dict_id = dict_id
        self.name = name
        self.data: dict[bytes, bytes] = {}


class SecretState:
    def __init__(self, secret_id: str, name: Optional[str], env: dict[str, str]):
        self.secret_id = secret_id
        self.name = name
        self.env = env


class NamedStore:
    """Deployed-object namespace: (environment, name) -> object id."""

    def __init__(self) -> None:
        self.by_name: dict[tuple[str, str], str] = {}

    def lookup(self, environment: str, name: str) -> Optional[str]:
        return self.by_name.get((environment, name))

    def set(self, environment: str, name: str, object_id: str) -> None:
        self.by_name[(environment, name)] = object_id


class Services:
    def __init__(self) -> None:
        self.queues: dict[str, QueueState] = {}
        self.dicts: dict[str, DictState] = {}
        self.secrets: dict[str, SecretState] = {}
        self.queue_names = NamedStore()
        self.dict_names = NamedStore()
        self.secret_names = NamedStore()

    # ---- queues --------------------------------------------------------
    def object_info(self, object_id: str) -> dict:
        """Name/metadata for a named resource handle (parity: the
        reference's *Info/metadata RPCs)."""
        if object_id.startswith("qu-"):
            q = self._queue(object_id)
            return {"name": q.name, "num_partitions": len(q.partitions)}
        if object_id.startswith("di-"):
            d = self._dict(object_id)
            return {"name": d.name, "len": len(d.data)}
        if object_id.startswith("st-"):
            st = self.secrets.get(object_id)
            if st is None:
                from ..exception import NotFoundError

                raise NotFoundError(f"Secret {object_id} not found")
            return {"name": st.name, "keys": sorted(st.env)}
        from ..exception import NotFoundError

        raise NotFoundError(f"no info for {object_id}")

    def queue_get_or_create(
        self, name: Optional[str], environment: str, create_if_missing: bool, ephemeral: bool
    ) -> str:
        if name and not ephemeral:
            existing = self.queue_names.lookup(environment, name)
            if existing:
                return existing
            if not create_if_missing:
                raise NotFoundError(f"Queue '{name}' not found")
        qid = new_id("queue")
        self.queues[qid] = QueueState(qid, name)
        if name and not ephemeral:
            self.queue_names.set(environment, name, qid)
        return qid

    def _queue(self, queue_id: str) -> QueueState:
        q = self.queues.get(queue_id)
        if q is None:
            raise NotFoundError(f"Queue {queue_id} not found")
        return q

    async def queue_put(
        self,
        queue_id: str,
        values: list[bytes],
        partition: Optional[bytes],
        block: bool,
        deadline: Optional[float],
    ) -> None:
        part = self._queue(queue_id).partition(partition)
        for value in values:
            while len(part.items) >= QUEUE_MAX_LEN:
                if not block:
                    raise _queue_full()
                timeout = None if deadline is None else deadline - time.time()
                if timeout is not None and timeout <= 0:
                    raise _queue_full()
                part.not_full.clear()
                try:
                    await asyncio.wait_for(part.not_full.wait(), timeout)
                except asyncio.TimeoutError:
                    raise _queue_full() from None
            part.items.append(value)
            part._update()

    async def queue_get(
        self,
        queue_id: str,
        partition: Optional[bytes],
        n_values: int,
        block: bool,
        deadline: Optional[float],
    ) -> list[bytes]:
        part = self._queue(queue_id).partition(partition)
        while not part.items:
            if not block:
                return []
            timeout = None if deadline is None else deadline - time.time()
            if timeout is not None and timeout <= 0:
                return []
            part.not_empty.clear()
            try:
                await asyncio.wait_for(part.not_empty.wait(), timeout)
            except asyncio.TimeoutError:
                return []
        out = []
        while part.items and len(out) < n_values:
            out.append(part.items.popleft())
        part._update()
        return out

    def queue_len(self, queue_id: str, partition: Optional[bytes], total: bool) -> int:
        q = self._queue(queue_id)
        if total:
            return sum(len(p.items) for p in q.partitions.values())
        part = q.partitions.get(partition or b"")
        return len(part.items) if part else 0

    def queue_clear(self, queue_id: str, partition: Optional[bytes], all_partitions: bool) -> None:
        q = self._queue(queue_id)
        if all_partitions:
            q.partitions.clear()
        else:
            part = q.partitions.get(partition or b"")
            if part:
                part.items.clear()
                part._update()

    def queue_peek(self, queue_id: str, partition: Optional[bytes], n: int) -> list[bytes]:
        q = self._queue(queue_id)
        part = q.partitions.get(partition or b"")
        if not part:
            return []
        return list(part.items)[:n]

    def queue_delete(self, queue_id: str) -> None:
        self.queues.pop(queue_id, None)
        self.queue_names.by_name = {
            k: v for k, v in self.queue_names.by_name.items() if v != queue_id
        }

    # ---- dicts ---------------------------------------------------------
    def dict_get_or_create(
        self,
        name: Optional[str],
        environment: str,
        create_if_missing: bool,
        ephemeral: bool,
        initial: Optional[dict[bytes, bytes]] = None,
    ) -> str:
        if name and not ephemeral:
            existing = self.dict_names.lookup(environment, name)
            if existing:
                return existing
            if not create_if_missing:
                raise NotFoundError(f"Dict '{name}' not found")
        did = new_id("dict")
        state = DictState(did, name)
        if initial:
            state.data.update(initial)
        self.dicts[did] = state
        if name and not ephemeral:
            self.dict_names.set(environment, name, did)
        return did

    def _dict(self, dict_id: str) -> DictState:
        d = self.dicts.get(dict_id)
        if d is None:
            raise NotFoundError(f"Dict {dict_id} not found")
        return d

    def dict_update(self, dict_id: str, updates: dict[bytes, bytes], if_not_exists: bool = False) -> bool:
        d = self._dict(dict_id)
        if if_not_exists:
            created = False
            for k, v in updates.items():
                if k not in d.data:
                    d.data[k] = v
                    created = True
            return created
        d.data.update(updates)
        return True

    def dict_get(self, dict_id: str, key: bytes) -> Optional[bytes]:
        return self._dict(dict_id).data.get(key)

    def dict_pop(self, dict_id: str, key: bytes) -> tuple[bool, Optional[bytes]]:
        d = self._dict(dict_id)
        if key in d.data:
            return True, d.data.pop(key)
        return False, None

    def dict_contains(self, dict_id: str, key: bytes) -> bool:
        return key in self._dict(dict_id).data

    def dict_len(self, dict_id: str) -> int:
        return len(self._dict(dict_id).data)

    def dict_items(self, dict_id: str) -> list[tuple[bytes, bytes]]:
        return list(self._dict(dict_id).data.items())

    def dict_clear(self, dict_id: str) -> None:
        self._dict(dict_id).data.clear()

    def dict_delete(self, dict_id: str) -> None:
        self.dicts.pop(dict_id, None)
        self.dict_names.by_name = {
            k: v for k, v in self.dict_names.by_name.items() if v != dict_id
        }

    # ---- secrets -------------------------------------------------------
    def secret_get_or_create(
        self,
        name: Optional[str],
        environment: str,
        env: Optional[dict[str, str]],
        overwrite: bool = True,
        required_keys: Optional[list[str]] = None,
    ) -> str:
        if name and env is None:
            existing = self.secret_names.lookup(environment, name)
            if existing is None:
                raise NotFoundError(f"Secret '{name}' not found")
            state = self.secrets[existing]
            for key in required_keys or []:
                if key not in state.env:
                    raise NotFoundError(f"Secret '{name}' is missing key '{key}'")
            return existing
        if name and not overwrite and self.secret_names.lookup(environment, name):
            raise AlreadyExistsError(f"Secret '{name}' already exists")
        sid = new_id("secret")
        self.secrets[sid] = SecretState(sid, name, dict(env or {}))
        if name:
            self.secret_names.set(environment, name, sid)
        return sid

    def secret_env(self, secret_id: str) -> dict[str, str]:
        s = self.secrets.get(secret_id)
        if s is None:
            raise NotFoundError(f"Secret {secret_id} not found")
        return dict(s.env)


def _queue_full() -> Exception:
    return QueueFullError("queue partition is full")
