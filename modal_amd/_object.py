"""Object layer: uniform lazy handles with explicit hydration.

Parity with the reference (/root/reference/py/modal/_object.py):

* Every resource is an ``_Object`` subclass keyed by an ID prefix
  (``__init_subclass__`` registry; reference :101).
* Lazy constructors (``from_name``, ``ephemeral``, factory DSLs) register a
  ``_load`` callback (reference ``_from_loader`` :199); ``hydrate()``
  (reference :322) drives ``Resolver.load`` (reference _resolver.py:39) which
  dedups by local identity + deduplication key, runs ``_load``, and stamps
  ``(object_id, client, metadata)`` via ``_hydrate`` (reference :161).
* ``live_method`` auto-hydrates on first use (reference :42).

Here the "server" the resolver talks to is the in-process scheduler reached
through a ``Client`` (see modal_amd/client.py); inside worker processes the
same handles bind to a socket-backed scheduler proxy.
"""

from __future__ import annotations

import functools
from typing import Any, Awaitable, Callable, ClassVar, Optional, TypeVar

from ._serialization import register_handle_factory
from .exception import ExecutionError, InvalidError
from .utils.ids import id_type

O = TypeVar("O", bound="_Object")

_TYPE_REGISTRY: dict[str, type] = {}


class _Object:
    _is_modal_object: ClassVar[bool] = True
    _type_kind: ClassVar[str] = ""  # e.g. "queue"; set via __init_subclass__

    _object_id: Optional[str]
    _client: Any
    _is_hydrated: bool
    _rep: str
    _load_fn: Optional[Callable[["_Object", "Resolver", Any], Awaitable[None]]]
    _deduplication_key: Optional[Callable[[], Awaitable[Any]]]
    _deps: Optional[Callable[[], list["_Object"]]]
    _is_another_app: bool

    def __init_subclass__(cls, type_kind: str = "", **kwargs: Any) -> None:
        super().__init_subclass__(**kwargs)
        if type_kind:
            cls._type_kind = type_kind
            _TYPE_REGISTRY[type_kind] = cls

    def __init__(self, *args: Any, **kwargs: Any) -> None:
        raise InvalidError(
            f"{type(self).__name__} objects must be created through factory methods "
            f"(from_name, ephemeral, lookup, ...)"
        )

    # -- construction ----------------------------------------------------
    @classmethod
    def _new(
        cls: type[O],
        rep: str = "",
        load: Optional[Callable] = None,
        is_another_app: bool = False,
        hydrate_lazily: bool = True,
        deps: Optional[Callable[[], list["_Object"]]] = None,
        deduplication_key: Optional[Callable] = None,
    ) -> O:
        obj = object.__new__(cls)
        obj._object_id = None
        obj._client = None
        obj._is_hydrated = False
        obj._rep = rep or cls.__name__
        obj._load_fn = load
        obj._deduplication_key = deduplication_key
        obj._deps = deps
        obj._is_another_app = is_another_app
        obj._init_attrs()
        return obj

    def _init_attrs(self) -> None:
        """Subclass hook for per-instance attribute defaults."""

    @classmethod
    def _from_loader(
        cls: type[O],
        load: Callable,
        rep: str,
        deps: Optional[Callable[[], list["_Object"]]] = None,
        deduplication_key: Optional[Callable] = None,
    ) -> O:
        return cls._new(rep=rep, load=load, deps=deps, deduplication_key=deduplication_key)

    @classmethod
    def _new_hydrated(
        cls: type[O], object_id: str, client: Any, metadata: Optional[dict] = None
    ) -> O:
        obj = cls._new(rep=f"{cls.__name__}({object_id})")
        obj._hydrate(object_id, client, metadata)
        return obj

    # -- hydration -------------------------------------------------------
    def _hydrate(self, object_id: str, client: Any, metadata: Optional[dict]) -> None:
        self._object_id = object_id
        self._client = client
        self._is_hydrated = True
        self._metadata = metadata or {}
        if metadata:
            self._hydrate_metadata(metadata)

    def _hydrate_metadata(self, metadata: dict) -> None:
        """Subclass hook: absorb server-side metadata on hydration."""

    def _get_metadata(self) -> dict:
        """Subclass hook: metadata carried when the handle is serialized."""
        return {}

    async def hydrate(self: O, client: Any = None) -> O:
        """Resolve this handle against the scheduler (reference _object.py:322)."""
        if self._is_hydrated:
            return self
        from .client import _Client

        resolver = Resolver(client or await _Client.from_env())
        await resolver.load(self)
        return self

    # -- accessors --------------------------------------------------------
    @property
    def object_id(self) -> str:
        if self._object_id is None:
            raise ExecutionError(f"{self._rep} has not been hydrated (no object id yet)")
        return self._object_id

    @property
    def is_hydrated(self) -> bool:
        return self._is_hydrated

    @property
    def client(self) -> Any:
        return self._client

    def __repr__(self) -> str:
        status = self._object_id if self._is_hydrated else "unhydrated"
        return f"<{type(self).__name__} {self._rep} [{status}]>"


class Resolver:
    """Loads objects, deduplicating concurrent loads.

    Parity: reference _resolver.py:39-100 — dedup by object identity and by
    (type, deduplication key); loads dependencies first.
    """

    def __init__(self, client: Any, environment_name: str = "", app_id: Optional[str] = None):
        self.client = client
        self.environment_name = environment_name
        self.app_id = app_id
        self._by_identity: dict[int, Any] = {}  # id(obj) -> asyncio.Future
        self._by_dedup_key: dict[Any, _Object] = {}

    async def load(self, obj: _Object) -> _Object:
        import asyncio

        if obj._is_hydrated and not obj._is_another_app:
            return obj
        existing = self._by_identity.get(id(obj))
        if existing is not None:
            await existing
            return obj

        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._by_identity[id(obj)] = fut
        try:
            dedup_key = None
            if obj._deduplication_key is not None:
                dedup_key = (type(obj), await obj._deduplication_key())
                prior = self._by_dedup_key.get(dedup_key)
                if prior is not None and prior._is_hydrated:
                    obj._hydrate(prior._object_id, prior._client, prior._get_metadata() or None)
                    fut.set_result(None)
                    return obj
            if obj._deps is not None:
                deps = obj._deps()
                if deps:
                    await asyncio.gather(*(self.load(dep) for dep in deps))
            if obj._load_fn is None:
                if not obj._is_hydrated:
                    raise ExecutionError(f"{obj._rep} has no loader and is not hydrated")
            else:
                await obj._load_fn(obj, self, None)
            if not obj._is_hydrated:
                raise ExecutionError(f"Loader for {obj._rep} did not hydrate it")
            if dedup_key is not None:
                self._by_dedup_key[dedup_key] = obj
            fut.set_result(None)
            return obj
        except BaseException as exc:
            fut.set_exception(exc)
            # consume so "exception never retrieved" warnings don't fire when no-one awaits
            fut.exception()
            del self._by_identity[id(obj)]
            raise


def live_method(fn: Callable) -> Callable:
    """Auto-hydrate on first use (reference _object.py:42)."""

    @functools.wraps(fn)
    async def wrapped(self: _Object, *args: Any, **kwargs: Any) -> Any:
        if not self._is_hydrated:
            await self.hydrate()
        return await fn(self, *args, **kwargs)

    return wrapped


def live_method_gen(fn: Callable) -> Callable:
    @functools.wraps(fn)
    async def wrapped(self: _Object, *args: Any, **kwargs: Any) -> Any:
        if not self._is_hydrated:
            await self.hydrate()
        async for item in fn(self, *args, **kwargs):
            yield item

    return wrapped


def _rebuild_handle(object_id: str, metadata: dict, client: Any) -> Any:
    """Handle factory used by the unpickler to resurrect serialized handles.

    Returns the *public wrapper* (what user code holds), not the impl.
    """
    from ._sync import wrap

    if client is None:
        from .client import _Client

        client = _Client._singleton  # same-process deserialization
    kind = id_type(object_id)
    cls = _TYPE_REGISTRY.get(kind)
    if cls is None:
        raise InvalidError(f"No object type registered for id {object_id!r}")
    return wrap(cls._new_hydrated(object_id, client, metadata))


register_handle_factory(_rebuild_handle)
