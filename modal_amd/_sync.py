"""Dual sync/async API machinery.

The reference SDK builds its entire public surface with the `synchronicity`
library: every public class is generated from an async implementation class,
giving each method a blocking form plus an ``.aio`` async twin driven from one
background event loop (/root/reference/py/modal/_utils/async_utils.py:327-330).

This module is a from-scratch, dependency-free equivalent:

* ``Synchronizer`` owns a daemon thread running a private asyncio loop. It is
  fork-aware (post-fork the loop is re-created lazily; parity with the
  reference's post-fork client reset, /root/reference/py/modal/client.py:356).
* ``synchronize_api(_Impl)`` produces a public wrapper class whose coroutine
  methods become blocking methods with ``.aio`` twins, with wrapper<->impl
  argument/result translation so users never see impl objects.
* Async generators bridge to both sync iterators and user-loop async
  iterators.
"""

from __future__ import annotations

import asyncio
import atexit
import concurrent.futures
import functools
import inspect
import os
import threading
from typing import Any, AsyncGenerator, Callable, Optional, TypeVar

_T = TypeVar("_T")

_STOP_SENTINEL = object()


class Synchronizer:
    """Owns the background event loop all impl coroutines run on."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._pid = os.getpid()
        self._stopped = False
        #: worker processes set this: blocking calls then run on a
        #: per-thread event loop IN the calling thread instead of hopping
        #: to the shared loop — a cross-thread wakeup costs ~1-2 ms on
        #: contended hosts, dominating per-item handle RPCs (Queue.get).
        #: Requires loop-affine transports (client.py UserCodeProxy keys
        #: its connections by running loop).
        self.thread_local_blocking = False
        self._tls = threading.local()

    def _thread_loop(self) -> asyncio.AbstractEventLoop:
        loop = getattr(self._tls, "loop", None)
        if loop is None or loop.is_closed():
            loop = asyncio.new_event_loop()
            self._tls.loop = loop
        return loop

    def _use_thread_local(self) -> bool:
        if not self.thread_local_blocking or self.in_loop_thread():
            return False
        try:
            asyncio.get_running_loop()
            return False  # caller is inside some loop: keep bridge semantics
        except RuntimeError:
            return True

    # -- loop lifecycle -------------------------------------------------
    def _ensure_loop(self) -> asyncio.AbstractEventLoop:
        with self._lock:
            if self._loop is not None and self._pid == os.getpid() and self._loop.is_running():
                return self._loop
            # fresh start (first use, post-fork, or after close)
            loop = asyncio.new_event_loop()
            ready = threading.Event()

            def _run() -> None:
                asyncio.set_event_loop(loop)
                loop.call_soon(ready.set)
                loop.run_forever()
                # drain pending tasks on shutdown
                try:
                    pending = asyncio.all_tasks(loop)
                    for task in pending:
                        task.cancel()
                    if pending:
                        loop.run_until_complete(asyncio.gather(*pending, return_exceptions=True))
                finally:
                    loop.close()

            thread = threading.Thread(target=_run, name="modal-amd-loop", daemon=True)
            thread.start()
            ready.wait()
            self._loop = loop
            self._thread = thread
            self._pid = os.getpid()
            self._stopped = False
            return loop

    @property
    def loop(self) -> asyncio.AbstractEventLoop:
        return self._ensure_loop()

    def in_loop_thread(self) -> bool:
        return self._thread is not None and threading.current_thread() is self._thread

    def stop(self) -> None:
        with self._lock:
            loop, thread = self._loop, self._thread
            self._loop = None
            self._thread = None
            self._stopped = True
        if loop is not None and loop.is_running():
            loop.call_soon_threadsafe(loop.stop)
            if thread is not None:
                thread.join(timeout=5)

    # -- execution ------------------------------------------------------
    def run(self, coro: Any) -> Any:
        """Run a coroutine on the background loop, blocking the caller."""
        if self.in_loop_thread():
            raise RuntimeError(
                "Blocking API called from within the framework event loop; use the .aio variant"
            )
        if self._use_thread_local():
            return self._thread_loop().run_until_complete(coro)
        fut = asyncio.run_coroutine_threadsafe(coro, self._ensure_loop())
        try:
            return fut.result()
        except KeyboardInterrupt:
            fut.cancel()
            raise

    def run_future(self, coro: Any) -> concurrent.futures.Future:
        return asyncio.run_coroutine_threadsafe(coro, self._ensure_loop())

    async def run_async(self, coro: Any) -> Any:
        """Await a coroutine on the background loop from the *user's* loop."""
        if self.in_loop_thread():
            return await coro
        fut = asyncio.run_coroutine_threadsafe(coro, self._ensure_loop())
        return await asyncio.wrap_future(fut)

    def run_generator_sync(self, agen: AsyncGenerator) -> Any:
        """Bridge an async generator to a plain (blocking) generator.

        Items stream through a thread-safe queue filled by a pump coroutine on
        the background loop — one cross-thread hop per *burst*, not per item
        (a per-item run_coroutine_threadsafe round-trip costs ~150 us and
        would dominate map() throughput).
        """
        if self._use_thread_local():
            # worker processes: drive the agen in THIS thread, no hops
            loop = self._thread_loop()
            while True:
                try:
                    yield loop.run_until_complete(agen.__anext__())
                except StopAsyncIteration:
                    return
        import queue as _queue

        loop = self._ensure_loop()
        q: _queue.Queue = _queue.Queue(maxsize=4096)
        ITEM, DONE, ERROR = 0, 1, 2

        async def _put(msg: tuple) -> None:
            while True:
                try:
                    q.put_nowait(msg)
                    return
                except _queue.Full:
                    await asyncio.sleep(0.002)

        async def pump() -> None:
            try:
                async for item in agen:
                    await _put((ITEM, item))
                await _put((DONE, None))
            except asyncio.CancelledError:
                raise
            except BaseException as exc:
                await _put((ERROR, exc))

        fut = asyncio.run_coroutine_threadsafe(pump(), loop)
        try:
            while True:
                kind, value = q.get()
                if kind == ITEM:
                    yield value
                elif kind == DONE:
                    return
                else:
                    raise value
        finally:
            if not fut.done():
                fut.cancel()
            closer = asyncio.run_coroutine_threadsafe(agen.aclose(), loop)
            try:
                closer.result(timeout=5)
            except Exception:
                pass

    async def run_generator_async(self, agen: AsyncGenerator) -> Any:
        """Bridge an async generator running on our loop to the user's loop.

        Items transfer in bursts through a shared buffer (GIL-atomic list
        ops) with a wakeup future on the user's loop — not one cross-loop
        round trip per item.
        """
        if self.in_loop_thread():
            async for item in agen:
                yield item
            return
        loop = self._ensure_loop()
        user_loop = asyncio.get_running_loop()
        buffer: list = []
        state: dict[str, Any] = {"done": False, "exc": None, "fut": user_loop.create_future()}

        def wake() -> None:
            fut = state["fut"]
            if not fut.done():
                fut.set_result(None)

        async def pump() -> None:
            try:
                async for item in agen:
                    buffer.append(item)
                    if len(buffer) == 1:
                        user_loop.call_soon_threadsafe(wake)
                    elif len(buffer) > 8192:
                        await asyncio.sleep(0.002)  # soft backpressure
            except asyncio.CancelledError:
                raise
            except BaseException as exc:
                state["exc"] = exc
            finally:
                state["done"] = True
                try:
                    user_loop.call_soon_threadsafe(wake)
                except RuntimeError:
                    pass

        pump_fut = asyncio.run_coroutine_threadsafe(pump(), loop)
        try:
            while True:
                if buffer:
                    batch = buffer.copy()
                    del buffer[: len(batch)]
                    for item in batch:
                        yield item
                    continue
                if state["done"]:
                    if state["exc"] is not None:
                        raise state["exc"]
                    return
                state["fut"] = user_loop.create_future()
                if buffer or state["done"]:
                    continue
                await state["fut"]
        finally:
            if not pump_fut.done():
                pump_fut.cancel()
            closer = asyncio.run_coroutine_threadsafe(agen.aclose(), loop)
            try:
                await asyncio.wrap_future(closer)
            except Exception:
                pass


#: process-global synchronizer, like the reference's module-level `synchronizer`
synchronizer = Synchronizer()
atexit.register(synchronizer.stop)


# ---------------------------------------------------------------------------
# wrapper <-> impl translation
# ---------------------------------------------------------------------------

_WRAPPER_BY_IMPL: dict[type, type] = {}


def unwrap(obj: Any) -> Any:
    """Translate public wrapper objects to impl objects (shallow containers)."""
    impl = getattr(obj, "_impl", None)
    if impl is not None and type(obj) in _WRAPPER_BY_IMPL.values():
        return impl
    if type(obj) is tuple:
        return tuple(unwrap(x) for x in obj)
    if type(obj) is list:
        return [unwrap(x) for x in obj]
    if type(obj) is dict:
        return {k: unwrap(v) for k, v in obj.items()}
    return obj


_WRAP_PASSTHROUGH = frozenset({int, float, str, bytes, bool, type(None)})


def wrap(obj: Any) -> Any:
    """Translate impl objects to their public wrappers (shallow containers)."""
    if type(obj) in _WRAP_PASSTHROUGH:  # the common case on hot paths
        return obj
    wrapper_cls = _WRAPPER_BY_IMPL.get(type(obj))
    if wrapper_cls is not None:
        return wrapper_cls._from_impl(obj)
    if type(obj) is tuple:
        return tuple(wrap(x) for x in obj)
    if type(obj) is list:
        return [wrap(x) for x in obj]
    if type(obj) is dict:
        return {k: wrap(v) for k, v in obj.items()}
    return obj


class _AioCallable:
    """The object bound as e.g. ``fn.remote``: callable (blocking) with ``.aio``."""

    __slots__ = ("_blocking", "aio", "__wrapped__")

    def __init__(self, blocking: Callable, aio: Callable, wrapped: Callable):
        self._blocking = blocking
        self.aio = aio
        self.__wrapped__ = wrapped

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._blocking(*args, **kwargs)

    def __repr__(self) -> str:
        return f"<dual method {self.__wrapped__.__qualname__}>"


def _make_dual(func: Callable, is_agen: bool) -> Any:
    """Build a descriptor exposing blocking + .aio forms of an impl coroutine."""

    class _Descriptor:
        def __set_name__(self, owner: type, name: str) -> None:
            self._name = name

        def __get__(self, wrapper_obj: Any, objtype: Any = None) -> Any:
            if wrapper_obj is None:
                return self
            impl_obj = wrapper_obj._impl

            if is_agen:

                @functools.wraps(func)
                def blocking(*args: Any, **kwargs: Any) -> Any:
                    agen = func(impl_obj, *unwrap(args), **unwrap(kwargs))
                    for item in synchronizer.run_generator_sync(agen):
                        yield wrap(item)

                @functools.wraps(func)
                async def aio(*args: Any, **kwargs: Any) -> Any:
                    agen = func(impl_obj, *unwrap(args), **unwrap(kwargs))
                    async for item in synchronizer.run_generator_async(agen):
                        yield wrap(item)

            else:

                @functools.wraps(func)
                def blocking(*args: Any, **kwargs: Any) -> Any:
                    coro = func(impl_obj, *unwrap(args), **unwrap(kwargs))
                    return wrap(synchronizer.run(coro))

                @functools.wraps(func)
                async def aio(*args: Any, **kwargs: Any) -> Any:
                    coro = func(impl_obj, *unwrap(args), **unwrap(kwargs))
                    return wrap(await synchronizer.run_async(coro))

            return _AioCallable(blocking, aio, func)

    return _Descriptor()


def _make_dual_classmethod(func: Callable, wrapper_holder: dict, is_agen: bool) -> Any:
    """Dual form for @classmethod coroutine factories (e.g. ``Queue.ephemeral``)."""

    class _Descriptor:
        def __get__(self, obj: Any, objtype: Any = None) -> Any:
            impl_cls = wrapper_holder["impl_cls"]

            @functools.wraps(func)
            def blocking(*args: Any, **kwargs: Any) -> Any:
                coro = func(impl_cls, *unwrap(args), **unwrap(kwargs))
                return wrap(synchronizer.run(coro))

            @functools.wraps(func)
            async def aio(*args: Any, **kwargs: Any) -> Any:
                coro = func(impl_cls, *unwrap(args), **unwrap(kwargs))
                return wrap(await synchronizer.run_async(coro))

            return _AioCallable(blocking, aio, func)

    return _Descriptor()


def _make_sync_passthrough(name: str) -> Any:
    def method(self: Any, *args: Any, **kwargs: Any) -> Any:
        return wrap(getattr(self._impl, name)(*unwrap(args), **unwrap(kwargs)))

    method.__name__ = name
    return method


def synchronize_api(impl_cls: type, name: Optional[str] = None) -> type:
    """Generate the public dual-API wrapper class for an async impl class.

    Parity: the reference's ``synchronize_api``
    (/root/reference/py/modal/_utils/async_utils.py:330) strips the ``_``
    prefix and produces a blocking class whose methods carry ``.aio`` twins.
    """
    public_name = name or impl_cls.__name__.lstrip("_")
    holder = {"impl_cls": impl_cls}

    ns: dict[str, Any] = {
        "__doc__": impl_cls.__doc__,
        "__module__": impl_cls.__module__,
        "_impl_cls": impl_cls,
    }

    def __init__(self: Any, *args: Any, **kwargs: Any) -> None:
        self._impl = impl_cls(*unwrap(args), **unwrap(kwargs))

    @classmethod
    def _from_impl(cls: type, impl: Any) -> Any:
        obj = object.__new__(cls)
        obj._impl = impl
        return obj

    ns["__init__"] = __init__
    ns["_from_impl"] = _from_impl

    def __repr__(self: Any) -> str:
        return repr(self._impl)

    ns["__repr__"] = __repr__

    def __eq__(self: Any, other: Any) -> Any:
        if isinstance(other, type(self)):
            return self._impl == other._impl
        return NotImplemented

    def __hash__(self: Any) -> int:
        return hash(self._impl)

    ns["__eq__"] = __eq__
    ns["__hash__"] = __hash__

    seen: set[str] = set()
    for klass in impl_cls.__mro__:
        if klass is object:
            continue
        for attr_name, attr in klass.__dict__.items():
            if attr_name in seen:
                continue
            seen.add(attr_name)
            if attr_name.startswith("__") and attr_name not in (
                "__aenter__",
                "__aexit__",
                "__call__",
                "__getitem__",
                "__setitem__",
                "__delitem__",
                "__len__",
                "__contains__",
                "__iter__",
                "__aiter__",
            ):
                continue
            if attr_name.startswith("_") and not attr_name.startswith("__"):
                continue

            if isinstance(attr, classmethod):
                inner = attr.__func__
                if inspect.iscoroutinefunction(inner):
                    ns[attr_name] = _make_dual_classmethod(inner, holder, False)
                else:

                    def make_cm(inner: Callable = inner) -> Any:
                        @classmethod
                        @functools.wraps(inner)
                        def cm(cls: type, *args: Any, **kwargs: Any) -> Any:
                            return wrap(inner(impl_cls, *unwrap(args), **unwrap(kwargs)))

                        return cm

                    ns[attr_name] = make_cm()
            elif isinstance(attr, staticmethod):
                inner = attr.__func__
                if inspect.iscoroutinefunction(inner):

                    def make_sm(inner: Callable = inner) -> Any:
                        @functools.wraps(inner)
                        def blocking(*args: Any, **kwargs: Any) -> Any:
                            return wrap(synchronizer.run(inner(*unwrap(args), **unwrap(kwargs))))

                        async def aio(*args: Any, **kwargs: Any) -> Any:
                            return wrap(
                                await synchronizer.run_async(inner(*unwrap(args), **unwrap(kwargs)))
                            )

                        blocking.aio = aio  # type: ignore[attr-defined]
                        return staticmethod(blocking)

                    ns[attr_name] = make_sm()
                else:
                    ns[attr_name] = attr
            elif isinstance(attr, property):

                def make_prop(attr: property = attr) -> property:
                    def getter(self: Any) -> Any:
                        return wrap(attr.fget(self._impl))

                    return property(getter)

                ns[attr_name] = make_prop(attr)
            elif inspect.isasyncgenfunction(attr):
                if attr_name == "__aiter__":
                    # expose as __iter__ (sync) and __aiter__ (async)
                    def make_iters(attr: Callable = attr) -> tuple:
                        def __iter__(self: Any) -> Any:
                            agen = attr(self._impl)
                            for item in synchronizer.run_generator_sync(agen):
                                yield wrap(item)

                        def __aiter__(self: Any) -> Any:
                            agen = attr(self._impl)

                            async def gen() -> Any:
                                async for item in synchronizer.run_generator_async(agen):
                                    yield wrap(item)

                            return gen()

                        return __iter__, __aiter__

                    ns["__iter__"], ns["__aiter__"] = make_iters()
                else:
                    ns[attr_name] = _make_dual(attr, True)
            elif inspect.iscoroutinefunction(attr):
                if attr_name == "__aenter__":

                    def __enter__(self: Any) -> Any:
                        return wrap(synchronizer.run(self._impl.__aenter__()))

                    async def __aenter__(self: Any) -> Any:
                        return wrap(await synchronizer.run_async(self._impl.__aenter__()))

                    ns["__enter__"] = __enter__
                    ns["__aenter__"] = __aenter__
                elif attr_name == "__aexit__":

                    def __exit__(self: Any, *exc: Any) -> Any:
                        return synchronizer.run(self._impl.__aexit__(*exc))

                    async def __aexit__(self: Any, *exc: Any) -> Any:
                        return await synchronizer.run_async(self._impl.__aexit__(*exc))

                    ns["__exit__"] = __exit__
                    ns["__aexit__"] = __aexit__
                else:
                    ns[attr_name] = _make_dual(attr, False)
            elif callable(attr) and not attr_name.startswith("__"):
                ns[attr_name] = _make_sync_passthrough(attr_name)
            elif attr_name in ("__getitem__", "__len__", "__contains__", "__call__"):
                if callable(attr):
                    ns[attr_name] = _make_sync_passthrough(attr_name)

    wrapper_cls = type(public_name, (), ns)
    holder["wrapper_cls"] = wrapper_cls
    _WRAPPER_BY_IMPL[impl_cls] = wrapper_cls
    return wrapper_cls


def dual_function(func: Callable) -> Callable:
    """Module-level dual function: blocking call + ``.aio`` twin."""
    if inspect.isasyncgenfunction(func):

        @functools.wraps(func)
        def blocking_gen(*args: Any, **kwargs: Any) -> Any:
            agen = func(*unwrap(args), **unwrap(kwargs))
            for item in synchronizer.run_generator_sync(agen):
                yield wrap(item)

        @functools.wraps(func)
        async def aio_gen(*args: Any, **kwargs: Any) -> Any:
            agen = func(*unwrap(args), **unwrap(kwargs))
            async for item in synchronizer.run_generator_async(agen):
                yield wrap(item)

        blocking_gen.aio = aio_gen  # type: ignore[attr-defined]
        return blocking_gen

    @functools.wraps(func)
    def blocking(*args: Any, **kwargs: Any) -> Any:
        return wrap(synchronizer.run(func(*unwrap(args), **unwrap(kwargs))))

    @functools.wraps(func)
    async def aio(*args: Any, **kwargs: Any) -> Any:
        return wrap(await synchronizer.run_async(func(*unwrap(args), **unwrap(kwargs))))

    blocking.aio = aio  # type: ignore[attr-defined]
    return blocking
