"""Testing utilities shipped with the SDK (parity: the reference ships its
interception DSL in the package proper so downstream users can test too —
/root/reference/py/modal/_utils/grpc_testing.py:19-49).

``intercept(client)`` wraps the client's control-plane service so tests can:
  * assert the exact sequence of control-plane calls,
  * inject canned responses or exceptions for specific methods.

Works with both the in-process scheduler and socket-attached proxies.
"""

from __future__ import annotations

import contextlib
from typing import Any, Callable, Iterator, Optional


class CallRecorder:
    def __init__(self, svc: Any):
        self._svc = svc
        self.recorded: list[tuple[str, dict]] = []
        self._overrides: dict[str, Callable] = {}

    # -- assertion helpers ----------------------------------------------
    def method_calls(self, name: str) -> list[dict]:
        return [kwargs for method, kwargs in self.recorded if method == name]

    def assert_called(self, name: str, times: Optional[int] = None) -> None:
        n = len(self.method_calls(name))
        if times is None:
            assert n > 0, f"{name} was never called (saw: {self.sequence()})"
        else:
            assert n == times, f"{name} called {n} times, expected {times}"

    def sequence(self) -> list[str]:
        return [method for method, _ in self.recorded]

    # -- response injection ----------------------------------------------
    def override(self, name: str, handler: Callable) -> None:
        """handler(**kwargs) -> value | raises; may be sync or async."""
        self._overrides[name] = handler

    def raise_on(self, name: str, exc: BaseException) -> None:
        def handler(**_kwargs: Any) -> None:
            raise exc

        self._overrides[name] = handler

    # -- proxy ------------------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        target = getattr(self._svc, name)

        if not callable(target):
            return target

        import asyncio
        import functools

        @functools.wraps(target)
        async def wrapper(*args: Any, **kwargs: Any) -> Any:
            self.recorded.append((name, kwargs or {"_args": args}))
            override = self._overrides.get(name)
            if override is not None:
                result = override(*args, **kwargs)
                if asyncio.iscoroutine(result):
                    result = await result
                return result
            result = target(*args, **kwargs)
            if asyncio.iscoroutine(result):
                return await result
            return result

        return wrapper

    @property
    def is_proxy(self) -> bool:  # transparent for _is_inproc checks
        return getattr(self._svc, "is_proxy", False)

    def __getattribute__(self, name: str) -> Any:
        # route attribute passthrough for scheduler internals used by the
        # in-proc fast paths (pool, calls, out_chunks, apps, blob_store, ...)
        if name in (
            "pool", "calls", "out_chunks", "apps", "blob_store", "functions",
            "services", "run_dir", "_extra", "sandbox_service", "volume_service",
            "image_service", "web_gateway",
        ):
            return getattr(object.__getattribute__(self, "_svc"), name)
        return object.__getattribute__(self, name)


@contextlib.contextmanager
def intercept(client: Any) -> Iterator[CallRecorder]:
    """Record (and optionally override) every control-plane call."""
    from ._sync import unwrap

    client = unwrap(client)
    recorder = CallRecorder(client.svc)
    original = client.svc
    client.svc = recorder
    try:
        yield recorder
    finally:
        client.svc = original
