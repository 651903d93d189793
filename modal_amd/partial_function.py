"""Method/lifecycle decorators for class services and web endpoints.

Parity: /root/reference/py/modal/_partial_function.py —
``modal.method`` (:283), ``modal.enter``/``modal.exit`` (:589,617),
``modal.batched`` (:640), ``modal.concurrent`` (:701), web decorators
``fastapi_endpoint``/``asgi_app``/``wsgi_app``/``web_server``
(:337,414,469,526).
"""

from __future__ import annotations

from typing import Any, Callable, Optional

from .exception import InvalidError


class PartialFunction:
    """A decorated method/endpoint captured before the class/app wires it up."""

    def __init__(self, raw_f: Callable, flags: dict):
        self.raw_f = raw_f
        self.flags = flags
        self.__name__ = getattr(raw_f, "__name__", "f")
        self.__doc__ = getattr(raw_f, "__doc__", None)

    def __get__(self, obj: Any, objtype: Any = None) -> Any:
        # accessed on an instance outside the service machinery: behave like
        # the raw function (local calls)
        if obj is None:
            return self
        return self.raw_f.__get__(obj, objtype)

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self.raw_f(*args, **kwargs)


def method(_warn_parentheses_missing: Any = None, *, is_generator: Optional[bool] = None) -> Callable:
    """Expose a class method as a remotely callable Function (reference :283)."""
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.method() with parentheses")

    def wrapper(raw_f: Callable) -> PartialFunction:
        if isinstance(raw_f, PartialFunction):
            raw = raw_f.raw_f
            flags = dict(raw_f.flags)
        else:
            raw, flags = raw_f, {}
        flags.update({"method": True, "is_generator": is_generator})
        return PartialFunction(raw, flags)

    return wrapper


def _lifecycle(kind: str) -> Callable:
    def deco(_warn_parentheses_missing: Any = None, *, snap: bool = False) -> Callable:
        if _warn_parentheses_missing is not None and callable(_warn_parentheses_missing):
            # bare usage @modal.enter without parens
            f = _warn_parentheses_missing
            f._modal_amd_lifecycle = kind
            return f

        def wrapper(f: Callable) -> Callable:
            # snap=True: runs BEFORE the memory snapshot is taken (parity:
            # reference _partial_function.py:589 enter(snap=...)); plain
            # enter hooks run after restore
            f._modal_amd_lifecycle = f"{kind}_snap" if snap and kind == "enter" else kind
            return f

        return wrapper

    return deco


enter = _lifecycle("enter")
exit = _lifecycle("exit")  # noqa: A001 - parity with reference name


def batched(_warn_parentheses_missing: Any = None, *, max_batch_size: int, wait_ms: int) -> Callable:
    """Dynamic batching (reference :640): the runtime accumulates up to
    ``max_batch_size`` inputs or ``wait_ms`` linger, transposes args, calls
    once, and splits the returned list per input."""
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.batched() with parentheses")
    if max_batch_size < 1:
        raise InvalidError("max_batch_size must be >= 1")
    if wait_ms < 0:
        raise InvalidError("wait_ms must be >= 0")

    def wrapper(raw_f: Callable) -> PartialFunction:
        if isinstance(raw_f, PartialFunction):
            raw, flags = raw_f.raw_f, dict(raw_f.flags)
        else:
            raw, flags = raw_f, {}
        flags.update({"batch_max_size": max_batch_size, "batch_linger_ms": wait_ms})
        return PartialFunction(raw, flags)

    return wrapper


def concurrent(
    _warn_parentheses_missing: Any = None, *, max_inputs: int, target_inputs: Optional[int] = None
) -> Callable:
    """Input concurrency within one worker (reference :701; slots semantics
    container_io_manager.py:485)."""
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.concurrent() with parentheses")
    if target_inputs and target_inputs > max_inputs:
        raise InvalidError("target_inputs must be <= max_inputs")

    def wrapper(raw_f: Callable) -> PartialFunction:
        if isinstance(raw_f, PartialFunction):
            raw, flags = raw_f.raw_f, dict(raw_f.flags)
        else:
            raw, flags = raw_f, {}
        flags.update(
            {"max_concurrent_inputs": max_inputs, "target_concurrent_inputs": target_inputs or 0}
        )
        return PartialFunction(raw, flags)

    return wrapper


def fastapi_endpoint(
    _warn_parentheses_missing: Any = None,
    *,
    method: str = "GET",
    label: Optional[str] = None,
    docs: bool = False,
    requires_proxy_auth: bool = False,
) -> Callable:
    """Wrap a function as a FastAPI endpoint served by the web layer
    (reference :337)."""
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.fastapi_endpoint() with parentheses")

    def wrapper(raw_f: Callable) -> PartialFunction:
        return PartialFunction(
            raw_f,
            {"web": {"type": "fastapi", "method": method, "label": label, "docs": docs}},
        )

    return wrapper


# deprecated alias kept for API parity
web_endpoint = fastapi_endpoint


def asgi_app(
    _warn_parentheses_missing: Any = None, *, label: Optional[str] = None, requires_proxy_auth: bool = False
) -> Callable:
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.asgi_app() with parentheses")

    def wrapper(raw_f: Callable) -> PartialFunction:
        return PartialFunction(raw_f, {"web": {"type": "asgi", "label": label}})

    return wrapper


def wsgi_app(
    _warn_parentheses_missing: Any = None, *, label: Optional[str] = None, requires_proxy_auth: bool = False
) -> Callable:
    if _warn_parentheses_missing is not None:
        raise InvalidError("Use @modal.wsgi_app() with parentheses")

    def wrapper(raw_f: Callable) -> PartialFunction:
        return PartialFunction(raw_f, {"web": {"type": "wsgi", "label": label}})

    return wrapper


def web_server(
    port: int, *, startup_timeout: float = 5.0, label: Optional[str] = None, requires_proxy_auth: bool = False
) -> Callable:
    """Expose a subprocess HTTP server on ``port`` (reference :526)."""
    if not isinstance(port, int):
        raise InvalidError("@modal.web_server(port) requires a port number")

    def wrapper(raw_f: Callable) -> PartialFunction:
        return PartialFunction(
            raw_f,
            {"web": {"type": "web_server", "port": port, "startup_timeout": startup_timeout, "label": label}},
        )

    return wrapper
