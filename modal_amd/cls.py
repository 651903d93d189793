"""Class services: ``@app.cls()`` + parametrization.

Parity: /root/reference/py/modal/cls.py — ``_Cls`` (:453), ``_Obj`` (:143),
parameter binding -> a derived function registration (reference
FunctionBindParams :181-250), ``modal.parameter()`` fields, ``@enter``
instance lifecycle (:346). One scheduler function row backs the whole class;
methods dispatch by name; parametrized instances bind to derived rows.
"""

from __future__ import annotations

import threading
from typing import Any, Optional

from ._serialization import serialize
from ._sync import synchronizer, unwrap, wrap
from .exception import InvalidError
from .functions import _Function
from .partial_function import PartialFunction


class _Parameter:
    """Marker produced by ``modal.parameter()`` (reference strict proto params)."""

    def __init__(self, default: Any = ..., init: bool = True):
        self.default = default
        self.init = init


def parameter(*, default: Any = ..., init: bool = True) -> Any:
    return _Parameter(default=default, init=init)


def _collect_parameters(user_cls: type) -> dict[str, _Parameter]:
    params = {}
    for name, value in vars(user_cls).items():
        if isinstance(value, _Parameter):
            params[name] = value
    # also honor dataclass-style annotations with parameter() defaults in bases
    return params


def _collect_methods(user_cls: type) -> dict[str, dict]:
    """Find @modal.method()-decorated members and web endpoints."""
    methods = {}
    for name in dir(user_cls):
        attr = None
        for klass in user_cls.__mro__:
            if name in vars(klass):
                attr = vars(klass)[name]
                break
        if isinstance(attr, PartialFunction):
            methods[name] = dict(attr.flags)
    return methods


class Obj:
    """A (possibly parametrized) instance handle; methods are remote-callable."""

    def __init__(self, cls_obj: "Cls", args: tuple, kwargs: dict):
        self._cls = cls_obj
        self._args = args
        self._kwargs = kwargs
        self._bound_fn_impl: Optional[_Function] = None
        self._bind_lock = threading.Lock()
        if args:
            raise InvalidError(
                "Class services take keyword-only parameters (modal.parameter fields)"
            )
        known = cls_obj._parameters
        for key in kwargs:
            if known and key not in known:
                raise InvalidError(f"Unknown parameter {key!r} for {cls_obj._user_cls.__name__}")

    def _bind(self) -> _Function:
        """Create/fetch the derived function row carrying these parameters
        (parity: FunctionBindParams, reference cls.py:181-250)."""
        with self._bind_lock:
            if self._bound_fn_impl is not None:
                return self._bound_fn_impl
            base_impl = unwrap(self._cls._service_function)
            if not self._kwargs:
                self._bound_fn_impl = base_impl
                return base_impl

            async def bind() -> _Function:
                if not base_impl.is_hydrated:
                    await base_impl.hydrate()
                client = base_impl._client
                spec = self._cls._make_spec(self._kwargs)
                opts = dict(getattr(base_impl, "_prepared_options", None) or {})
                opts["is_class_service"] = True
                resp = await client.svc.function_create(
                    app_id=self._cls._app._app_id or "",
                    name=f"{self._cls._user_cls.__name__}.bound",
                    definition=serialize(spec),
                    options=opts,
                )
                impl = _Function._new_hydrated(resp["function_id"], client, resp["metadata"])
                return impl

            self._bound_fn_impl = synchronizer.run(bind())
            return self._bound_fn_impl

    def __getattr__(self, name: str) -> Any:
        methods = self._cls._methods
        if name in methods:
            impl = self._bind()
            method_impl = _Function._new_hydrated(impl.object_id, impl._client, impl._get_metadata())
            method_impl._method_name = name
            flags = methods[name]
            raw = None
            for klass in self._cls._user_cls.__mro__:
                if name in vars(klass):
                    raw = vars(klass)[name]
                    break
            if isinstance(raw, PartialFunction):
                import inspect

                rf = raw.raw_f
                method_impl._is_generator = bool(
                    flags.get("is_generator")
                    or inspect.isgeneratorfunction(rf)
                    or inspect.isasyncgenfunction(rf)
                )
            return wrap(method_impl)
        # fall back to a local instance attribute (local usage of the class)
        raise AttributeError(name)

    def local_instance(self) -> Any:
        """Instantiate the underlying class locally (for .local() flows)."""
        inst = self._cls._instantiate_local(self._kwargs)
        return inst


class Cls:
    """The registered class service (``cs-``)."""

    def __init__(self, app: Any, user_cls: type, function_kwargs: dict):
        self._app = app
        self._user_cls = user_cls
        self._function_kwargs = function_kwargs
        self._parameters = _collect_parameters(user_cls)
        self._methods = _collect_methods(user_cls)
        self._service_function = self._make_service_function()

    def _make_spec(self, kwargs: dict) -> dict:
        return {
            "kind": "cls_service",
            "cls": self._user_cls,
            "args": (),
            "kwargs": dict(kwargs),
        }

    def _make_service_function(self) -> Any:
        from .app import _build_options

        user_cls = self._user_cls
        fk = dict(self._function_kwargs)
        options = _build_options(
            raw_f=user_cls,  # only the name is used here
            flags={},
            name=user_cls.__name__,
            gpu=fk.pop("gpu", None),
            timeout=fk.pop("timeout", None),
            retries=fk.pop("retries", None),
            image=fk.pop("image", None) or self._app._image,
            secrets=list(self._app._secrets) + list(fk.pop("secrets", ())),
            volumes={**self._app._volumes, **(fk.pop("volumes", None) or {})},
            mounts=fk.pop("mounts", ()),
            schedule=None,
            cpu=fk.pop("cpu", None),
            memory=fk.pop("memory", None),
            min_containers=fk.pop("min_containers", None),
            max_containers=fk.pop("max_containers", None),
            buffer_containers=fk.pop("buffer_containers", None),
            scaledown_window=fk.pop("scaledown_window", None),
            serialized=fk.pop("serialized", False),
            cloud=fk.pop("cloud", None),
            region=fk.pop("region", None),
            is_generator=None,
            extra=fk,
        )
        # concurrency/batching flags can come from method decorators on the class
        for flags in self._methods.values():
            for key in ("max_concurrent_inputs", "target_concurrent_inputs", "batch_max_size", "batch_linger_ms"):
                if flags.get(key):
                    options[key] = flags[key]
        options["is_class_service"] = True
        options["metadata"] = {"methods": sorted(self._methods.keys())}
        options["_definition_provider"] = lambda: self._make_spec({})
        impl = _Function.from_local(user_cls, self._app, options)
        return wrap(impl)

    def _instantiate_local(self, kwargs: dict) -> Any:
        cls = self._user_cls
        has_custom_init = "__init__" in vars(cls)
        if has_custom_init:
            return cls(**kwargs)
        inst = cls.__new__(cls)
        for name, param in self._parameters.items():
            if param.default is not ...:
                setattr(inst, name, param.default)
        for key, value in kwargs.items():
            setattr(inst, key, value)
        return inst

    def __call__(self, *args: Any, **kwargs: Any) -> Obj:
        return Obj(self, args, kwargs)

    def with_options(self, **kwargs: Any) -> "Cls":
        merged = {**self._function_kwargs, **kwargs}
        return Cls(self._app, self._user_cls, merged)

    @staticmethod
    def validate_construction_mechanism(user_cls: type) -> None:
        """Reject classes mixing a custom __init__ with modal.parameter()
        fields, and unannotated parameters (parity: reference cls.py:566)."""
        params = {k for k, v in vars(user_cls).items() if isinstance(v, _Parameter)}
        has_custom_init = user_cls.__init__ is not object.__init__
        if params and has_custom_init:
            raise InvalidError(
                "A class can't have both a custom __init__ constructor "
                "and dataclass-style modal.parameter() annotations"
            )
        annotations = getattr(user_cls, "__annotations__", {})
        missing = params - set(annotations)
        if missing:
            raise InvalidError(
                "All modal.parameter() specifications need to be type-annotated "
                f"(missing: {sorted(missing)})"
            )

    @staticmethod
    def from_local(user_cls: type, app: Any, class_service_function: Any = None) -> "Cls":
        """Build a Cls from a local class definition (parity: reference
        cls.py:604 from_local — validation + registration in one step)."""
        Cls.validate_construction_mechanism(user_cls)
        return Cls(app, user_cls, {})

    def with_concurrency(self, *, max_inputs: int, target_inputs: int = 0) -> "Cls":
        """Parity: reference cls.py Cls.with_concurrency."""
        return self.with_options(
            max_concurrent_inputs=max_inputs,
            target_concurrent_inputs=target_inputs or max_inputs,
        )

    def with_batching(self, *, max_batch_size: int, wait_ms: int = 0) -> "Cls":
        """Parity: reference cls.py Cls.with_batching."""
        return self.with_options(batch_max_size=max_batch_size, batch_linger_ms=wait_ms)

    @classmethod
    def from_name(
        cls, app_name: str, name: str, *, environment_name: str = ""
    ) -> "_LazyRemoteCls":
        return _LazyRemoteCls(app_name, name, environment_name)


class _LazyRemoteCls:
    """Cls.from_name: methods resolve against a deployed class service."""

    def __init__(self, app_name: str, name: str, environment_name: str):
        self._app_name = app_name
        self._name = name
        self._environment_name = environment_name

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        if args:
            raise InvalidError("Class services take keyword-only parameters")
        return _LazyRemoteObj(self, kwargs)


class _LazyRemoteObj:
    def __init__(self, lazy_cls: _LazyRemoteCls, kwargs: dict):
        self._lazy_cls = lazy_cls
        self._kwargs = kwargs
        self._impl: Optional[_Function] = None
        self._methods: list[str] = []

    def _resolve(self) -> _Function:
        if self._impl is not None:
            return self._impl

        async def resolve() -> _Function:
            from .client import _Client

            client = await _Client.from_env()
            resp = await client.svc.function_lookup(
                app_name=self._lazy_cls._app_name,
                name=self._lazy_cls._name,
                environment=self._lazy_cls._environment_name,
            )
            impl = _Function._new_hydrated(resp["function_id"], client, resp["metadata"])
            self._methods = resp["metadata"].get("methods", [])
            return impl

        self._impl = synchronizer.run(resolve())
        return self._impl

    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        impl = self._resolve()
        method_impl = _Function._new_hydrated(impl.object_id, impl._client, None)
        method_impl._method_name = name
        return wrap(method_impl)


def make_cls(app: Any, user_cls: type, function_kwargs: dict) -> Cls:
    return Cls(app, user_cls, function_kwargs)
