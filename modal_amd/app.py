"""App framework: registry of functions/classes + run/deploy orchestration.

Parity: /root/reference/py/modal/app.py — ``_App`` (:140), decorators
``@app.function`` (:782), ``@app.cls`` (:1039), ``@app.local_entrypoint``
(:707), ``app.run()`` (:425), ``app.deploy()`` (:492), ``app.include()``
(:1480).
"""

from __future__ import annotations

import typing
from typing import Any, Callable, Optional, Sequence, Union

from ._sync import synchronizer, unwrap, wrap
from .exception import InvalidError
from .functions import Function, _Function
from .partial_function import PartialFunction
from .retries import Retries
from .schedule import Schedule

if typing.TYPE_CHECKING:
    pass


class _LocalEntrypoint:
    def __init__(self, raw_f: Callable, app: "App"):
        self.raw_f = raw_f
        self.app = app
        self.__name__ = raw_f.__name__

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self.raw_f(*args, **kwargs)


def _parse_gpu(gpu: Union[None, bool, str, int]) -> tuple[bool, int]:
    """Accept the reference's gpu= spellings: None/False, True, "any", "MI355X",
    "MI355X:4", or a count."""
    if gpu in (None, False):
        return False, 0
    if gpu is True:
        return True, 1
    if isinstance(gpu, int):
        return True, max(gpu, 1)
    if isinstance(gpu, str):
        count = 1
        spec = gpu
        if ":" in gpu:
            spec, _, count_s = gpu.partition(":")
            count = int(count_s)
        return True, count
    raise InvalidError(f"Unsupported gpu= value: {gpu!r}")


def _build_options(
    raw_f: Callable,
    flags: dict,
    *,
    name: Optional[str],
    gpu: Any,
    timeout: Optional[float],
    retries: Union[None, int, Retries],
    image: Any,
    secrets: Sequence[Any],
    volumes: dict,
    mounts: Sequence[Any],
    schedule: Optional[Schedule],
    cpu: Optional[float],
    memory: Optional[int],
    min_containers: Optional[int],
    max_containers: Optional[int],
    buffer_containers: Optional[int],
    scaledown_window: Optional[float],
    serialized: bool,
    cloud: Optional[str],
    region: Any,
    is_generator: Optional[bool],
    cluster_size: int = 0,
    extra: Optional[dict] = None,
) -> dict:
    needs_gpu, gpu_count = _parse_gpu(gpu)
    retry_dict = None
    if isinstance(retries, int):
        retry_dict = Retries(max_retries=retries, initial_delay=1.0)._to_policy_dict()
    elif isinstance(retries, Retries):
        retry_dict = retries._to_policy_dict()
    options: dict[str, Any] = {
        "name": name or raw_f.__name__,
        "needs_gpu": needs_gpu,
        "gpu_count": gpu_count,
        "timeout": timeout,
        "retries": retry_dict,
        "serialized": serialized,
        "cpu": cpu,
        "memory": memory,
        "min_containers": min_containers or 0,
        "max_containers": max_containers or 0,
        "buffer_containers": buffer_containers or 0,
        "scaledown_window": scaledown_window or 60.0,
        "cloud": cloud,
        "region": region,
        "is_generator": is_generator,
        "cluster_size": cluster_size,
        "secret_ids": [],  # filled at load time by the resolver
        "volume_mounts": {},  # path -> volume id, filled at load time
    }
    options["_secrets"] = list(secrets)
    options["_volumes"] = dict(volumes)
    options["_image"] = image
    options["_mounts"] = list(mounts)
    options["_schedule"] = schedule
    for key in (
        "max_concurrent_inputs",
        "target_concurrent_inputs",
        "batch_max_size",
        "batch_linger_ms",
        "cluster_size",
        "rdma",
    ):
        if key in flags and flags[key]:
            options[key] = flags[key]
    if flags.get("is_generator") is not None:
        options["is_generator"] = flags["is_generator"]
    if flags.get("web"):
        options["web_config"] = flags["web"]
    if extra:
        options.update(extra)
    placement = options.get("placement")
    if placement is not None and hasattr(placement, "to_dict"):
        options["placement"] = placement.to_dict()
    return options


class App:
    """An application: a named collection of functions and classes.

    The reference's ``modal.App`` (reference app.py:140). Public and
    synchronous by design — the heavy lifting happens in ``runner``.
    """

    _all_apps: dict[Optional[str], list["App"]] = {}

    def __init__(
        self,
        name: Optional[str] = None,
        *,
        image: Any = None,
        secrets: Sequence[Any] = (),
        volumes: Optional[dict] = None,
        include_source: bool = True,
    ):
        if name is not None and not isinstance(name, str):
            raise InvalidError("App name must be a string")
        self._name = name
        self._description = name
        self._image = image
        self._secrets = list(secrets)
        self._volumes = dict(volumes or {})
        self._include_source = include_source
        self._functions: dict[str, Function] = {}
        self._classes: dict[str, Any] = {}
        self._local_entrypoints: dict[str, _LocalEntrypoint] = {}
        self._app_id: Optional[str] = None
        self._running_client: Any = None
        self._run_state: Any = None
        App._all_apps.setdefault(name, []).append(self)

    # -- accessors -------------------------------------------------------
    @property
    def name(self) -> Optional[str]:
        return self._name

    @property
    def app_id(self) -> Optional[str]:
        return self._app_id

    @property
    def description(self) -> Optional[str]:
        return self._description or self._name

    @property
    def is_interactive(self) -> bool:
        return False

    def set_description(self, description: str) -> None:
        self._description = description

    @property
    def registered_functions(self) -> dict[str, Function]:
        return dict(self._functions)

    @property
    def registered_web_endpoints(self) -> list[str]:
        """Names of functions carrying web configs (parity: reference)."""
        out = []
        for name, fn in self._functions.items():
            impl = getattr(fn, "_impl", fn)
            if getattr(impl, "_web_config", None) or (
                getattr(impl, "_options", None) or {}
            ).get("web_config"):
                out.append(name)
        return out

    @property
    def image(self) -> Any:
        """The app-level default image, when one was set."""
        return getattr(self, "_image", None)

    @image.setter
    def image(self, value: Any) -> None:
        self._image = value

    def set_tags(self, tags: dict) -> None:
        """Attach metadata tags to the running app (parity: AppCreate tags)."""
        from ._sync import synchronizer

        client = self._running_client
        if client is None or self._app_id is None:
            raise InvalidError("set_tags requires a running app")
        synchronizer.run(client.svc.app_set_tags(app_id=self._app_id, tags=dict(tags)))

    def get_tags(self) -> dict:
        from ._sync import synchronizer

        client = self._running_client
        if client is None or self._app_id is None:
            raise InvalidError("get_tags requires a running app")
        return synchronizer.run(client.svc.app_get_tags(app_id=self._app_id))

    def logs(self, *, timeout: 'Any' = None) -> Any:
        """Iterate this app's log entries as text (parity: App.logs)."""
        if self._app_id is None:
            raise InvalidError("logs() requires a running or deployed app")
        from .logs_manager import tail_app_logs

        for entry in tail_app_logs(self._app_id, self._running_client, timeout=timeout):
            yield entry.get("data", "")

    @property
    def registered_classes(self) -> dict[str, Any]:
        return dict(self._classes)

    @property
    def registered_entrypoints(self) -> dict[str, _LocalEntrypoint]:
        return dict(self._local_entrypoints)

    # -- decorators ------------------------------------------------------
    def function(
        self,
        _warn_parentheses_missing: Any = None,
        *,
        name: Optional[str] = None,
        gpu: Any = None,
        image: Any = None,
        secrets: Sequence[Any] = (),
        volumes: Optional[dict] = None,
        mounts: Sequence[Any] = (),
        schedule: Optional[Schedule] = None,
        timeout: Optional[float] = None,
        retries: Union[None, int, Retries] = None,
        cpu: Optional[float] = None,
        memory: Optional[int] = None,
        min_containers: Optional[int] = None,
        max_containers: Optional[int] = None,
        buffer_containers: Optional[int] = None,
        scaledown_window: Optional[float] = None,
        serialized: bool = False,
        cloud: Optional[str] = None,
        region: Any = None,
        is_generator: Optional[bool] = None,
        enable_memory_snapshot: bool = False,
        **extra_kwargs: Any,
    ) -> Callable:
        """Register a function (reference app.py:782)."""
        if _warn_parentheses_missing is not None:
            raise InvalidError("Use @app.function() with parentheses")

        def decorator(f: Union[Callable, PartialFunction]) -> Function:
            flags: dict = {}
            raw_f = f
            if isinstance(f, PartialFunction):
                flags = dict(f.flags)
                raw_f = f.raw_f
            options = _build_options(
                raw_f,
                flags,
                name=name,
                gpu=gpu,
                timeout=timeout,
                retries=retries,
                image=image if image is not None else self._image,
                secrets=list(self._secrets) + list(secrets),
                volumes={**self._volumes, **(volumes or {})},
                mounts=mounts,
                schedule=schedule,
                cpu=cpu,
                memory=memory,
                min_containers=min_containers,
                max_containers=max_containers,
                buffer_containers=buffer_containers,
                scaledown_window=scaledown_window,
                serialized=serialized,
                cloud=cloud,
                region=region,
                is_generator=is_generator,
                extra={"enable_memory_snapshot": enable_memory_snapshot, **extra_kwargs},
            )
            fn_impl = _Function.from_local(raw_f, self, options)
            fn = wrap(fn_impl)
            tag = options["name"]
            if tag in self._functions:
                raise InvalidError(f"Function name '{tag}' is already registered on this app")
            self._functions[tag] = fn
            # if the app is live, register eagerly so .remote works right away
            if self._app_id is not None and self._running_client is not None:
                from .runner import register_function_live

                register_function_live(self, fn_impl)
            return fn

        return decorator

    def cls(
        self,
        _warn_parentheses_missing: Any = None,
        **function_kwargs: Any,
    ) -> Callable:
        """Register a class service (reference app.py:1039)."""
        if _warn_parentheses_missing is not None:
            raise InvalidError("Use @app.cls() with parentheses")

        def decorator(user_cls: type) -> Any:
            from .cls import make_cls

            cls_obj = make_cls(self, user_cls, function_kwargs)
            self._classes[user_cls.__name__] = cls_obj
            return cls_obj

        return decorator

    def local_entrypoint(
        self, _warn_parentheses_missing: Any = None, *, name: Optional[str] = None
    ) -> Callable:
        """Register a CLI entrypoint that runs locally inside ``modal run``
        (reference app.py:707)."""
        if _warn_parentheses_missing is not None:
            raise InvalidError("Use @app.local_entrypoint() with parentheses")

        def decorator(raw_f: Callable) -> _LocalEntrypoint:
            entrypoint = _LocalEntrypoint(raw_f, self)
            self._local_entrypoints[name or raw_f.__name__] = entrypoint
            return entrypoint

        return decorator

    def server(
        self,
        _warn_parentheses_missing: Any = None,
        *,
        port: int = 8000,
        startup_timeout: float = 30.0,
        **function_kwargs: Any,
    ) -> Callable:
        """Register a server class (reference app.py:1280 + _server.py):
        the class's @modal.enter starts an HTTP server on ``port``; the
        returned handle's start() waits for readiness (``startup_timeout``)
        inside the worker, then ``url`` serves on 127.0.0.1."""
        if _warn_parentheses_missing is not None:
            raise InvalidError("Use @app.server() with parentheses")

        def decorator(user_cls: type) -> Any:
            from .server import make_server

            return make_server(self, user_cls, port, startup_timeout, function_kwargs)

        return decorator

    def include(self, other: "App") -> "App":
        """Merge another app's registrations (reference app.py:1480)."""
        for tag, fn in other._functions.items():
            if tag in self._functions:
                raise InvalidError(f"Function '{tag}' exists in both apps")
            self._functions[tag] = fn
        for tag, cls_obj in other._classes.items():
            self._classes[tag] = cls_obj
        return self

    # -- run/deploy ------------------------------------------------------
    def run(
        self,
        *,
        client: Any = None,
        detach: bool = False,
        environment_name: str = "",
        interactive: bool = False,
    ) -> Any:
        """Context manager that makes the app live (reference app.py:425)."""
        from .runner import AppRunContext

        return AppRunContext(self, client=client, detach=detach, environment_name=environment_name)

    def deploy(
        self,
        *,
        name: Optional[str] = None,
        client: Any = None,
        environment_name: str = "",
    ) -> "App":
        """Deploy: register functions durably under the app name
        (reference app.py:492, runner.py:590)."""
        from .runner import deploy_app

        deploy_name = name or self._name
        if not deploy_name:
            raise InvalidError("Deploying requires a name: App('name') or deploy(name=...)")
        deploy_app(self, deploy_name, client=client, environment_name=environment_name)
        return self

    async def deploy_async(self, *, name: Optional[str] = None, client: Any = None) -> "App":
        from .runner import deploy_app_async

        deploy_name = name or self._name
        if not deploy_name:
            raise InvalidError("Deploying requires a name")
        await deploy_app_async(self, deploy_name, client=client)
        return self

    @classmethod
    def lookup(cls, name: str, *, client: Any = None, environment_name: str = "") -> "App":
        """Look up a deployed app by name."""
        from .client import _Client

        async def _lookup() -> Any:
            c = unwrap(client) if client is not None else await _Client.from_env()
            return await c.svc.app_lookup(name=name, environment=environment_name)

        resp = synchronizer.run(_lookup())
        app = cls(name)
        app._app_id = resp["app_id"]
        return app

    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        functions = self.__dict__.get("_functions", {})
        classes = self.__dict__.get("_classes", {})
        if name in functions:
            return functions[name]
        if name in classes:
            return classes[name]
        raise AttributeError(f"App has no registered function or class '{name}'")
