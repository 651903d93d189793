"""App run/deploy orchestration.

Parity: /root/reference/py/modal/runner.py — ``_run_app`` (:369): AppCreate,
15 s heartbeat loop (:431), log streaming (:447), ``_create_all_objects``
(:136-205) with function precreate-then-create (reference
_functions.py:912-933,1142-1149), AppPublish (:273), and
AppClientDisconnect + final log drain on exit (:326-366).
"""

from __future__ import annotations

import asyncio
from typing import Any, Optional

from ._object import Resolver
from ._serialization import serialize
from ._sync import synchronizer, unwrap
from .client import HEARTBEAT_INTERVAL, _Client


def _sanitize_options(options: dict) -> dict:
    """Strip local-only keys (underscore-prefixed handles) before the registry."""
    out = {}
    for key, value in options.items():
        if key.startswith("_"):
            continue
        out[key] = value
    return out


async def _prepare_function_options(fn_impl: Any, resolver: Resolver) -> dict:
    """Hydrate secrets/volumes/images referenced by the function and record ids."""
    options = dict(fn_impl._options)
    secret_ids = []
    for secret in options.get("_secrets") or []:
        impl = unwrap(secret)
        await resolver.load(impl)
        secret_ids.append(impl.object_id)
    options["secret_ids"] = secret_ids
    volume_mounts = {}
    for path, volume in (options.get("_volumes") or {}).items():
        impl = unwrap(volume)
        await resolver.load(impl)
        opts = getattr(impl, "_mount_options", None)
        if opts:
            volume_mounts[str(path)] = {"volume_id": impl.object_id, **opts}
        else:
            volume_mounts[str(path)] = impl.object_id
    options["volume_mounts"] = volume_mounts
    image = options.get("_image")
    if image is not None:
        impl = unwrap(image)
        try:
            await resolver.load(impl)
            options["image_id"] = impl.object_id
        except Exception:
            pass  # image building is best-effort locally
    proxy = options.pop("proxy", None)
    if proxy is not None:
        impl = unwrap(proxy)
        await resolver.load(impl)
        options["proxy_url"] = (getattr(impl, "_metadata", None) or {}).get("url")
    schedule = options.get("_schedule")
    if schedule is not None:
        from .schedule import Cron, Period

        if isinstance(schedule, Cron):
            options["schedule"] = {"cron": schedule.cron_string}
        elif isinstance(schedule, Period):
            options["schedule"] = {"period": schedule.total_seconds}
    region = options.get("region")
    if region is not None and not isinstance(region, (str, list)):
        options["region"] = str(region)
    return options


async def _register_functions(app: Any, client: _Client, app_id: str) -> None:
    """Two-phase registration: precreate ids for every function (so
    cross-references between functions serialize as hydrated handles), then
    upload real definitions (parity: FunctionPrecreate flow,
    reference _functions.py:912-933)."""
    resolver = Resolver(client, app_id=app_id)
    fn_impls = []
    for tag, fn in app._functions.items():
        fn_impls.append((tag, unwrap(fn)))
    # include class service functions
    for cls_obj in app._classes.values():
        service_fn = getattr(cls_obj, "_service_function", None)
        if service_fn is not None:
            fn_impls.append((cls_obj._user_cls.__name__, unwrap(service_fn)))

    # phase 1: precreate with empty definitions
    for tag, impl in fn_impls:
        if impl.is_hydrated:
            continue
        options = await _prepare_function_options(impl, resolver)
        impl._prepared_options = _sanitize_options(options)
        resp = await client.svc.function_create(
            app_id=app_id,
            name=options.get("name") or tag,
            definition=b"",
            options=impl._prepared_options,
        )
        impl._hydrate(resp["function_id"], client, resp["metadata"])
    # phase 2: real (cloudpickled) definitions — all handles now serializable
    for tag, impl in fn_impls:
        definition = impl._serialize_definition()
        await client.svc.function_update(
            function_id=impl.object_id, definition=definition
        )


def register_function_live(app: Any, fn_impl: Any) -> None:
    """Register a function added while the app is already running."""

    async def _register() -> None:
        client = app._running_client
        resolver = Resolver(client, app_id=app._app_id)
        options = await _prepare_function_options(fn_impl, resolver)
        resp = await client.svc.function_create(
            app_id=app._app_id,
            name=options["name"],
            definition=b"",
            options=_sanitize_options(options),
        )
        fn_impl._hydrate(resp["function_id"], client, resp["metadata"])
        await client.svc.function_update(
            function_id=fn_impl.object_id, definition=fn_impl._serialize_definition()
        )

    synchronizer.run(_register())


class AppRunContext:
    """Dual sync/async context manager returned by ``app.run()``."""

    def __init__(self, app: Any, client: Any = None, detach: bool = False, environment_name: str = ""):
        self.app = app
        self.client: Optional[_Client] = unwrap(client) if client is not None else None
        self.detach = detach
        self.environment_name = environment_name
        self._heartbeat_task: Optional[asyncio.Task] = None
        self._log_task: Optional[asyncio.Task] = None
        self._log_queue: Optional[asyncio.Queue] = None

    # -- async core ------------------------------------------------------
    async def _aenter(self) -> Any:
        app = self.app
        if self.client is None:
            self.client = await _Client.from_env()
        client = self.client
        resp = await client.svc.app_create(
            description=app.description or "(anonymous)",
            ephemeral=not self.detach,
            environment=self.environment_name,
        )
        app._app_id = resp["app_id"]
        app._running_client = client
        from .output import get_output_manager

        manager = get_output_manager()
        if manager is not None:
            with manager.status("Creating objects..."):
                await _register_functions(app, client, app._app_id)
        else:
            await _register_functions(app, client, app._app_id)
        if manager is not None:  # parity: OutputManager step lines
            manager.print_step(f"Initialized app {app.description or app._app_id}.")
            names = sorted(app._functions.keys()) if getattr(app, "_functions", None) else []
            if names:
                manager.print_step(f"Created functions: {', '.join(names)}.")
        self._heartbeat_task = asyncio.get_running_loop().create_task(self._heartbeat_loop())
        self._start_log_stream()
        return app

    async def _aexit(self, exc_type: Any, exc: Any, tb: Any) -> None:
        app = self.app
        client = self.client
        for task in (self._heartbeat_task, self._log_task):
            if task is not None:
                task.cancel()
        try:
            if client is not None and app._app_id is not None:
                await client.svc.app_client_disconnect(app_id=app._app_id)
        finally:
            app._running_client = None

    async def _heartbeat_loop(self) -> None:
        while True:
            await asyncio.sleep(HEARTBEAT_INTERVAL)
            try:
                await self.client.svc.app_heartbeat(app_id=self.app._app_id)
            except Exception:
                return

    def _start_log_stream(self) -> None:
        svc = self.client.svc
        from .output import get_output_manager

        manager = get_output_manager()
        if manager is None:
            return
        if getattr(svc, "is_proxy", False):
            # daemon-attached: tail over the offset-resumable long-poll RPC
            app_id = self.app._app_id

            async def drain_remote() -> None:
                offset = 0
                while True:
                    try:
                        resp = await svc.app_get_logs(
                            app_id=app_id, offset=offset, timeout=10.0
                        )
                    except asyncio.CancelledError:
                        raise
                    except Exception:
                        await asyncio.sleep(0.5)
                        continue
                    for entry in resp["entries"]:
                        manager.print_log(entry)
                    offset = resp["next_offset"]

            self._log_task = asyncio.get_running_loop().create_task(drain_remote())
            return
        state = svc.apps.get(self.app._app_id)
        if state is None:
            return
        queue: asyncio.Queue = asyncio.Queue()
        state.log_subscribers.append(queue)
        self._log_queue = queue

        async def drain() -> None:
            while True:
                entry = await queue.get()
                manager.print_log(entry)

        self._log_task = asyncio.get_running_loop().create_task(drain())

    # -- dual context manager -------------------------------------------
    def __enter__(self) -> Any:
        return synchronizer.run(self._aenter())

    def __exit__(self, exc_type: Any, exc: Any, tb: Any) -> None:
        synchronizer.run(self._aexit(exc_type, exc, tb))

    async def __aenter__(self) -> Any:
        return await synchronizer.run_async(self._aenter())

    async def __aexit__(self, exc_type: Any, exc: Any, tb: Any) -> None:
        await synchronizer.run_async(self._aexit(exc_type, exc, tb))


async def deploy_app_async(
    app: Any, name: str, client: Any = None, environment_name: str = ""
) -> str:
    """Deploy: create app, register functions, publish under ``name``
    (parity: reference runner.py:590 _deploy_app)."""
    client = unwrap(client) if client is not None else await _Client.from_env()
    resp = await client.svc.app_create(
        description=name, ephemeral=False, environment=environment_name
    )
    app_id = resp["app_id"]
    app._app_id = app_id
    app._running_client = client
    await _register_functions(app, client, app_id)
    await client.svc.app_publish(app_id=app_id, name=name)
    return app_id


def deploy_app(app: Any, name: str, client: Any = None, environment_name: str = "") -> str:
    return synchronizer.run(deploy_app_async(app, name, client, environment_name))
