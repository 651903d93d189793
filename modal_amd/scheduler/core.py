"""The in-process scheduler: single-node re-implementation of the control plane.

The reference client speaks 240 gRPC RPCs to Modal's closed cloud
(/root/reference/modal_proto/api.proto, service at :4680). Here the control
plane is this class: called directly (no serialization) from the client
process, and over Unix-socket RPC (scheduler/transport.py) from workers,
sandboxes and CLI tooling. Method names and semantics track the RPC groups
(App*, Function*, Queue*, Dict*, Secret*, Blob*) so behavior matches the
reference's mock servicer (/root/reference/py/test/conftest.py:701) where the
reference documents it.
"""

from __future__ import annotations

import asyncio
import logging
import os
import tempfile
import time
from collections import deque
from typing import Any, Optional

from ..exception import InvalidError, NotFoundError
from ..utils.ids import new_id
from .blobs import INLINE_LIMIT, BlobStore
from .calls import (
    GENERIC_STATUS_FAILURE,
    GENERIC_STATUS_INTERNAL_FAILURE,
    GENERIC_STATUS_SUCCESS,
    GENERIC_STATUS_TERMINATED,
    MAX_INPUTS_OUTSTANDING_DEFAULT,
    MAX_INTERNAL_FAILURE_COUNT,
    CallRecord,
    FunctionDef,
    InputRecord,
    RetryPolicy,
)
from .services import Services
from .workerhost import WorkerPool

OUTPUT_POLL_TIMEOUT = 55.0  # parity: function_utils.py:498 (55 s long-poll)


def _load_or_create_auth_token(run_dir: str) -> str:
    """Socket-handshake secret, stored 0600 inside the (0700) run_dir.

    Possessing the token == being able to read the run_dir, so the Unix
    socket no longer grants the RPC surface to arbitrary local processes."""
    path = os.path.join(run_dir, "auth.token")
    try:
        with open(path) as f:
            token = f.read().strip()
        if token:
            return token
    except OSError:
        pass
    import secrets

    token = secrets.token_hex(16)
    fd = os.open(path, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, 0o600)
    with os.fdopen(fd, "w") as f:
        f.write(token)
    return token


def read_auth_token(socket_path: str) -> str:
    """Client-side: the token lives next to the socket."""
    try:
        with open(os.path.join(os.path.dirname(socket_path), "auth.token")) as f:
            return f.read().strip()
    except OSError:
        return ""


class AppState:
    def __init__(self, app_id: str, description: str, ephemeral: bool, environment: str):
        self.app_id = app_id
        self.description = description
        self.ephemeral = ephemeral
        self.environment = environment
        self.state = "running"  # running | stopped | deployed
        self.created_at = time.time()
        self.last_heartbeat = time.time()
        self.objects: dict[str, tuple[str, dict]] = {}  # tag -> (object_id, metadata)
        self.deployment_name: Optional[str] = None
        self.logs: deque = deque(maxlen=10_000)
        self.log_subscribers: list[asyncio.Queue] = []
        self.log_seq = 0  # total entries EVER appended (offset-resume base)
        self.log_event = asyncio.Event()  # pulsed on each append (long-poll)


class RPCAdapter:
    """Whitelisted surface exposed to socket peers (workers, sandboxes, CLI)."""

    _ALLOWED = {
        "app_create", "app_publish", "app_heartbeat", "app_client_disconnect", "app_stop",
        "app_list", "app_history", "app_rollback", "app_set_objects", "app_get_object",
        "function_create", "function_update", "function_update_autoscaler",
        "queue_get_or_create", "queue_put", "queue_get", "queue_len", "queue_clear",
        "queue_peek", "queue_delete",
        "dict_get_or_create", "dict_update", "dict_get", "dict_pop", "dict_contains",
        "dict_len", "dict_items", "dict_clear", "dict_delete",
        "secret_get_or_create", "secret_env",
        "blob_put", "blob_get", "blob_path",
        "function_lookup", "function_map", "function_put_inputs", "function_put_chunk",
        "function_finish_inputs",
        "function_get_outputs", "function_call_cancel", "function_call_info",
        "function_get_current_stats", "generator_poll", "node_stats", "node_metrics",
        "container_exec", "container_stop", "endpoint_list",
        "app_lookup", "app_get_layout", "cluster_hello",
        "volume_get_or_create", "volume_put_file_blocks", "volume_get_file", "volume_list_files",
        "volume_remove_file", "volume_copy_files", "volume_commit", "volume_reload",
        "volume_delete", "volume_rename", "volume_dir",
        "sandbox_create", "sandbox_wait", "sandbox_terminate", "sandbox_poll", "sandbox_stdio_read",
        "sandbox_stdin_write", "sandbox_exec", "sandbox_list", "sandbox_set_tags",
        "sandbox_from_name", "sandbox_snapshot_fs", "sandbox_fs_op", "sandbox_resize",
        "sandbox_snapshot_dir", "sandbox_mount_image", "sandbox_unmount_image",
        "sandbox_connect_token", "named_objects_list", "named_object_delete", "image_publish", "image_from_name",
        "proxy_get_or_create", "proxy_stats",
        "image_get_or_create", "image_info", "mount_get_or_create",
        "device_transfer", "tensor_pull_relay",
        "worker_snapshot", "worker_restore", "start_grpc_bridge", "app_get_logs",
        "object_info", "secret_update", "volume_info", "app_set_tags", "app_get_tags",
    }

    def __init__(self, scheduler: "Scheduler"):
        self._scheduler = scheduler

    def __getattr__(self, name: str) -> Any:
        if name in RPCAdapter._ALLOWED:
            return getattr(self._scheduler, name)
        raise AttributeError(name)


class Scheduler:
    def __init__(self, run_dir: Optional[str] = None):
        from ..config import config

        self.run_dir = run_dir or config.get("run_dir") or tempfile.mkdtemp(prefix="modal-amd-")
        os.makedirs(self.run_dir, exist_ok=True)
        try:
            os.chmod(self.run_dir, 0o700)  # the socket grants full control
        except OSError:
            pass
        # connection handshake token: any process that can READ the run_dir
        # may use the control socket; anything else is rejected at hello
        # (round-1 review: the socket was an unauthenticated control plane)
        self.auth_token = _load_or_create_auth_token(self.run_dir)
        self.apps: dict[str, AppState] = {}
        self.app_names: dict[tuple[str, str], str] = {}  # (env, name) -> app_id
        self.functions: dict[str, FunctionDef] = {}
        self.function_names: dict[tuple[str, str, str], str] = {}  # (env, app_name, fn) -> fu-id
        self.calls: dict[str, CallRecord] = {}
        # shared output chunks: chunk_id -> {"data": bytes, "refs": int}
        self.out_chunks: dict[str, dict] = {}
        self._out_chunk_order: list[str] = []
        self._out_chunk_seq = 0
        self.services = Services()
        self.blob_store = BlobStore(os.path.join(self.run_dir, "blobs"))
        from .sandboxes import SandboxService
        from .volumes import VolumeService

        self.sandbox_service = SandboxService(self.run_dir)
        self.volume_service = VolumeService(self.run_dir, self.blob_store)
        from .images import ImageService

        self.image_service = ImageService(self.run_dir, self.blob_store)
        self.mounts: dict[str, str] = {}  # mount_id -> materialized dir
        self._mounts_by_hash: dict[str, str] = {}
        from .web import WebGateway

        self.web_gateway = WebGateway(self)
        self.pool = WorkerPool(self)
        self.rpc_adapter = RPCAdapter(self)
        self._started = False
        self._start_lock = asyncio.Lock()
        self.default_environment = "main"
        self._extra: dict[str, Any] = {}  # extension services (volumes, sandboxes, images)
        self._persist_task: Optional[asyncio.Task] = None
        self._persist_digest = b""
        # observability counters (SURVEY §5.5 MI355X line: items/sec +
        # p50/p99 in the scheduler itself; exposed via node_metrics)
        from collections import deque as _deque

        self.metrics_counters = {"inputs_total": 0, "outputs_total": 0, "failures_total": 0}
        self._unary_latencies: Any = _deque(maxlen=4096)
        self._rate_window: Any = _deque(maxlen=64)
        from . import persist

        try:  # same run_dir => same deployments (durable control plane)
            persist.load(self)
        except Exception:
            logging.getLogger("modal_amd.scheduler").warning(
                "state restore failed", exc_info=True
            )
        try:  # unfinished detached spawn calls re-enter the dispatch queue
            from . import wal

            wal.replay(self)
        except Exception:
            logging.getLogger("modal_amd.scheduler").warning(
                "WAL replay failed", exc_info=True
            )

    # -- lifecycle -------------------------------------------------------
    async def start(self) -> None:
        async with self._start_lock:
            if self._started:
                return
            await self.pool.start()
            from .cron import ScheduleRunner

            self.schedule_runner = ScheduleRunner(self)
            self.schedule_runner.start()
            self._persist_task = asyncio.get_running_loop().create_task(self._persist_loop())
            self._call_gc_task = asyncio.get_running_loop().create_task(self._call_gc_loop())
            if os.environ.get("MODAL_AMD_GRPC") == "1":
                await self.start_grpc_bridge()
            self._started = True

    async def start_grpc_bridge(self) -> str:
        """api.proto gRPC plane on <run_dir>/grpc.sock (proto/bridge.py);
        lazy because grpc import costs ~0.5 s the fast path never pays."""
        bridge = self._extra.get("grpc_bridge")
        if bridge is None:
            from ..proto.bridge import GrpcBridge

            bridge = GrpcBridge(self)
            await bridge.start()
            self._extra["grpc_bridge"] = bridge
        return bridge.socket_path

    async def stop(self) -> None:
        if not self._started:
            return
        bridge = self._extra.pop("grpc_bridge", None)
        if bridge is not None:
            try:
                await bridge.stop()
            except Exception:
                pass
        if getattr(self, "schedule_runner", None) is not None:
            self.schedule_runner.stop()
        if self._persist_task is not None:
            self._persist_task.cancel()
            self._persist_task = None
        if getattr(self, "_call_gc_task", None) is not None:
            self._call_gc_task.cancel()
            self._call_gc_task = None
        from . import persist

        try:  # final snapshot so a clean stop never loses deployments
            self._persist_digest = persist.save_if_changed(self, self._persist_digest)
        except Exception:
            pass
        await self.sandbox_service.shutdown()
        await self.web_gateway.stop()
        await self.pool.stop()
        self._started = False

    #: completed call records are kept this long for late .get()/gather
    #: (parity: the reference expires spawn results server-side too)
    CALL_RETENTION_SECONDS = float(os.environ.get("MODAL_AMD_CALL_RETENTION", "3600"))

    def _gc_calls_once(self) -> int:
        now = time.time()
        dropped = 0
        for call_id, rec in list(self.calls.items()):
            if (
                rec.done_event.is_set()
                and rec.finished_at is not None
                and now - rec.finished_at > self.CALL_RETENTION_SECONDS
            ):
                self.calls.pop(call_id, None)
                # release any undrained chunk payloads (unlinks one-shot
                # xfer spill files a cancelled/abandoned map left behind)
                for chunk_id in list(rec.chunks):
                    rec._drop_chunk(chunk_id)
                dropped += 1
        # stopped ephemeral apps: logs deques (10k entries each) are the cost
        for app_id, app in list(self.apps.items()):
            if (
                app.state == "stopped"
                and now - app.last_heartbeat > self.CALL_RETENTION_SECONDS
            ):
                self.apps.pop(app_id, None)
                dropped += 1
        # function rows of dropped apps hold cloudpickled definitions
        for fid, fdef in list(self.functions.items()):
            if fdef.app_id and fdef.app_id not in self.apps:
                self.functions.pop(fid, None)
                dropped += 1
        self.function_names = {
            k: v for k, v in self.function_names.items() if v in self.functions
        }
        # finished sandboxes keep stdio buffers: drop them after retention
        svc = self.sandbox_service
        for sb_id, sb in list(svc.sandboxes.items()):
            if (
                sb.main.returncode is not None
                and all(e.returncode is not None for e in sb.execs.values())
                and now - sb.created_at > self.CALL_RETENTION_SECONDS
            ):
                svc.sandboxes.pop(sb_id, None)
                dropped += 1
        return dropped

    async def _call_gc_loop(self) -> None:
        """Drop long-completed call records so a long-lived daemon does
        not accumulate them unboundedly."""
        gc_rounds = 0
        while True:
            await asyncio.sleep(60.0)
            gc_rounds += 1
            if gc_rounds % 10 == 0:
                try:  # leftover per-sandbox cgroup dirs (members long dead)
                    from .isolation import cleanup_stale_cgroups

                    cleanup_stale_cgroups()
                except Exception:
                    pass
            try:
                self._gc_calls_once()
            except Exception:
                pass

    async def _persist_loop(self) -> None:
        """Snapshot named/deployed state every 2 s when it changed
        (scheduler/persist.py; writes are atomic replaces)."""
        from . import persist

        while True:
            await asyncio.sleep(2.0)
            try:
                # on the loop: snapshot iterates live dicts that only the
                # loop mutates (state is small; the write is one file)
                self._persist_digest = persist.save_if_changed(self, self._persist_digest)
            except Exception:
                logging.getLogger("modal_amd.scheduler").warning(
                    "state snapshot failed", exc_info=True
                )

    def log(self, message: str) -> None:
        import logging

        logging.getLogger("modal_amd.scheduler").debug(message)

    # -- apps ------------------------------------------------------------
    async def app_create(
        self, description: str = "", ephemeral: bool = True, environment: str = ""
    ) -> dict:
        app_id = new_id("app")
        env = environment or self.default_environment
        self.apps[app_id] = AppState(app_id, description, ephemeral, env)
        return {"app_id": app_id}

    async def app_publish(self, app_id: str, name: str) -> dict:
        app = self._app(app_id)
        app.deployment_name = name
        app.state = "deployed"
        app.ephemeral = False
        previous = self.app_names.get((app.environment, name))
        self.app_names[(app.environment, name)] = app_id
        history = self._extra.setdefault("deploy_history", {}).setdefault(
            (app.environment, name), []
        )
        history.append({"app_id": app_id, "deployed_at": time.time(), "version": len(history) + 1})
        for tag, (object_id, _meta) in app.objects.items():
            if object_id.startswith("fu-"):
                self.function_names[(app.environment, name, tag)] = object_id
        return {"app_id": app_id, "url": f"local://{name}", "version": len(history)}

    async def app_history(self, name: str, environment: str = "") -> list:
        env = environment or self.default_environment
        return list(self._extra.get("deploy_history", {}).get((env, name), []))

    async def app_rollback(self, name: str, environment: str = "", version: int = -1) -> dict:
        """Re-point the deployed name at an earlier version (parity:
        modal app rollback / deployment history RPCs, reference runner.py:590)."""
        env = environment or self.default_environment
        history = self._extra.get("deploy_history", {}).get((env, name), [])
        if not history:
            raise NotFoundError(f"No deployment history for '{name}'")
        entry = history[version if version != -1 else -2 if len(history) > 1 else -1]
        app_id = entry["app_id"]
        app = self._app(app_id)
        self.app_names[(env, name)] = app_id
        for tag, (object_id, _meta) in app.objects.items():
            if object_id.startswith("fu-"):
                self.function_names[(env, name, tag)] = object_id
        return {"app_id": app_id, "version": entry["version"]}

    async def app_lookup(self, name: str, environment: str = "") -> dict:
        env = environment or self.default_environment
        app_id = self.app_names.get((env, name))
        if app_id is None:
            raise NotFoundError(f"App '{name}' not found in environment '{env}'")
        return {"app_id": app_id}

    async def app_set_objects(self, app_id: str, objects: dict[str, tuple[str, dict]]) -> None:
        self._app(app_id).objects.update(objects)

    async def app_get_object(self, app_name: str, tag: str, environment: str = "") -> dict:
        """Look up any object published by a deployed app, by tag (parity:
        reference api.proto AppGetObjects / deployment object lookup)."""
        env = environment or self.default_environment
        app_id = self.app_names.get((env, app_name))
        if app_id is None:
            raise NotFoundError(f"App '{app_name}' not found in environment '{env}'")
        entry = self._app(app_id).objects.get(tag)
        if entry is None:
            raise NotFoundError(f"App '{app_name}' has no object tagged '{tag}'")
        object_id, meta = entry
        return {"object_id": object_id, "metadata": meta or {}}

    async def app_get_layout(self, app_id: str) -> dict:
        return self.app_layout(app_id)

    def app_layout(self, app_id: str) -> dict:
        app = self.apps.get(app_id)
        if app is None:
            return {"app_id": app_id, "objects": {}}
        return {
            "app_id": app_id,
            "objects": {tag: [oid, meta] for tag, (oid, meta) in app.objects.items()},
        }

    async def app_heartbeat(self, app_id: str) -> None:
        self._app(app_id).last_heartbeat = time.time()

    async def app_client_disconnect(self, app_id: str) -> None:
        app = self.apps.get(app_id)
        if app is None:
            return
        if app.ephemeral:
            app.state = "stopped"
            await self._cancel_app_calls(app_id)
            await self._notify_app_stop(app_id)

    async def _notify_app_stop(self, app_id: str) -> None:
        """Tell workers to tear down this app's services and WAIT for them:
        @exit hooks + exit-time volume commit finish before app.run()
        returns (parity: container shutdown lifecycle,
        task_lifecycle_manager.py:78,117)."""

        async def one(w: Any) -> None:
            try:
                await w.conn.call("app_stop", {"app_id": app_id}, timeout=30)
            except Exception:
                pass

        await asyncio.gather(*(one(w) for w in list(self.pool.workers.values())))

    async def app_stop(self, app_id: str) -> None:
        app = self._app(app_id)
        app.state = "stopped"
        if app.deployment_name:
            self.app_names.pop((app.environment, app.deployment_name), None)
        await self._cancel_app_calls(app_id)
        await self._notify_app_stop(app_id)

    async def app_list(self, environment: str = "") -> list[dict]:
        env = environment or self.default_environment
        return [
            {
                "app_id": a.app_id,
                "description": a.description,
                "state": a.state,
                "name": a.deployment_name,
                "created_at": a.created_at,
            }
            for a in self.apps.values()
            if a.environment == env
        ]

    async def _cancel_app_calls(self, app_id: str) -> None:
        fids = {fid for fid, f in self.functions.items() if f.app_id == app_id}
        for call in self.calls.values():
            if call.function_id in fids and not call.done_event.is_set():
                await self.function_call_cancel(call.call_id, terminate_containers=False)

    def _app(self, app_id: str) -> AppState:
        app = self.apps.get(app_id)
        if app is None:
            raise NotFoundError(f"App {app_id} not found")
        return app

    # -- function registry ------------------------------------------------
    async def function_create(
        self,
        app_id: str,
        name: str,
        definition: bytes,
        options: Optional[dict] = None,
    ) -> dict:
        options = options or {}
        fid = new_id("function")
        fdef = FunctionDef(
            function_id=fid,
            app_id=app_id,
            name=name,
            definition=definition,
            definition_kind=options.get("definition_kind", "serialized"),
            is_generator=bool(options.get("is_generator")),
            needs_gpu=bool(options.get("needs_gpu")),
            gpu_count=int(options.get("gpu_count", 1 if options.get("needs_gpu") else 0)),
            timeout=options.get("timeout"),
            retry_policy=RetryPolicy.from_dict(options.get("retries")),
            max_concurrent_inputs=int(options.get("max_concurrent_inputs", 1)),
            target_concurrent_inputs=int(options.get("target_concurrent_inputs", 0)),
            batch_max_size=int(options.get("batch_max_size", 0)),
            batch_linger_ms=int(options.get("batch_linger_ms", 0)),
            is_method=bool(options.get("is_method")),
            is_class_service=bool(options.get("is_class_service")),
            cluster_size=int(options.get("cluster_size", 0)),
            min_containers=int(options.get("min_containers", 0)),
            max_containers=int(options.get("max_containers", 0)),
            buffer_containers=int(options.get("buffer_containers", 0)),
            scaledown_window=float(options.get("scaledown_window", 60.0)),
            metadata={
                **(options.get("metadata") or {}),
                **(
                    {"enable_memory_snapshot": True}
                    if options.get("enable_memory_snapshot")
                    else {}
                ),
            },
            web_config=options.get("web_config"),
            secret_ids=list(options.get("secret_ids") or []),
            volume_mounts=dict(options.get("volume_mounts") or {}),
            schedule=options.get("schedule"),
            image_id=options.get("image_id"),
            placement=options.get("placement"),
            proxy_url=options.get("proxy_url"),
        )
        self.functions[fid] = fdef
        if fdef.min_containers or fdef.buffer_containers:
            asyncio.get_running_loop().create_task(self.pool.ensure_min(fdef))
        if fdef.web_config:
            await self.web_gateway.ensure_started()
            label = fdef.web_config.get("label") or name
            fdef.metadata["web_url"] = self.web_gateway.register(label, fid)
        app = self.apps.get(app_id)
        if app is not None:
            app.objects[name] = (fid, fdef.public_metadata())
        return {"function_id": fid, "metadata": fdef.public_metadata()}

    async def function_update(self, function_id: str, definition: bytes, options: Optional[dict] = None) -> None:
        fdef = self.functions[function_id]
        fdef.definition = definition
        fdef.definition_version += 1
        for key, value in (options or {}).items():
            if hasattr(fdef, key):
                setattr(fdef, key, value)

    async def function_lookup(self, app_name: str, name: str, environment: str = "") -> dict:
        env = environment or self.default_environment
        fid = self.function_names.get((env, app_name, name))
        if fid is None:
            raise NotFoundError(f"Function '{app_name}/{name}' not found")
        return {"function_id": fid, "metadata": self.functions[fid].public_metadata()}

    async def node_stats(self) -> dict:
        """Node-level gauges: workers, in-flight inputs, per-GPU HBM
        (SURVEY §5.5's MI355X metrics obligation)."""
        workers = []
        for w in self.pool.workers.values():
            row = {
                "worker_id": w.worker_id,
                "task_id": w.task_id,
                "gpu_index": w.gpu_index,
                "alive": w.alive,
                "inflight": len(w.inflight),
                "last_heartbeat_age": max(0.0, time.time() - w.last_heartbeat),
            }
            if w.gpu_stats:
                row["gpu"] = w.gpu_stats
            workers.append(row)
        return {
            "workers": workers,
            "pending_inputs": sum(len(q) for q in self.pool.pending.values()),
            "active_calls": sum(
                1 for c in self.calls.values() if not c.done_event.is_set()
            ),
        }

    def _worker_by_task(self, task_id: str) -> Any:
        for w in self.pool.workers.values():
            if w.task_id == task_id:
                return w
        raise NotFoundError(f"Container {task_id} not found")

    async def container_exec(self, task_id: str, cmd: list, timeout: float = 60.0) -> dict:
        """Run a command inside a worker's context (parity: modal
        container exec, reference cli/container.py:297)."""
        w = self._worker_by_task(task_id)
        return await w.conn.call("exec_command", {"cmd": cmd, "timeout": timeout}, timeout=timeout + 10)

    async def container_stop(self, task_id: str) -> None:
        """Stop one worker (parity: modal container stop, cli/container.py:318)."""
        w = self._worker_by_task(task_id)
        w.draining = True
        await w.conn.send({"t": "shutdown"})

    async def endpoint_list(self) -> list:
        """Registered web endpoints (parity: modal endpoint list,
        reference cli/endpoint.py)."""
        gw = self.web_gateway
        return [
            {"label": label, "function_id": fid,
             "url": gw.url_for(label) if gw.port else None}
            for label, fid in gw.routes.items()
        ]

    async def node_metrics(self) -> str:
        """Prometheus-exposition snapshot of the gauges/counters SURVEY §5.5
        obligates: per-GPU HBM, worker liveness, items/sec, p50/p99 latency."""
        now = time.time()
        counters = self.metrics_counters
        self._rate_window.append((now, counters["outputs_total"]))
        lines = [
            "# TYPE modal_amd_inputs_total counter",
            f"modal_amd_inputs_total {counters['inputs_total']}",
            "# TYPE modal_amd_outputs_total counter",
            f"modal_amd_outputs_total {counters['outputs_total']}",
            "# TYPE modal_amd_failures_total counter",
            f"modal_amd_failures_total {counters['failures_total']}",
        ]
        if len(self._rate_window) >= 2:
            (t0, c0), (t1, c1) = self._rate_window[0], self._rate_window[-1]
            if t1 > t0:
                lines += [
                    "# TYPE modal_amd_items_per_sec gauge",
                    f"modal_amd_items_per_sec {(c1 - c0) / (t1 - t0):.3f}",
                ]
        if self._unary_latencies:
            lat = sorted(self._unary_latencies)
            lines += [
                "# TYPE modal_amd_unary_latency_seconds summary",
                f'modal_amd_unary_latency_seconds{{quantile="0.5"}} {lat[len(lat) // 2]:.6f}',
                f'modal_amd_unary_latency_seconds{{quantile="0.99"}} {lat[min(len(lat) - 1, int(len(lat) * 0.99))]:.6f}',
            ]
        lines += [
            "# TYPE modal_amd_workers gauge",
            f"modal_amd_workers {len(self.pool.workers)}",
            "# TYPE modal_amd_pending_inputs gauge",
            f"modal_amd_pending_inputs {sum(len(q) for q in self.pool.pending.values())}",
            "# TYPE modal_amd_active_calls gauge",
            f"modal_amd_active_calls {sum(1 for c in self.calls.values() if not c.done_event.is_set())}",
        ]
        gpu_lines = []
        for w in self.pool.workers.values():
            if w.gpu_stats and w.gpu_index is not None:
                g = w.gpu_stats
                gpu_lines.append(
                    f'modal_amd_gpu_hbm_free_bytes{{gpu="{w.gpu_index}",worker="{w.worker_id}"}} {g.get("hbm_free", 0)}'
                )
                gpu_lines.append(
                    f'modal_amd_gpu_hbm_total_bytes{{gpu="{w.gpu_index}",worker="{w.worker_id}"}} {g.get("hbm_total", 0)}'
                )
        if gpu_lines:
            lines.append("# TYPE modal_amd_gpu_hbm_free_bytes gauge")
            lines.extend(gpu_lines)
        return "\n".join(lines) + "\n"

    async def function_get_current_stats(self, function_id: str) -> dict:
        backlog = len(self.pool.pending.get(function_id, ()))
        runners = sum(
            1 for w in self.pool.workers.values() if w.outstanding.get(function_id, 0) > 0
        )
        return {"backlog": backlog, "num_total_tasks": runners}

    async def function_update_autoscaler(
        self,
        function_id: str,
        min_containers: Optional[int] = None,
        max_containers: Optional[int] = None,
        buffer_containers: Optional[int] = None,
        scaledown_window: Optional[float] = None,
    ) -> None:
        fdef = self.functions[function_id]
        if min_containers is not None:
            fdef.min_containers = min_containers
        if max_containers is not None:
            fdef.max_containers = max_containers
        if buffer_containers is not None:
            fdef.buffer_containers = buffer_containers
        if scaledown_window is not None:
            fdef.scaledown_window = scaledown_window

    # -- invocation -------------------------------------------------------
    async def function_map(
        self,
        function_id: str,
        kind: str = "unary",
        pipelined_inputs: Optional[list] = None,
        return_exceptions: bool = False,
    ) -> dict:
        """Create a function call; optionally enqueue the first inputs in the
        same round-trip (parity: FunctionMap w/ pipelined_inputs,
        reference _functions.py:163-188)."""
        if function_id not in self.functions:
            raise NotFoundError(f"Function {function_id} not found")
        record = CallRecord(function_id, kind, return_exceptions)
        self.calls[record.call_id] = record
        from . import wal

        if wal.journaled(self, record):
            # detached spawn work on a deployed app survives scheduler
            # restarts (SURVEY hard part 5: WAL in scheduler)
            record.durable = True
            wal.journal_created(self, record)
        if pipelined_inputs:
            await self.function_put_inputs(record.call_id, pipelined_inputs)
        return {
            "function_call_id": record.call_id,
            "retry_policy": self.functions[function_id].retry_policy.to_dict(),
            # server-sized (the reference makes this server-overridable,
            # parallel_map.py:387): the default 1,000 items is ~7 chunks of
            # 128 — enough for one worker's pipeline but it STARVES a
            # multi-GPU pool. Scale with the live worker count sized for
            # ~4 chunks of up to 256 items in flight PER worker (pipeline
            # depth; a cap of one chunk per worker leaves workers idle
            # between completions — measured 372k -> 1M+ items/s on the
            # batched-noop probe when deepened)
            "max_inputs_outstanding": max(
                MAX_INPUTS_OUTSTANDING_DEFAULT,
                1024 * max(
                    sum(1 for w in self.pool.workers.values() if w.alive), 1
                ),
            ),
            "sync_client_retries_enabled": True,
        }

    async def function_put_inputs(
        self, function_call_id: str, items: list, chunks: Optional[dict] = None
    ) -> list:
        """items: [{"payload": bytes, "method": str} | {"chunk": id, "ci": i}]
        -> [{"idx", "input_id"}]. ``chunks`` carries shared pickled arg-lists
        (the map fast path: one pickle per ~64 inputs)."""
        record = self._call(function_call_id)
        self.metrics_counters["inputs_total"] += len(items)
        fdef = self.functions.get(record.function_id)
        if chunks:
            for chunk_id, data in chunks.items():
                record.chunks[chunk_id] = {"data": data, "refs": 0}
        out = []
        batch_recs = []
        for item in items:
            if isinstance(item, (bytes, bytearray)):
                item = {"payload": bytes(item)}
            if fdef is not None and fdef.cluster_size > 1:
                recs = await self._put_gang_input(record, fdef, item)
                out.append({"idx": recs[0].idx, "input_id": recs[0].input_id})
                continue
            rec = record.add_input(
                item.get("payload") or b"",
                item.get("method", ""),
                tensors=item.get("tensors"),
                payload_blob=item.get("payload_blob"),
            )
            if getattr(record, "durable", False):
                from . import wal

                wal.journal_input(self, record, rec)
            chunk_id = item.get("chunk")
            if chunk_id:
                rec.chunk_id = chunk_id
                rec.chunk_index = item.get("ci", 0)
                chunk = record.chunks.get(chunk_id)
                if chunk is not None:
                    chunk["refs"] += 1
            batch_recs.append(rec)
            out.append({"idx": rec.idx, "input_id": rec.input_id})
        if batch_recs:
            self.pool.enqueue_many(record.function_id, batch_recs)
        return out

    async def _put_gang_input(self, record: CallRecord, fdef: FunctionDef, item: dict) -> list:
        """Gang scheduling for @clustered(size=n): the same input lands on n
        distinct workers simultaneously, each with rank/world identity; the
        rank-0 output is the call's result (parity: reference clustered
        semantics, _clustered_functions.py:42-94)."""
        import socket

        n = fdef.cluster_size
        cluster_id = new_id("task")
        # pre-allocate the rendezvous port for torch.distributed/RCCL
        sock = socket.socket()
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
        sock.close()
        recs = []
        for rank in range(n):
            rec = record.add_input(
                item.get("payload") or b"",
                item.get("method", ""),
                payload_blob=item.get("payload_blob"),
            )
            rec.cluster = {
                "rank": rank,
                "size": n,
                "cluster_id": cluster_id,
                "master_port": port,
            }
            recs.append(rec)
        await self.pool.dispatch_gang(fdef, recs)
        return recs

    async def function_put_chunk(
        self,
        function_call_id: str,
        chunk_id: str,
        payload: Any,
        count: int,
        method: str = "",
    ) -> dict:
        """Range-protocol intake: one call registers a whole chunk of inputs
        (~64) with NO per-item records; items materialize only on failure.
        Falls back to per-item intake for functions needing per-item
        treatment (gangs, batching, generators, timeouts)."""
        record = self._call(function_call_id)
        fdef = self.functions.get(record.function_id)
        fast = (
            fdef is not None
            and fdef.cluster_size <= 1
            and not fdef.is_generator
            and not fdef.timeout
            and not fdef.web_config
            and not os.environ.get("MODAL_AMD_NO_RANGE")  # A/B escape hatch
        )
        if not fast:
            items = [{"chunk": chunk_id, "ci": ci, "method": method} for ci in range(count)]
            return {
                "items": await self.function_put_inputs(
                    function_call_id, items, chunks={chunk_id: payload}
                )
            }
        group = record.add_chunk(chunk_id, payload, count, method)
        self.metrics_counters["inputs_total"] += count
        self.pool.enqueue_chunk(record, group, record.function_id)
        return {"idx_base": group.base_idx, "count": count}

    async def function_finish_inputs(self, function_call_id: str) -> None:
        record = self._call(function_call_id)
        record.finish_inputs()
        if getattr(record, "durable", False):
            from . import wal

            wal.journal_finish(self, record)

    async def function_wait_output(
        self, function_call_id: str, idx: int = 0, timeout: Optional[float] = None
    ) -> InputRecord:
        """In-process fast path: await one input's final output directly."""
        return await self._call(function_call_id).wait_output(idx, timeout)

    async def function_get_outputs(
        self,
        function_call_id: str,
        max_values: int = 256,
        timeout: float = OUTPUT_POLL_TIMEOUT,
        clear_on_success: bool = True,
    ) -> list[dict]:
        """Streaming poll: completed outputs in completion order
        (parity: FunctionGetOutputs long-poll, reference _functions.py:224-263)."""
        record = self._call(function_call_id)
        out: list[dict] = []
        n_vals = 0  # TRUE number of outputs in this response (groups expanded)

        def expand(entry: Any) -> None:
            # honest max_values: a chunk-group entry counts as its item count;
            # a group that does not fit is SPLIT — the head ships now, the
            # tail goes to output_pushback for the next poll
            nonlocal n_vals
            if type(entry) is tuple and entry and entry[0] == "g":
                _tag, base, count, out_chunk, cis = entry[:5]
                val_off = entry[5] if len(entry) > 5 else 0
                n = count if cis is None else len(cis)
                budget = max_values - n_vals
                if n > budget:
                    cis_full = list(range(count)) if cis is None else cis
                    head, tail = cis_full[:budget], cis_full[budget:]
                    if head:
                        out.append(
                            {
                                "group": True,
                                "idx_base": base,
                                "count": count,
                                "cis": head,
                                "val_off": val_off,
                                "out_chunk": out_chunk,
                                "status": 1,
                            }
                        )
                        n_vals += len(head)
                    record.output_pushback.append(
                        ("g", base, count, out_chunk, tail, val_off + len(head))
                    )
                    return
                out.append(
                    {
                        "group": True,
                        "idx_base": base,
                        "count": count,
                        "cis": cis,
                        "val_off": val_off,
                        "out_chunk": out_chunk,
                        "status": 1,
                    }
                )
                n_vals += n
            elif isinstance(entry, list):
                for pos, idx in enumerate(entry):
                    if n_vals >= max_values:
                        record.output_pushback.append(entry[pos:])
                        return
                    out.append(self._output_item(record, idx))
                    n_vals += 1
            else:
                out.append(self._output_item(record, entry))
                n_vals += 1

        def take_nowait() -> Any:
            if record.output_pushback:
                return record.output_pushback.popleft()
            return record.output_ready.get_nowait()

        deadline = time.time() + timeout
        while not out:
            if record.output_pushback:
                entry = record.output_pushback.popleft()
            else:
                try:
                    remaining = deadline - time.time()
                    if remaining <= 0:
                        break
                    entry = await asyncio.wait_for(record.output_ready.get(), remaining)
                except asyncio.TimeoutError:
                    break
            expand(entry)
            while n_vals < max_values:
                try:
                    entry = take_nowait()
                except asyncio.QueueEmpty:
                    break
                expand(entry)
        # attach shared output-chunk bytes once per chunk per response
        chunks_seen: set[str] = set()
        for item in out:
            cid = item.get("out_chunk")
            if cid and cid not in chunks_seen:
                chunks_seen.add(cid)
                chunk = self.out_chunks.get(cid)
                if chunk is not None:
                    item["chunk_data"] = chunk["data"]
        if clear_on_success:
            for item in out:
                cid = item.get("out_chunk")
                if item.get("group"):
                    n = item["count"] if item["cis"] is None else len(item["cis"])
                else:
                    n = 1
                    rec = record.inputs.get(item["idx"])
                    if rec is not None:
                        rec.payload = b""  # release memory; result extracted
                if cid:
                    chunk = self.out_chunks.get(cid)
                    if chunk is not None:
                        chunk["refs"] -= n
                        if chunk["refs"] <= 0:
                            self.out_chunks.pop(cid, None)
        return out

    def _output_item(self, record: CallRecord, idx: int) -> dict:
        rec = record.inputs[idx]
        item = {
            "idx": idx,
            "input_id": rec.input_id,
            "status": rec.status,
            "data": rec.output,
            "data_blob": rec.output_blob,
            "format": rec.output_format,
            "exc": rec.exc_repr,
            "retry_count": rec.retry_count,
        }
        if rec.out_chunk:
            item["out_chunk"] = rec.out_chunk
            item["out_ci"] = rec.out_ci
        return item

    async def function_call_cancel(
        self, function_call_id: str, terminate_containers: bool = False
    ) -> None:
        record = self._call(function_call_id)
        record.cancelled = True
        tokens = []
        for rec in record.inputs.values():
            if not rec.final:
                rec.cancelled = True
                tokens.append(rec.token)
                self.finalize_input(
                    rec, GENERIC_STATUS_TERMINATED, None, 0, "input cancelled", rec.retry_count
                )
        for group in record.chunk_groups.values():
            if group.state != "done":
                group.state = "done"
                tokens.append(group.token)
                for ci in range(group.count):
                    if record.inputs.get(group.base_idx + ci) is not None:
                        continue  # already materialized + finalized above
                    rec = record.materialize_chunk_item(group, ci)
                    rec.cancelled = True
                    self.finalize_input(
                        rec, GENERIC_STATUS_TERMINATED, None, 0, "input cancelled", rec.retry_count
                    )
        await self.pool.cancel_inputs(tokens, terminate=terminate_containers)

    async def function_call_info(self, function_call_id: str) -> dict:
        record = self._call(function_call_id)
        return {
            "function_call_id": record.call_id,
            "function_id": record.function_id,
            "kind": record.kind,
            **record.stats(),
        }

    def _call(self, function_call_id: str) -> CallRecord:
        record = self.calls.get(function_call_id)
        if record is None:
            raise NotFoundError(f"Function call {function_call_id} not found")
        return record

    # -- worker callbacks --------------------------------------------------
    def register_out_chunk(self, data: bytes, refs: int) -> str:
        self._out_chunk_seq += 1
        chunk_id = f"oc{self._out_chunk_seq}"
        self.out_chunks[chunk_id] = {"data": data, "refs": refs}
        self._out_chunk_order.append(chunk_id)
        while len(self._out_chunk_order) > 20_000:  # stale-delivery leak bound
            old = self._out_chunk_order.pop(0)
            self.out_chunks.pop(old, None)
        return chunk_id

    def on_worker_output(
        self,
        call_id: str,
        idx: int,
        retry_count: int,
        status: int,
        output: Optional[bytes],
        output_format: int,
        exc_repr: Optional[str],
        output_blob: Optional[str] = None,
        out_chunk: Optional[str] = None,
        out_ci: int = 0,
    ) -> None:
        record = self.calls.get(call_id)
        if record is None:
            return
        rec = record.inputs.get(idx)
        if rec is None or rec.final or rec.retry_count != retry_count:
            return  # stale attempt (dedup parity: parallel_map.py:1416-1431)
        fdef = self.functions.get(record.function_id)
        if rec.cluster is not None and status in (
            GENERIC_STATUS_FAILURE,
            GENERIC_STATUS_INTERNAL_FAILURE,
        ):
            # gang fate-sharing: a solo retry of one rank can never
            # rendezvous with peers that already ran, so a failed gang
            # member fails its whole gang (parity: clustered containers
            # fate-share in the reference's scheduler)
            asyncio.get_running_loop().create_task(
                self._fail_gang_siblings(record, rec, exc_repr)
            )
        elif status == GENERIC_STATUS_FAILURE and fdef is not None:
            policy = fdef.retry_policy
            if rec.retry_count < policy.max_retries and not rec.cancelled:
                rec.retry_count += 1
                delay_s = policy.delay_ms(rec.retry_count) / 1000.0
                if delay_s <= 0:
                    self.pool.enqueue(rec)
                else:
                    self.pool.enqueue_delayed(rec, delay_s)
                return
        elif status == GENERIC_STATUS_INTERNAL_FAILURE:
            rec.internal_failures += 1
            if rec.internal_failures <= MAX_INTERNAL_FAILURE_COUNT and not rec.cancelled:
                self.pool.enqueue(rec, front=True)
                return
        self.finalize_input(
            rec, status, output, output_format, exc_repr, retry_count, output_blob,
            out_chunk, out_ci,
        )

    async def _fail_gang_siblings(self, record: CallRecord, failed: InputRecord, exc_repr: Optional[str]) -> None:
        """Terminate the other ranks of a failed gang member's cluster:
        cancel their in-flight executions and finalize them so the caller
        unblocks (stale ranks would otherwise hang in collectives)."""
        try:
            await self._fail_gang_siblings_inner(record, failed, exc_repr)
        except Exception:
            logging.getLogger("modal_amd.scheduler").warning(
                "gang teardown failed", exc_info=True
            )

    async def _fail_gang_siblings_inner(
        self, record: CallRecord, failed: InputRecord, exc_repr: Optional[str]
    ) -> None:
        cluster_id = failed.cluster["cluster_id"]
        siblings = [
            r
            for r in record.inputs.values()
            if r.cluster is not None
            and r.cluster["cluster_id"] == cluster_id
            and r.idx != failed.idx
            and not r.final
        ]
        tokens = []
        for r in siblings:
            r.cancelled = True
            tokens.append(r.token)
        if tokens:
            await self.pool.cancel_inputs(tokens, terminate=True)
        for r in siblings:
            if not r.final:
                self.finalize_input(
                    r,
                    GENERIC_STATUS_TERMINATED,
                    None,
                    0,
                    f"gang member rank {failed.cluster['rank']} failed: {exc_repr}",
                    r.retry_count,
                )

    def finalize_input(
        self,
        rec: InputRecord,
        status: int,
        output: Optional[bytes],
        output_format: int,
        exc_repr: Optional[str],
        retry_count: int,
        output_blob: Optional[str] = None,
        out_chunk: Optional[str] = None,
        out_ci: int = 0,
    ) -> None:
        record = self.calls.get(rec.call_id)
        if record is None:
            return
        self.metrics_counters["outputs_total"] += 1
        if status != GENERIC_STATUS_SUCCESS:
            self.metrics_counters["failures_total"] += 1
        if record.kind == "unary":
            self._unary_latencies.append(time.time() - record.created_at)
        record.post_output(
            rec.idx, status, output, output_format, exc_repr, retry_count, output_blob,
            out_chunk, out_ci,
        )
        if getattr(record, "durable", False) and rec.final:
            from . import wal

            wal.journal_result(self, record, rec)
            if (
                record.num_inputs_final is not None
                and record.completed >= record.num_inputs_final
            ):
                wal.drop(self, record.call_id)

    def on_generator_data(self, msg: dict) -> None:
        call_id, idx_s, _ = msg["token"].rsplit(":", 2)
        record = self.calls.get(call_id)
        if record is None:
            return
        record.gen_queue(int(idx_s)).put_nowait(
            (msg.get("index", 0), msg.get("data"), msg.get("format", 0), bool(msg.get("done")))
        )

    async def generator_poll(
        self, function_call_id: str, idx: int = 0, timeout: float = OUTPUT_POLL_TIMEOUT
    ) -> list:
        """RPC-friendly poll of a generator's data-out channel."""
        record = self._call(function_call_id)
        q = record.gen_queue(idx)
        out = []
        try:
            item = await asyncio.wait_for(q.get(), timeout)
            out.append(list(item))
            while True:
                out.append(list(q.get_nowait()))
        except (asyncio.TimeoutError, asyncio.QueueEmpty):
            pass
        return out

    def on_worker_log(self, handle: Any, msg: dict) -> None:
        app_id = msg.get("app_id", "")
        entry = {
            "ts": time.time(),
            "task_id": handle.task_id,
            "fd": msg.get("fd", 1),
            "data": msg.get("data", ""),
            "app_id": app_id,
        }
        app = self.apps.get(app_id)
        if app is not None:
            app.logs.append(entry)
            app.log_seq += 1
            app.log_event.set()
            for sub in app.log_subscribers:
                sub.put_nowait(entry)
        else:
            for a in self.apps.values():
                if a.state == "running":
                    for sub in a.log_subscribers:
                        sub.put_nowait(entry)

    def resolve_function_env(self, fdef: FunctionDef) -> dict[str, str]:
        """Merge image env then secret env bundles (secrets win; parity:
        image env + secrets applied to container env)."""
        env: dict[str, str] = {}
        if fdef.image_id:
            state = self.image_service.by_id.get(fdef.image_id)
            if state is not None:
                env.update(state.env)
        for secret_id in fdef.secret_ids:
            try:
                env.update(self.services.secret_env(secret_id))
            except Exception:
                pass
        if fdef.proxy_url:
            env["HTTP_PROXY"] = env["http_proxy"] = fdef.proxy_url
            env["HTTPS_PROXY"] = env["https_proxy"] = fdef.proxy_url
        return env

    def resolve_function_pythonpaths(self, fdef: FunctionDef) -> list[str]:
        if fdef.image_id:
            state = self.image_service.by_id.get(fdef.image_id)
            if state is not None:
                return list(state.python_paths)
        return []

    # -- device mesh (RCCL/xGMI tensor plane) ------------------------------
    async def ensure_mesh(self) -> dict:
        """Form the worker collective plane once: rank assignment + readiness
        (the tensor-transfer substrate of runtime/devicemesh.py)."""
        state = self._extra.setdefault(
            "mesh", {"formed": False, "ranks": {}, "lock": asyncio.Lock(), "ready": {}}
        )
        async with state["lock"]:
            if state["formed"]:
                return state
            members = [w for w in self.pool.workers.values() if w.alive]
            gpu_members = [w for w in members if w.has_gpu]
            if gpu_members and self.pool._gpu_count() > 0:
                members = gpu_members
                backend = "nccl"
            else:
                backend = "gloo"
            if len(members) < 2:
                raise NotFoundError("device mesh needs >= 2 workers")
            members.sort(key=lambda w: w.worker_id)
            import socket as socket_mod

            sock = socket_mod.socket()
            sock.bind(("127.0.0.1", 0))
            port = sock.getsockname()[1]
            sock.close()
            ranks = {w.task_id: i for i, w in enumerate(members)}
            events = {w.task_id: asyncio.Event() for w in members}
            state["ready"] = events
            for w in members:
                await w.conn.send(
                    {
                        "t": "mesh_init",
                        "rank": ranks[w.task_id],
                        "world": len(members),
                        "port": port,
                        "backend": backend,
                    }
                )
            await asyncio.wait_for(
                asyncio.gather(*(e.wait() for e in events.values())), timeout=120
            )
            state["ranks"] = ranks
            state["backend"] = backend
            state["formed"] = True
            return state

    def on_mesh_ready(self, handle: Any, ok: bool) -> None:
        state = self._extra.get("mesh")
        if state:
            event = state["ready"].get(handle.task_id)
            if event is not None and ok:
                event.set()

    def _worker_by_task(self, task_id: str) -> Any:
        for w in self.pool.workers.values():
            if w.task_id == task_id:
                return w
        return None

    async def device_transfer(
        self, owner_task: str, token: str, meta: dict, dest_task: str
    ) -> dict:
        """Coordinate a p2p tensor move: owner sends, dest receives
        (RCCL over xGMI on GPU pools; gloo on CPU pools)."""
        try:
            state = await self.ensure_mesh()
        except Exception:
            return {"fallback": True}
        ranks = state["ranks"]
        if owner_task not in ranks or dest_task not in ranks or owner_task == dest_task:
            return {"fallback": True}
        owner = self._worker_by_task(owner_task)
        dest = self._worker_by_task(dest_task)
        if owner is None or dest is None:
            return {"fallback": True}
        xfer_id = new_id("blob")
        # dest's dev_recv is ordered before this RPC's reply on its socket
        await dest.conn.send(
            {"t": "dev_recv", "xfer_id": xfer_id, "meta": meta, "src_rank": ranks[owner_task]}
        )
        await owner.conn.send(
            {"t": "dev_send", "token": token, "dst_rank": ranks[dest_task], "xfer_id": xfer_id}
        )
        return {"xfer_id": xfer_id, "src_rank": ranks[owner_task]}

    async def tensor_pull_relay(self, owner_task: str, token: str) -> Optional[bytes]:
        """Host-staged fallback for non-mesh consumers."""
        owner = self._worker_by_task(owner_task)
        if owner is None:
            return None
        return await owner.conn.call("tensor_pull", {"token": token}, timeout=120)

    # -- cluster rendezvous ------------------------------------------------
    async def cluster_hello(self, cluster_id: str, rank: int, world_size: int, addr: str = "") -> dict:
        """In-proc TaskClusterHello: collect member addresses, release when full
        (parity: reference _clustered_functions.py:74-86)."""
        key = f"cluster:{cluster_id}"
        state = self._extra.setdefault(key, {"members": {}, "event": asyncio.Event()})
        state["members"][rank] = addr
        if len(state["members"]) >= world_size:
            state["event"].set()
        await state["event"].wait()
        return {
            "cluster_id": cluster_id,
            "rank": rank,
            "world_size": world_size,
            "addrs": [state["members"].get(r, "") for r in range(world_size)],
        }

    async def app_get_logs(
        self,
        app_id: str,
        offset: int = 0,
        timeout: float = 55.0,
        max_entries: int = 1000,
    ) -> dict:
        """Offset-resumable log long-poll (parity: the reference's
        reconnecting deadline'd AppGetLogs streams, _logs_manager.py:24-26).

        ``offset`` is an absolute sequence number; entries older than the
        ring keeps are reported as ``dropped``. Works over the proxy
        transport, so a daemon-attached CLI can tail live."""
        app = self._app(app_id)
        deadline = time.time() + min(timeout, OUTPUT_POLL_TIMEOUT)
        while app.log_seq <= offset:
            remaining = deadline - time.time()
            if remaining <= 0 or app.state == "stopped":
                return {"entries": [], "next_offset": offset, "dropped": 0,
                        "app_state": app.state}
            app.log_event.clear()
            try:
                await asyncio.wait_for(app.log_event.wait(), remaining)
            except asyncio.TimeoutError:
                pass
        oldest = app.log_seq - len(app.logs)
        start = max(offset, oldest)
        dropped = start - offset
        skip = start - oldest
        entries = list(app.logs)[skip : skip + max_entries]
        return {
            "entries": entries,
            "next_offset": start + len(entries),
            "dropped": dropped,
            "app_state": app.state,
        }

    # -- worker GPU snapshots ---------------------------------------------
    async def worker_snapshot(self, worker_id: int) -> dict:
        """Capture a worker's snapshot-visible GPU state (registered
        tensors + tracked raw hipMalloc allocations + RNG) into the CAS.
        Parity: the snapshot half of ContainerCheckpoint
        (reference task_lifecycle_manager.py:195)."""
        w = self.pool.workers.get(int(worker_id))
        if w is None:
            raise NotFoundError(f"worker {worker_id} not connected")
        data = await w.conn.call("snapshot_state", timeout=300)
        blob_id = self.blob_store.put(data)
        return {"snapshot_id": blob_id}

    async def worker_restore(
        self, snapshot_id: str, gpu_index: Optional[int] = None
    ) -> dict:
        """Spawn a FRESH worker that rehydrates a snapshot via the
        restore-state.json contract; on sentinel exit 222 the spawn is
        retried WITHOUT the snapshot (degraded, parity: reference
        gpu_memory_snapshot.py:20-23)."""
        import json as _json

        state_path = os.path.join(self.run_dir, f"restore-{new_id('task')[3:]}.json")
        state = {
            "task_id": new_id("task"),
            "snapshot_id": snapshot_id,
            "snapshot_path": self.blob_store.open_path(snapshot_id),
            "env": {},
            "snapshot_debug": False,
        }
        with open(state_path, "w") as f:
            _json.dump(state, f)
        # spawn_worker stamps MODAL_AMD_WORKER_ID from _next_worker_id, so
        # the restored worker's id is known a priori (autoscaled respawns
        # racing this cannot be confused with it)
        wid = self.pool._next_worker_id
        proc = await self.pool.spawn_worker(
            gpu_index=gpu_index,
            extra_env={"MODAL_AMD_RESTORE_STATE_PATH": state_path},
        )
        deadline = time.time() + 120
        while time.time() < deadline:
            if wid in self.pool.workers:
                return {"worker_id": wid, "degraded": False}
            rc = proc.poll() if proc is not None else None
            if rc is not None:
                # exit-222 contract: retry without the snapshot
                self.pool._pending_spawns = max(0, self.pool._pending_spawns - 1)
                wid2 = self.pool._next_worker_id
                await self.pool.spawn_worker(gpu_index=gpu_index)
                while time.time() < deadline:
                    if wid2 in self.pool.workers:
                        return {"worker_id": wid2, "degraded": True, "exit_code": rc}
                    await asyncio.sleep(0.1)
                break
            await asyncio.sleep(0.1)
        raise InvalidError("restored worker never connected")

    async def named_objects_list(self, kind: str, environment: str = "") -> list:
        """List deployed named objects of one kind (parity: reference
        <Type>.objects.list() managers, e.g. queue.py _QueueManager.list)."""
        env = environment or self.default_environment
        if kind == "volume":
            store = self.volume_service.by_name
        elif kind == "queue":
            store = self.services.queue_names.by_name
        elif kind == "dict":
            store = self.services.dict_names.by_name
        elif kind == "secret":
            store = self.services.secret_names.by_name
        else:
            raise InvalidError(f"Unknown object kind {kind!r}")
        return [
            {"name": name, "object_id": oid}
            for (e, name), oid in sorted(store.items())
            if e == env
        ]

    async def named_object_delete(self, kind: str, name: str, environment: str = "") -> None:
        """Delete a deployed named object (parity: <Type>.objects.delete())."""
        env = environment or self.default_environment
        rows = await self.named_objects_list(kind, env)
        oid = next((r["object_id"] for r in rows if r["name"] == name), None)
        if oid is None:
            raise NotFoundError(f"{kind} '{name}' not found in environment '{env}'")
        if kind == "volume":
            await self.volume_service.delete(oid)
        elif kind == "queue":
            self.services.queue_delete(oid)
        elif kind == "dict":
            self.services.dict_delete(oid)
        elif kind == "secret":
            self.services.secrets.pop(oid, None)
            self.services.secret_names.by_name = {
                k: v for k, v in self.services.secret_names.by_name.items() if v != oid
            }

    async def object_info(self, object_id: str) -> dict:
        return self.services.object_info(object_id)

    async def volume_info(self, volume_id: str) -> dict:
        vol = self.volume_service._get(volume_id)
        return {
            "name": vol.name,
            "files": len(vol.manifest),
            "created_at": getattr(vol, "created_at", None),
        }

    async def secret_update(self, secret_id: str, env: dict) -> None:
        st = self.services.secrets.get(secret_id)
        if st is None:
            raise NotFoundError(f"Secret {secret_id} not found")
        st.env.update({k: str(v) for k, v in env.items()})

    async def app_set_tags(self, app_id: str, tags: dict) -> None:
        app = self._app(app_id)
        if not hasattr(app, "tags"):
            app.tags = {}
        app.tags.update({str(k): str(v) for k, v in tags.items()})

    async def app_get_tags(self, app_id: str) -> dict:
        return dict(getattr(self._app(app_id), "tags", {}) or {})

    # -- blobs -------------------------------------------------------------
    async def blob_put(self, data: bytes) -> dict:
        digest = await asyncio.get_running_loop().run_in_executor(None, self.blob_store.put, data)
        return {"blob_id": digest}

    async def blob_get(self, blob_id: str) -> bytes:
        return await asyncio.get_running_loop().run_in_executor(None, self.blob_store.get, blob_id)

    async def blob_path(self, blob_id: str) -> str:
        return self.blob_store.open_path(blob_id)

    def maybe_blob(self, payload: bytes, limit: int = INLINE_LIMIT) -> dict:
        """Inline small payloads; CAS-store large ones (parity: should_upload,
        reference function_utils.py:586, thresholds blob_utils.py:36-39)."""
        if len(payload) <= limit:
            return {"payload": payload}
        digest = self.blob_store.put(payload)
        return {"payload_blob": digest}

    # -- volumes -----------------------------------------------------------
    async def volume_get_or_create(self, name=None, environment="main", create_if_missing=False, ephemeral=False) -> dict:
        return await self.volume_service.get_or_create(name, environment, create_if_missing, ephemeral)

    async def volume_put_file_blocks(
        self, volume_id, rel_path, block_digests, size, mode=0o644, content_tmp=None
    ) -> dict:
        return await self.volume_service.put_file_blocks(
            volume_id, rel_path, block_digests, size, mode, content_tmp=content_tmp
        )

    async def volume_get_file(self, volume_id, rel_path, offset=0, n_bytes=-1) -> bytes:
        return await self.volume_service.get_file(volume_id, rel_path, offset, n_bytes)

    async def volume_list_files(self, volume_id, rel_path="/", recursive=True) -> list:
        return await self.volume_service.list_files(volume_id, rel_path, recursive)

    async def volume_remove_file(self, volume_id, rel_path, recursive=False) -> None:
        return await self.volume_service.remove_file(volume_id, rel_path, recursive)

    async def volume_copy_files(self, volume_id, src_paths, dst_path) -> None:
        return await self.volume_service.copy_files(volume_id, src_paths, dst_path)

    async def volume_commit(self, volume_id) -> dict:
        return await self.volume_service.commit(volume_id)

    async def volume_reload(self, volume_id) -> None:
        return await self.volume_service.reload(volume_id)

    async def volume_delete(self, volume_id) -> None:
        return await self.volume_service.delete(volume_id)

    async def volume_rename(self, volume_id, new_name, environment="main") -> None:
        return await self.volume_service.rename(volume_id, new_name, environment)

    async def volume_dir(self, volume_id) -> str:
        return self.volume_service.volume_dir(volume_id)

    # -- sandboxes ---------------------------------------------------------
    async def sandbox_create(self, **kwargs: Any) -> dict:
        volume_mounts = kwargs.pop("volume_mounts", None) or {}
        volume_paths = {}
        for path, spec in volume_mounts.items():
            vid = spec["volume_id"] if isinstance(spec, dict) else spec
            vol_dir = self.volume_service.volume_dir(vid)
            if isinstance(spec, dict) and spec.get("sub_path"):
                vol_dir = os.path.join(vol_dir, spec["sub_path"].strip("/"))
                os.makedirs(vol_dir, exist_ok=True)
            volume_paths[path] = vol_dir
        restore_image = kwargs.pop("restore_image_id", None)
        restore_blob = None
        image_fsroot = None
        if restore_image:
            # image ids minted by sandbox_snapshot_fs carry their tar blob
            restore_blob = self._extra.get("snapshot_blobs", {}).get(restore_image)
            # built images contribute their filesystem layer as an extra
            # overlay lower for the sandbox root (isolation.py)
            img = self.image_service.by_id.get(restore_image)
            if img is not None and os.path.isdir(img.fsroot) and os.listdir(img.fsroot):
                image_fsroot = img.fsroot
        return await self.sandbox_service.create(
            volume_paths=volume_paths,
            restore_blob=restore_blob,
            image_fsroot=image_fsroot,
            blob_store=self.blob_store,
            **kwargs,
        )

    async def sandbox_exec(
        self, sandbox_id, cmd, env=None, workdir=None, timeout=None, exec_id=None,
        pty=False, rows=24, cols=80,
    ) -> dict:
        return await self.sandbox_service.exec(
            sandbox_id, cmd, env, workdir, timeout, exec_id, pty=pty, rows=rows, cols=cols
        )

    async def sandbox_resize(self, target_id, rows, cols) -> None:
        await self.sandbox_service.resize(target_id, rows, cols)

    async def sandbox_stdio_read(self, target_id, fd, offset=0, max_bytes=1 << 20, timeout=55.0) -> dict:
        return await self.sandbox_service.stdio_read(target_id, fd, offset, max_bytes, timeout)

    async def sandbox_stdin_write(self, target_id, offset, data, eof=False) -> int:
        return await self.sandbox_service.stdin_write(target_id, offset, data, eof)

    async def sandbox_wait(self, target_id, timeout=None, raise_on_timeout=True) -> dict:
        return await self.sandbox_service.wait(target_id, timeout, raise_on_timeout)

    async def sandbox_poll(self, target_id) -> dict:
        return await self.sandbox_service.poll(target_id)

    async def sandbox_terminate(self, sandbox_id) -> None:
        return await self.sandbox_service.terminate(sandbox_id)

    async def sandbox_list(self, app_id=None, tags=None) -> list:
        return await self.sandbox_service.list(app_id, tags)

    async def sandbox_set_tags(self, sandbox_id, tags) -> None:
        return await self.sandbox_service.set_tags(sandbox_id, tags)

    async def sandbox_from_name(self, name, environment="main") -> dict:
        sid = await self.sandbox_service.from_name(name, environment)
        return {"sandbox_id": sid}

    async def sandbox_snapshot_fs(self, sandbox_id) -> dict:
        resp = await self.sandbox_service.snapshot_fs(sandbox_id, self.blob_store)
        self._extra.setdefault("snapshot_blobs", {})[resp["image_id"]] = resp["blob_id"]
        return resp

    async def sandbox_snapshot_dir(self, sandbox_id: str, path: str) -> dict:
        """Snapshot one sandbox directory into a new Image whose fsroot holds
        the tree (parity: reference Sandbox.snapshot_directory,
        sandbox.py:1643)."""
        import shutil

        sb = self.sandbox_service._get(sandbox_id)
        src = os.path.normpath(os.path.join(sb.workdir, path.lstrip("/")))
        if not os.path.isdir(src):
            raise InvalidError(f"{path!r} is not a directory in sandbox {sandbox_id}")
        resp = await self.image_service.get_or_create(
            [{"kind": "base", "name": "snapshot", "sandbox": sandbox_id, "path": path,
              "nonce": os.urandom(8).hex()}],
            build=False,
        )
        state = self.image_service.by_id[resp["image_id"]]
        dst = os.path.join(state.fsroot, path.lstrip("/"))
        await asyncio.get_running_loop().run_in_executor(
            None, lambda: shutil.copytree(src, dst, symlinks=True)
        )
        state.built = True
        state.build_log = f"snapshot of {sandbox_id}:{path}"
        return {"image_id": state.image_id}

    async def sandbox_mount_image(self, sandbox_id: str, path: str, image_id: str) -> None:
        """Expose an image's filesystem layer at `path` inside the sandbox
        workdir (parity: Sandbox.mount_image, reference sandbox.py:1548)."""
        sb = self.sandbox_service._get(sandbox_id)
        state = self.image_service.by_id.get(image_id)
        if state is None:
            raise NotFoundError(f"Image {image_id} not found")
        link = os.path.normpath(os.path.join(sb.workdir, path.lstrip("/")))
        src = os.path.join(state.fsroot, path.lstrip("/"))
        if not os.path.isdir(src):
            src = state.fsroot
        os.makedirs(os.path.dirname(link), exist_ok=True)
        if os.path.islink(link):
            os.unlink(link)
        elif os.path.exists(link):
            raise InvalidError(f"{path!r} already exists in sandbox {sandbox_id}")
        os.symlink(src, link)

    async def sandbox_unmount_image(self, sandbox_id: str, path: str) -> None:
        sb = self.sandbox_service._get(sandbox_id)
        link = os.path.normpath(os.path.join(sb.workdir, path.lstrip("/")))
        if os.path.islink(link):
            os.unlink(link)

    async def sandbox_connect_token(
        self, sandbox_id: str, port: int = 8080, user_metadata: Any = None
    ) -> dict:
        """Mint a bearer token for HTTP access to a sandbox-hosted server
        (parity: SandboxCreateConnectToken, reference sandbox.py:1799).
        Locally the "proxy" is direct: the URL targets 127.0.0.1:<port>."""
        if not isinstance(port, int) or not (1 <= port <= 65535):
            raise InvalidError("port must be between 1 and 65535")
        self.sandbox_service._get(sandbox_id)  # raises if unknown
        import secrets as _secrets

        token = _secrets.token_urlsafe(24)
        self._extra.setdefault("sandbox_connect_tokens", {})[token] = {
            "sandbox_id": sandbox_id,
            "port": port,
            "user_metadata": user_metadata,
            "created_at": time.time(),
        }
        return {"url": f"http://127.0.0.1:{port}", "token": token}

    async def sandbox_fs_op(self, sandbox_id: str, op: str, path: str = "", **kwargs: Any) -> Any:
        """Typed remote-FS operations inside a sandbox workdir (parity:
        reference sandbox filesystem API, sandbox_fs.py:68, file_io.py:135)."""
        sb = self.sandbox_service._get(sandbox_id)
        base = sb.workdir
        full = os.path.normpath(os.path.join(base, path.lstrip("/"))) if path else base
        loop = asyncio.get_running_loop()
        if op == "read":
            def _read() -> bytes:
                with open(full, "rb") as f:
                    f.seek(kwargs.get("offset", 0))
                    return f.read(kwargs.get("n", -1))

            return await loop.run_in_executor(None, _read)
        if op == "write":
            def _write() -> int:
                mode = "ab" if kwargs.get("append") else ("r+b" if kwargs.get("offset") else "wb")
                if kwargs.get("offset") and not os.path.exists(full):
                    open(full, "wb").close()
                with open(full, mode) as f:
                    if kwargs.get("offset"):
                        f.seek(kwargs["offset"])
                    return f.write(kwargs.get("data", b""))

            return await loop.run_in_executor(None, _write)
        if op == "ls":
            return sorted(os.listdir(full))
        if op == "stat":
            st = os.stat(full)
            return {"size": st.st_size, "mtime": st.st_mtime, "is_dir": os.path.isdir(full)}
        if op == "mkdir":
            os.makedirs(full, exist_ok=kwargs.get("parents", False) or kwargs.get("exist_ok", False))
            return None
        if op == "rm":
            if os.path.isdir(full) and not os.path.islink(full):
                if kwargs.get("recursive"):
                    import shutil

                    shutil.rmtree(full)
                else:
                    os.rmdir(full)
            else:
                os.unlink(full)
            return None
        if op == "exists":
            return os.path.exists(full)
        raise InvalidError(f"Unknown fs op {op!r}")

    # -- proxies -----------------------------------------------------------
    async def proxy_get_or_create(self, name: str, environment: str = "") -> dict:
        """Start (once) the shared local forward proxy and register `name`
        on it (parity: Proxy.from_name, reference proxy.py:57 — locally all
        named proxies share one 127.0.0.1 egress point)."""
        from .localproxy import LocalForwardProxy

        proxy = self._extra.get("local_proxy")
        if proxy is None:
            proxy = LocalForwardProxy()
            await proxy.start()
            self._extra["local_proxy"] = proxy
        env = environment or self.default_environment
        ids = self._extra.setdefault("proxy_ids", {})
        pid = ids.get((env, name))
        if pid is None:
            pid = new_id("tunnel")
            ids[(env, name)] = pid
        return {"proxy_id": pid, "url": proxy.url, "port": proxy.port}

    async def proxy_stats(self) -> dict:
        proxy = self._extra.get("local_proxy")
        if proxy is None:
            return {"running": False}
        return {
            "running": True,
            "port": proxy.port,
            "connections": proxy.connections,
            "bytes_relayed": proxy.bytes_relayed,
        }

    # -- images ------------------------------------------------------------
    async def image_get_or_create(self, recipe: list) -> dict:
        return await self.image_service.get_or_create(recipe)

    async def image_info(self, image_id: str) -> dict:
        return await self.image_service.info(image_id)

    async def image_publish(self, image_id: str, name: str, environment: str = "") -> dict:
        """Publish a built image under a workspace name; ':latest' implied
        (parity: reference Image.publish, _image.py:3010)."""
        if self.image_service.by_id.get(image_id) is None:
            raise NotFoundError(f"Image {image_id} not found")
        env = environment or self.default_environment
        if ":" not in name:
            name = f"{name}:latest"
        self._extra.setdefault("image_names", {})[(env, name)] = image_id
        return {"name": name}

    async def image_from_name(self, name: str, environment: str = "") -> dict:
        env = environment or self.default_environment
        if ":" not in name:
            name = f"{name}:latest"
        image_id = self._extra.get("image_names", {}).get((env, name))
        if image_id is None:
            raise NotFoundError(f"Image '{name}' not found in environment '{env}'")
        return {"image_id": image_id}

    # -- mounts ------------------------------------------------------------
    async def mount_get_or_create(self, manifest: list) -> dict:
        """manifest: [[remote_path, blob_id, mode], ...] — content-addressed
        dedup like the reference's MountGetOrCreate (api.proto:4838-4840);
        files materialize as hard links out of the CAS."""
        import hashlib
        import json as _json

        key = hashlib.sha256(
            _json.dumps(sorted(manifest), sort_keys=True).encode()
        ).hexdigest()
        existing = self._mounts_by_hash.get(key)
        if existing:
            return {"mount_id": existing, "dir": self.mounts[existing]}
        mount_id = new_id("mount")
        root = os.path.join(self.run_dir, "mounts", mount_id)
        for remote_path, blob_id, _mode in manifest:
            dest = os.path.join(root, remote_path.lstrip("/"))
            self.blob_store.materialize(blob_id, dest)
        self.mounts[mount_id] = root
        self._mounts_by_hash[key] = mount_id
        return {"mount_id": mount_id, "dir": root}

    # -- service passthrough (queues/dicts/secrets) -------------------------
    def __getattr__(self, name: str) -> Any:
        # delegate queue_/dict_/secret_ methods to Services, wrapped async
        if name.startswith(("queue_", "dict_", "secret_")):
            target = getattr(self.services, name, None)
            if target is not None:
                if asyncio.iscoroutinefunction(target):
                    return target

                async def _async_wrap(*args: Any, **kwargs: Any) -> Any:
                    return target(*args, **kwargs)

                return _async_wrap
        raise AttributeError(name)
