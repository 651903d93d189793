"""api.proto gRPC plane over the in-process scheduler.

A ``grpc.aio`` server on ``<run_dir>/grpc.sock`` speaking the ModalClient
service (modal_proto/api.proto:4680) for the App*/Function*/Queue*/Dict*/
Secret*/Blob* groups, adapting protos to the scheduler's native surface.
The msgpack socket remains the fast path; this plane exists so
reference-style clients, mock-servicer tests, and wire-format checks work
against the real scheduler (round-1 review, Missing #1).

Payload translation: reference FunctionInput.args is a pickled
``(args, kwargs)`` 2-tuple (reference _serialization.py); the native worker
accepts that form directly (runtime/worker.py deserialize_payload handles
both it and the native ``("P", ...)`` envelope).
"""

from __future__ import annotations

import asyncio
import os
import pickle
from typing import Any, Optional

from .compiler import load


class GrpcBridge:
    def __init__(self, scheduler: Any):
        self.scheduler = scheduler
        self.api, self.router_pb = load()
        self._server: Any = None
        self.socket_path = os.path.join(scheduler.run_dir, "grpc.sock")
        self._blob_dir = os.path.join(scheduler.run_dir, "grpcblobs")
        self._out_chunk_cache: dict[str, list] = {}

    # -- lifecycle ---------------------------------------------------------

    async def start(self) -> str:
        import grpc

        os.makedirs(self._blob_dir, exist_ok=True)
        server = grpc.aio.server()
        api = self.api
        pool = api.DESCRIPTOR.pool
        from google.protobuf import message_factory

        def make_handlers(svc: Any) -> dict:
            handlers = {}
            for method in svc.method:
                impl = getattr(self, method.name, None)
                if impl is None:
                    continue
                req_cls = message_factory.GetMessageClass(
                    pool.FindMessageTypeByName(method.input_type.lstrip("."))
                )
                if method.server_streaming:
                    handlers[method.name] = grpc.unary_stream_rpc_method_handler(
                        impl,
                        request_deserializer=req_cls.FromString,
                        response_serializer=lambda msg: msg.SerializeToString(),
                    )
                else:
                    handlers[method.name] = grpc.unary_unary_rpc_method_handler(
                        impl,
                        request_deserializer=req_cls.FromString,
                        response_serializer=lambda msg: msg.SerializeToString(),
                    )
            return handlers

        server.add_generic_rpc_handlers((
            grpc.method_handlers_generic_handler(
                "modal.client.ModalClient", make_handlers(api.SERVICES["ModalClient"])
            ),
            grpc.method_handlers_generic_handler(
                "modal.task_command_router.TaskCommandRouter",
                make_handlers(self.router_pb.SERVICES["TaskCommandRouter"]),
            ),
        ))
        server.add_insecure_port(f"unix:{self.socket_path}")
        await server.start()
        self._server = server
        return self.socket_path

    async def stop(self) -> None:
        if self._server is not None:
            await self._server.stop(0.5)
            self._server = None

    # -- App ---------------------------------------------------------------

    async def AppCreate(self, request: Any, context: Any) -> Any:
        resp = await self.scheduler.app_create(
            description=request.description,
            ephemeral=True,
            environment=request.environment_name,
        )
        return self.api.AppCreateResponse(app_id=resp["app_id"])

    async def AppClientDisconnect(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.app_client_disconnect(request.app_id)
        return empty_pb2.Empty()

    async def AppGetOrCreate(self, request: Any, context: Any) -> Any:
        try:
            resp = await self.scheduler.app_lookup(
                request.app_name, request.environment_name or "main"
            )
            return self.api.AppGetOrCreateResponse(app_id=resp["app_id"])
        except Exception:
            pass
        resp = await self.scheduler.app_create(
            description=request.app_name, ephemeral=False,
            environment=request.environment_name or "main",
        )
        await self.scheduler.app_publish(resp["app_id"], request.app_name)
        return self.api.AppGetOrCreateResponse(app_id=resp["app_id"])

    async def AppDeploy(self, request: Any, context: Any) -> Any:
        await self.scheduler.app_publish(request.app_id, request.name)
        return self.api.AppDeployResponse(url=f"local://{request.name}")

    async def AppLookup(self, request: Any, context: Any) -> Any:
        try:
            resp = await self.scheduler.app_lookup(
                request.app_name, request.environment_name or "main"
            )
        except Exception as exc:
            await self._abort_not_found(context, exc)
            raise
        return self.api.AppLookupResponse(app_id=resp["app_id"])

    async def FunctionGet(self, request: Any, context: Any) -> Any:
        try:
            resp = await self.scheduler.function_lookup(
                request.app_name, request.object_tag, request.environment_name or "main"
            )
        except Exception as exc:
            await self._abort_not_found(context, exc)
            raise
        out = self.api.FunctionGetResponse(function_id=resp["function_id"])
        meta = resp.get("metadata") or {}
        out.handle_metadata.function_name = meta.get("function_name", "")
        return out

    async def FunctionGetCurrentStats(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.function_get_current_stats(request.function_id)
        return self.api.FunctionStats(
            backlog=native.get("backlog", 0),
            num_total_tasks=native.get("num_total_tasks", 0),
        )

    async def DictContents(self, request: Any, context: Any) -> Any:
        items = await self.scheduler.dict_items(request.dict_id)
        for key, value in items:
            entry = self.api.DictEntry()
            entry.key = bytes(key)
            entry.value = bytes(value)
            yield entry

    async def AppStop(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.app_stop(request.app_id)
        return empty_pb2.Empty()

    # -- Function ------------------------------------------------------------

    async def FunctionCreate(self, request: Any, context: Any) -> Any:
        fn = request.function
        is_gen = fn.function_type == self.api.Function.FunctionType.FUNCTION_TYPE_GENERATOR
        options = {
            "definition_kind": "serialized",
            "is_generator": is_gen,
            "timeout": fn.timeout_secs or None,
            "max_concurrent_inputs": max(
                getattr(fn, "max_concurrent_inputs", 0) or 0, 1
            ),
        }
        if fn.resources.gpu_config.count:
            options["needs_gpu"] = True
            options["gpu_count"] = fn.resources.gpu_config.count
        if fn.retry_policy.retries:
            options["retries"] = {
                "max_retries": fn.retry_policy.retries,
                "initial_delay": fn.retry_policy.initial_delay_ms / 1000.0,
                "backoff_coefficient": fn.retry_policy.backoff_coefficient or 1.0,
            }
        resp = await self.scheduler.function_create(
            app_id=request.app_id,
            name=fn.function_name or fn.module_name or "grpc-function",
            definition=bytes(fn.function_serialized),
            options=options,
        )
        out = self.api.FunctionCreateResponse(function_id=resp["function_id"])
        out.handle_metadata.function_name = fn.function_name
        return out

    def _put_item_to_native(self, item: Any) -> dict:
        native: dict = {}
        which = item.input.WhichOneof("args_oneof")
        if which == "args_blob_id":
            native["payload_blob"] = item.input.args_blob_id
            native["payload"] = b""
        else:
            native["payload"] = bytes(item.input.args)
        if item.input.HasField("method_name"):
            native["method"] = item.input.method_name
        return native

    async def FunctionMap(self, request: Any, context: Any) -> Any:
        kind_map = {
            self.api.FUNCTION_CALL_TYPE_UNARY: "unary",
            self.api.FUNCTION_CALL_TYPE_MAP: "map",
        }
        kind = kind_map.get(request.function_call_type, "map")
        if request.from_spawn_map:
            kind = "spawn_map"
        items = [self._put_item_to_native(i) for i in request.pipelined_inputs]
        try:
            resp = await self.scheduler.function_map(
                function_id=request.function_id,
                kind=kind,
                pipelined_inputs=items or None,
                return_exceptions=request.return_exceptions,
            )
        except Exception as exc:
            await self._abort_not_found(context, exc)
            raise
        out = self.api.FunctionMapResponse(
            function_call_id=resp["function_call_id"],
            sync_client_retries_enabled=bool(resp.get("sync_client_retries_enabled")),
            max_inputs_outstanding=resp.get("max_inputs_outstanding") or 1000,
            function_call_jwt=resp["function_call_id"],  # local: the id is the token
        )
        retries = resp.get("retry_policy") or {}
        out.retry_policy.retries = int(retries.get("max_retries", 0))
        out.retry_policy.initial_delay_ms = int(
            float(retries.get("initial_delay", 1.0)) * 1000
        )
        out.retry_policy.backoff_coefficient = float(
            retries.get("backoff_coefficient", 1.0)
        )
        for idx, item in enumerate(items):
            pi = out.pipelined_inputs.add()
            pi.idx = idx
            pi.input_id = f"in-{resp['function_call_id'][3:]}-{idx}"
            pi.input_jwt = pi.input_id
        return out

    async def FunctionPutInputs(self, request: Any, context: Any) -> Any:
        items = [self._put_item_to_native(i) for i in request.inputs]
        try:
            placed = await self.scheduler.function_put_inputs(
                function_call_id=request.function_call_id, items=items
            )
        except Exception as exc:
            await self._abort_not_found(context, exc)
            raise
        out = self.api.FunctionPutInputsResponse()
        for rec, req_item in zip(placed, request.inputs):
            ri = out.inputs.add()
            ri.idx = req_item.idx
            ri.input_id = rec["input_id"]
            ri.input_jwt = rec["input_id"]
        # the reference marks the end of inputs with final_input on the last
        # item (function_utils.py should_upload / FunctionInput.final_input)
        if any(i.input.final_input for i in request.inputs):
            await self.scheduler.function_finish_inputs(
                function_call_id=request.function_call_id
            )
        return out

    async def FunctionGetOutputs(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.function_get_outputs(
            function_call_id=request.function_call_id,
            max_values=request.max_values or 256,
            timeout=min(request.timeout or 55.0, 55.0),
            clear_on_success=request.clear_on_success,
        )
        out = self.api.FunctionGetOutputsResponse()
        record = self.scheduler.calls.get(request.function_call_id)
        call_suffix = request.function_call_id[3:]
        for item in native:
            if item.get("group"):
                values = pickle.loads(item["chunk_data"])
                cis = item["cis"]
                voff = item.get("val_off", 0)
                base = item["idx_base"]
                pairs = (
                    enumerate(values) if cis is None else zip(cis, values[voff:])
                )
                for ci, value in pairs:
                    o = out.outputs.add()
                    o.idx = base + ci
                    o.input_id = f"in-{call_suffix}-{base + ci}"
                    o.result.status = (
                        self.api.GenericResult.GenericStatus.GENERIC_STATUS_SUCCESS
                    )
                    o.result.data = pickle.dumps(value, 4)
                    o.data_format = self.api.DATA_FORMAT_PICKLE
                    out.idxs.append(o.idx)
                continue
            o = out.outputs.add()
            o.idx = item.get("idx", 0)
            o.input_id = f"in-{call_suffix}-{o.idx}"
            o.retry_count = item.get("retry_count", 0)
            o.result.status = item.get("status", 0)
            data = item.get("data")
            cid = item.get("out_chunk")
            if cid:
                # shared worker output chunk: bytes attach once per chunk per
                # response; cache the decoded list for that chunk's siblings
                values = self._out_chunk_cache.get(cid)
                if values is None and item.get("chunk_data") is not None:
                    values = pickle.loads(item["chunk_data"])
                    self._out_chunk_cache[cid] = values
                    while len(self._out_chunk_cache) > 64:
                        self._out_chunk_cache.pop(next(iter(self._out_chunk_cache)))
                if values is not None:
                    data = pickle.dumps(values[item.get("out_ci", 0)], 4)
            if data is not None:
                o.result.data = data
            if item.get("data_blob"):
                o.result.data_blob_id = item["data_blob"]
            if item.get("exc"):
                o.result.exception = item["exc"]
            o.data_format = self.api.DATA_FORMAT_PICKLE
            out.idxs.append(o.idx)
        if record is not None and record.num_inputs_final is not None:
            out.num_unfinished_inputs = max(
                0, record.num_inputs_final - record.completed
            )
        return out

    async def FunctionCallGetDataOut(self, request: Any, context: Any) -> Any:
        """Generator data stream (parity: FunctionCallGetDataOut,
        reference function_utils.py:437-495): DataChunks in index order,
        ending after the GENERATOR_DONE sentinel."""
        call_id = request.function_call_id or request.attempt_token
        next_index = int(request.last_index)
        buffered: dict[int, tuple] = {}
        done_index: Any = None
        while True:
            entries = await self.scheduler.generator_poll(
                function_call_id=call_id, idx=0, timeout=5.0
            )
            if not entries:
                info = await self.scheduler.function_call_info(function_call_id=call_id)
                if info.get("completed", 0) >= info.get("total", 0) and info.get("final"):
                    return  # call finished without (more) generator data
                continue
            for index, data, fmt, is_done in entries:
                buffered[index] = (data, fmt, is_done)
            while next_index in buffered:
                data, fmt, is_done = buffered.pop(next_index)
                chunk = self.api.DataChunk(index=next_index)
                if is_done and data is None:
                    # terminal sentinel (parity: GeneratorDone chunk)
                    from .._serialization import (
                        DataFormat, GeneratorDone, serialize_data_format,
                    )

                    chunk.data_format = self.api.DATA_FORMAT_GENERATOR_DONE
                    chunk.data = serialize_data_format(
                        GeneratorDone(items_total=next_index), DataFormat.GENERATOR_DONE
                    )
                else:
                    chunk.data_format = fmt or self.api.DATA_FORMAT_PICKLE
                    if data is not None:
                        chunk.data = data
                yield chunk
                next_index += 1
                if is_done:
                    return

    async def FunctionCallCancel(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.function_call_cancel(
            function_call_id=request.function_call_id,
            terminate_containers=request.terminate_containers,
        )
        return empty_pb2.Empty()

    async def ClientHello(self, request: Any, context: Any) -> Any:
        """The reference client's first RPC on connect (client.py:209)."""
        return self.api.ClientHelloResponse(image_builder_version="local")

    async def AppList(self, request: Any, context: Any) -> Any:
        rows = await self.scheduler.app_list(request.environment_name or "main")
        out = self.api.AppListResponse()
        state_map = {
            "running": self.api.APP_STATE_EPHEMERAL,
            "deployed": self.api.APP_STATE_DEPLOYED,
            "stopped": self.api.APP_STATE_STOPPED,
        }
        for row in rows:
            item = out.apps.add()
            item.app_id = row["app_id"]
            item.description = row.get("description") or ""
            item.state = state_map.get(row.get("state"), 0)
        return out

    # -- input plane (Attempt trio) -------------------------------------------
    # parity: _InputPlaneInvocation (reference _functions.py:396-549) —
    # locally the "input plane" IS the scheduler, and the attempt token is
    # the call id (SURVEY §2 row 11's stated plan).

    async def AttemptStart(self, request: Any, context: Any) -> Any:
        item = self._put_item_to_native(request.input)
        try:
            resp = await self.scheduler.function_map(
                function_id=request.function_id,
                kind="unary",
                pipelined_inputs=[item],
            )
        except Exception as exc:
            await self._abort_not_found(context, exc)
            raise
        await self.scheduler.function_finish_inputs(
            function_call_id=resp["function_call_id"]
        )
        out = self.api.AttemptStartResponse(attempt_token=resp["function_call_id"])
        retries = resp.get("retry_policy") or {}
        out.retry_policy.retries = int(retries.get("max_retries", 0))
        return out

    async def AttemptAwait(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.function_get_outputs(
            function_call_id=request.attempt_token,
            max_values=1,
            timeout=min(request.timeout_secs or 55.0, 55.0),
            clear_on_success=False,
        )
        out = self.api.AttemptAwaitResponse()
        if native:
            item = native[0]
            o = out.output
            o.idx = item.get("idx", 0)
            o.result.status = item.get("status", 0)
            data = item.get("data")
            cid = item.get("out_chunk")
            if cid and item.get("chunk_data") is not None:
                data = pickle.dumps(
                    pickle.loads(item["chunk_data"])[item.get("out_ci", 0)], 4
                )
            if data is not None:
                o.result.data = data
            if item.get("exc"):
                o.result.exception = item["exc"]
            o.data_format = self.api.DATA_FORMAT_PICKLE
        return out

    async def AttemptRetry(self, request: Any, context: Any) -> Any:
        # a fresh attempt for the same logical input (new local call)
        resp = await self.AttemptStart(
            self.api.AttemptStartRequest(
                function_id=request.function_id, input=request.input
            ),
            context,
        )
        return self.api.AttemptRetryResponse(attempt_token=resp.attempt_token)

    # -- Queue ---------------------------------------------------------------

    async def QueueGetOrCreate(self, request: Any, context: Any) -> Any:
        qid = await self.scheduler.queue_get_or_create(
            name=request.deployment_name or None,
            environment=request.environment_name or "main",
            create_if_missing=True,
            ephemeral=not request.deployment_name,
        )
        return self.api.QueueGetOrCreateResponse(queue_id=qid)

    async def QueuePut(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.queue_put(
            queue_id=request.queue_id,
            values=[bytes(v) for v in request.values],
            partition=bytes(request.partition_key) or None,
            block=False,
            deadline=None,
        )
        return empty_pb2.Empty()

    async def QueueGet(self, request: Any, context: Any) -> Any:
        import time as _time

        deadline = _time.time() + request.timeout if request.timeout else None
        values = await self.scheduler.queue_get(
            queue_id=request.queue_id,
            partition=bytes(request.partition_key) or None,
            n_values=request.n_values or 1,
            block=bool(request.timeout),
            deadline=deadline,
        )
        return self.api.QueueGetResponse(values=values)

    async def QueueLen(self, request: Any, context: Any) -> Any:
        n = self.scheduler.services.queue_len(
            request.queue_id, bytes(request.partition_key) or None, request.total
        )
        return self.api.QueueLenResponse(len=n)

    # -- Dict ----------------------------------------------------------------

    async def DictGetOrCreate(self, request: Any, context: Any) -> Any:
        did = await self.scheduler.dict_get_or_create(
            name=request.deployment_name or None,
            environment=request.environment_name or "main",
            create_if_missing=True,
            ephemeral=not request.deployment_name,
        )
        return self.api.DictGetOrCreateResponse(dict_id=did)

    async def DictUpdate(self, request: Any, context: Any) -> Any:
        updates = {bytes(e.key): bytes(e.value) for e in request.updates}
        created = await self.scheduler.dict_update(
            dict_id=request.dict_id,
            updates=updates,
            if_not_exists=request.if_not_exists,
        )
        return self.api.DictUpdateResponse(created=bool(created))

    async def DictGet(self, request: Any, context: Any) -> Any:
        value = await self.scheduler.dict_get(request.dict_id, bytes(request.key))
        if value is None:
            return self.api.DictGetResponse(found=False)
        return self.api.DictGetResponse(found=True, value=value)

    async def QueueClear(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.queue_clear(
            request.queue_id, bytes(request.partition_key) or None, request.all_partitions
        )
        return empty_pb2.Empty()

    async def QueueDelete(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.queue_delete(request.queue_id)
        return empty_pb2.Empty()

    async def DictLen(self, request: Any, context: Any) -> Any:
        n = await self.scheduler.dict_len(request.dict_id)
        return self.api.DictLenResponse(len=n)

    async def DictClear(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.dict_clear(request.dict_id)
        return empty_pb2.Empty()

    async def DictDelete(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.dict_delete(request.dict_id)
        return empty_pb2.Empty()

    async def VolumeRemoveFile(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.volume_remove_file(
            request.volume_id, request.path, recursive=request.recursive
        )
        return empty_pb2.Empty()

    async def VolumeDelete(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.volume_delete(request.volume_id)
        return empty_pb2.Empty()

    async def SandboxGetLogs(self, request: Any, context: Any) -> Any:
        """Sandbox stdout/stderr as TaskLogsBatch stream (offset tracked in
        last_entry_id for reconnect parity)."""
        fd = 2 if request.file_descriptor == 2 else 1
        offset = int(request.last_entry_id) if request.last_entry_id else 0
        while True:
            resp = await self.scheduler.sandbox_stdio_read(
                target_id=request.sandbox_id, fd=fd, offset=offset,
                timeout=min(request.timeout or 5.0, 55.0),
            )
            if resp["data"]:
                batch = self.api.TaskLogsBatch(entry_id=str(resp["next_offset"]))
                item = batch.items.add()
                item.data = resp["data"].decode("utf-8", errors="replace")
                offset = resp["next_offset"]
                yield batch
            if resp["eof"]:
                out = self.api.TaskLogsBatch(eof=True, entry_id=str(offset))
                yield out
                return
            if not resp["data"]:
                return  # poll window elapsed: client reconnects at entry_id

    # -- Secret --------------------------------------------------------------

    async def SecretGetOrCreate(self, request: Any, context: Any) -> Any:
        sid = await self.scheduler.secret_get_or_create(
            name=request.deployment_name or None,
            env=dict(request.env_dict),
            environment=request.environment_name or "main",
        )
        return self.api.SecretGetOrCreateResponse(secret_id=sid)

    # -- Blob ----------------------------------------------------------------

    async def BlobCreate(self, request: Any, context: Any) -> Any:
        from ..utils.ids import new_id

        blob_id = new_id("blob")
        path = os.path.join(self._blob_dir, blob_id)
        return self.api.BlobCreateResponse(
            blob_id=blob_id, upload_url=f"file://{path}"
        )

    async def BlobGet(self, request: Any, context: Any) -> Any:
        path = os.path.join(self._blob_dir, request.blob_id)
        if not os.path.exists(path):
            # fall through to the CAS (native blob ids are digests)
            try:
                path = await self.scheduler.blob_path(request.blob_id)
            except Exception:
                import grpc

                await context.abort(grpc.StatusCode.NOT_FOUND, "blob not found")
        return self.api.BlobGetResponse(download_url=f"file://{path}")

    # -- Sandbox -------------------------------------------------------------

    async def SandboxCreate(self, request: Any, context: Any) -> Any:
        d = request.definition
        res = d.resources
        resp = await self.scheduler.sandbox_create(
            entrypoint_args=list(d.entrypoint_args),
            workdir=d.workdir if d.HasField("workdir") else None,
            timeout=d.timeout_secs or None,
            app_id=request.app_id,
            cpu=(res.milli_cpu / 1000.0) if res.milli_cpu else None,
            memory=res.memory_mb or None,
            gpu=0 if res.gpu_config.count else None,
        )
        out = self.api.SandboxCreateResponse(sandbox_id=resp["sandbox_id"])
        return out

    async def SandboxWait(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.sandbox_wait(
            request.sandbox_id,
            timeout=request.timeout or None,
            raise_on_timeout=False,
        )
        out = self.api.SandboxWaitResponse()
        rc = native.get("returncode")
        if rc is not None:
            out.result.status = (
                self.api.GenericResult.GenericStatus.GENERIC_STATUS_SUCCESS
                if rc == 0
                else self.api.GenericResult.GenericStatus.GENERIC_STATUS_FAILURE
            )
            out.result.exitcode = rc
        return out

    async def SandboxTerminate(self, request: Any, context: Any) -> Any:
        await self.scheduler.sandbox_terminate(request.sandbox_id)
        return self.api.SandboxTerminateResponse()

    async def SandboxGetTaskId(self, request: Any, context: Any) -> Any:
        sb = self.scheduler.sandbox_service.sandboxes.get(request.sandbox_id)
        task_id = sb.task_id if sb is not None else ""
        return self.api.SandboxGetTaskIdResponse(task_id=task_id)

    async def SandboxList(self, request: Any, context: Any) -> Any:
        rows = await self.scheduler.sandbox_list(app_id=request.app_id or None)
        out = self.api.SandboxListResponse()
        for row in rows:
            info = out.sandboxes.add()
            info.id = row.get("sandbox_id", "")
            info.created_at = row.get("created_at", 0.0)
        return out

    # -- Volume --------------------------------------------------------------

    async def VolumeGetOrCreate(self, request: Any, context: Any) -> Any:
        resp = await self.scheduler.volume_get_or_create(
            name=request.deployment_name or None,
            environment=request.environment_name or "main",
            create_if_missing=True,
            ephemeral=not request.deployment_name,
        )
        return self.api.VolumeGetOrCreateResponse(volume_id=resp["volume_id"])

    async def VolumeListFiles(self, request: Any, context: Any) -> Any:
        entries = await self.scheduler.volume_list_files(
            request.volume_id, request.path or "/", recursive=request.recursive
        )
        resp = self.api.VolumeListFilesResponse()
        type_map = {"file": self.api.FileEntry.FileType.FILE,
                    "dir": self.api.FileEntry.FileType.DIRECTORY}
        for e in entries:
            fe = resp.entries.add()
            fe.path = e["path"]
            fe.type = type_map.get(e.get("type", "file"), self.api.FileEntry.FileType.FILE)
            fe.size = int(e.get("size", 0))
            fe.mtime = int(e.get("mtime", 0))
        yield resp  # server-streaming: one batch

    async def VolumeCommit(self, request: Any, context: Any) -> Any:
        await self.scheduler.volume_commit(request.volume_id)
        return self.api.VolumeCommitResponse()

    async def VolumeReload(self, request: Any, context: Any) -> Any:
        from google.protobuf import empty_pb2

        await self.scheduler.volume_reload(request.volume_id)
        return empty_pb2.Empty()

    # -- Image ---------------------------------------------------------------

    async def ImageGetOrCreate(self, request: Any, context: Any) -> Any:
        # translate the reference's dockerfile-command image definition into
        # the local layer recipe (scheduler/images.py)
        recipe: list = [{"kind": "base", "name": "local"}]
        for cmd in request.image.dockerfile_commands:
            line = cmd.strip()
            upper = line.upper()
            if upper.startswith("RUN "):
                recipe.append({"kind": "run_commands", "commands": [line[4:]]})
            elif upper.startswith("ENV "):
                body = line[4:]
                if "=" in body:
                    key, _, value = body.partition("=")
                    recipe.append({"kind": "env", "vars": {key.strip(): value.strip()}})
            elif upper.startswith("WORKDIR "):
                recipe.append({"kind": "workdir", "path": line[8:].strip()})
            else:
                recipe.append({"kind": "dockerfile_commands", "commands": [line]})
        resp = await self.scheduler.image_get_or_create(recipe)
        out = self.api.ImageGetOrCreateResponse(image_id=resp["image_id"])
        out.metadata.image_builder_version = "local"
        return out

    async def ImageJoinStreaming(self, request: Any, context: Any) -> Any:
        # local builds are synchronous: report the (finished) result once
        info = await self.scheduler.image_info(request.image_id)
        resp = self.api.ImageJoinStreamingResponse()
        resp.result.status = (
            self.api.GenericResult.GenericStatus.GENERIC_STATUS_SUCCESS
            if info.get("built")
            else self.api.GenericResult.GenericStatus.GENERIC_STATUS_FAILURE
        )
        entry = resp.task_logs.add()
        entry.data = info.get("build_log", "")
        yield resp

    # -- TaskCommandRouter (the second gRPC plane) -----------------------------
    # The reference client dials the WORKER directly for exec/stdio
    # (task_command_router.proto:373, task_command_router_client.py:211).
    # Locally the same service answers on the scheduler's grpc socket; the
    # offset-resume semantics ride the native sandbox stdio buffers.

    def _sandbox_by_task(self, task_id: str) -> Any:
        for sb in self.scheduler.sandbox_service.sandboxes.values():
            if sb.task_id == task_id:
                return sb
        return None

    async def TaskExecStart(self, request: Any, context: Any) -> Any:
        sb = self._sandbox_by_task(request.task_id)
        if sb is None:
            import grpc

            await context.abort(grpc.StatusCode.NOT_FOUND, "task not found")
        await self.scheduler.sandbox_exec(
            sb.sandbox_id,
            list(request.command_args),
            env=dict(request.env) or None,
            workdir=request.workdir if request.HasField("workdir") else None,
            timeout=request.timeout_secs if request.HasField("timeout_secs") else None,
            exec_id=request.exec_id,
        )
        return self.router_pb.TaskExecStartResponse()

    async def TaskExecStdioRead(self, request: Any, context: Any) -> Any:
        fd = 2 if request.file_descriptor == 1 else 1  # enum: 0=stdout, 1=stderr
        offset = request.offset
        while True:
            resp = await self.scheduler.sandbox_stdio_read(
                target_id=request.exec_id, fd=fd, offset=offset, timeout=5.0
            )
            if resp["data"]:
                out = self.router_pb.TaskExecStdioReadResponse()
                out.data = resp["data"]
                offset = resp["next_offset"]
                yield out
            if resp["eof"]:
                return

    async def TaskExecStdinWrite(self, request: Any, context: Any) -> Any:
        await self.scheduler.sandbox_stdin_write(
            target_id=request.exec_id,
            offset=request.offset,
            data=bytes(request.data),
            eof=request.eof,
        )
        return self.router_pb.TaskExecStdinWriteResponse()

    async def TaskExecWait(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.sandbox_wait(
            request.exec_id, timeout=None, raise_on_timeout=False
        )
        out = self.router_pb.TaskExecWaitResponse()
        rc = native.get("returncode")
        if rc is not None:
            if rc < 0:
                out.signal = -rc
            else:
                out.code = rc
        return out

    async def TaskExecPoll(self, request: Any, context: Any) -> Any:
        native = await self.scheduler.sandbox_poll(request.exec_id)
        out = self.router_pb.TaskExecPollResponse()
        rc = native.get("returncode")
        if rc is not None:
            if rc < 0:
                out.signal = -rc
            else:
                out.code = rc
        return out

    # -- helpers -------------------------------------------------------------

    async def _abort_not_found(self, context: Any, exc: Exception) -> None:
        import grpc

        from ..exception import NotFoundError

        if isinstance(exc, NotFoundError):
            await context.abort(grpc.StatusCode.NOT_FOUND, str(exc))
