"""Web gateway: HTTP ingress for web-decorated functions.

The reference serves web endpoints through Modal's edge with an ASGI bridge
in the container (/root/reference/py/modal/_runtime/asgi.py:99). Locally the
gateway is an aiohttp server in the scheduler process: requests become
function invocations (DataFormat.ASGI-shaped payloads), workers run the
FastAPI/ASGI/WSGI wrapper, and responses come back through the output plane.
``web_server(port)`` functions are proxied straight to their port (same
node, same network namespace).
"""

from __future__ import annotations

import asyncio
from typing import TYPE_CHECKING, Any, Optional

from .._serialization import serialize_fast
from .calls import GENERIC_STATUS_SUCCESS

if TYPE_CHECKING:
    from .core import Scheduler


class WebGateway:
    def __init__(self, scheduler: "Scheduler"):
        self.scheduler = scheduler
        self.port: Optional[int] = None
        self._runner: Any = None
        self._site: Any = None
        self.routes: dict[str, str] = {}  # label -> function_id
        self._lock = asyncio.Lock()

    @property
    def base_url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def url_for(self, label: str) -> str:
        return f"{self.base_url}/{label}"

    async def ensure_started(self) -> None:
        async with self._lock:
            if self.port is not None:
                return
            from aiohttp import web

            app = web.Application(client_max_size=1 << 28)
            app.router.add_route("*", "/{label}{tail:(/.*)?}", self._handle)
            self._runner = web.AppRunner(app)
            await self._runner.setup()
            self._site = web.TCPSite(self._runner, "127.0.0.1", 0)
            await self._site.start()
            self.port = self._site._server.sockets[0].getsockname()[1]

    async def stop(self) -> None:
        if self._runner is not None:
            await self._runner.cleanup()
            self.port = None

    def register(self, label: str, function_id: str) -> str:
        self.routes[label] = function_id
        return self.url_for(label)

    async def _handle(self, request: Any) -> Any:
        from aiohttp import web

        label = request.match_info["label"]
        tail = request.match_info.get("tail") or "/"
        if label == "_metrics":  # Prometheus scrape target
            return web.Response(
                text=await self.scheduler.node_metrics(), content_type="text/plain"
            )
        function_id = self.routes.get(label)
        if function_id is None:
            return web.Response(status=404, text=f"No web function '{label}'")
        body = await request.read()
        req = {
            "method": request.method,
            "path": tail,
            "query_string": request.query_string,
            "headers": [[k, v] for k, v in request.headers.items()],
            "body": body,
        }
        payload = serialize_fast(("P", ((req,), {})))  # primitives only: C pickler
        resp = await self.scheduler.function_map(
            function_id=function_id, kind="unary", pipelined_inputs=[{"payload": payload, "method": "__web__"}]
        )
        call_id = resp["function_call_id"]
        await self.scheduler.function_finish_inputs(function_call_id=call_id)
        # The worker streams {status, headers} then body chunks over the
        # generator data channel; forward them as they arrive. The final
        # output (GeneratorDone or a failure) settles the call.
        from .._serialization import deserialize

        record = self.scheduler._call(call_id)
        q = record.gen_queue(0)
        output_task = asyncio.ensure_future(
            self.scheduler.function_wait_output(call_id, 0, timeout=300)
        )
        stream: Any = None

        async def forward(entry: tuple) -> bool:
            """Apply one gen-channel entry; True when the stream is done."""
            nonlocal stream
            _index, data, _fmt, done_flag = entry
            if done_flag:
                return True
            value = deserialize(data)
            try:
                if stream is None:
                    headers = {k: v for k, v in value.get("headers", [])}
                    headers.pop("Content-Length", None)
                    headers.pop("content-length", None)
                    stream = web.StreamResponse(status=value.get("status", 200), headers=headers)
                    await stream.prepare(request)
                else:
                    await stream.write(value)
            except (ConnectionResetError, OSError):
                return True  # client went away: stop forwarding
            return False

        try:
            finished = False
            while not finished:
                get_task = asyncio.ensure_future(q.get())
                await asyncio.wait({get_task, output_task}, return_when=asyncio.FIRST_COMPLETED)
                if get_task.done():
                    finished = await forward(get_task.result())
                    continue
                get_task.cancel()
                rec = output_task.result()
                if rec.status != GENERIC_STATUS_SUCCESS:
                    if stream is None:
                        return web.Response(status=500, text=rec.exc_repr or "function failed")
                    break  # mid-stream failure: truncate the response
                # success settled first: drain whatever is already queued
                while not finished:
                    try:
                        finished = await forward(q.get_nowait())
                    except asyncio.QueueEmpty:
                        finished = True
        finally:
            if not output_task.done():
                output_task.cancel()
        if stream is None:
            return web.Response(status=500, text="web function produced no response")
        try:
            await stream.write_eof()
        except ConnectionResetError:
            pass  # client went away mid-stream
        return stream
