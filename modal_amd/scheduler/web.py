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

from .._serialization import serialize
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
        payload = serialize(("P", ((req,), {})))
        resp = await self.scheduler.function_map(
            function_id=function_id, kind="unary", pipelined_inputs=[{"payload": payload, "method": "__web__"}]
        )
        call_id = resp["function_call_id"]
        await self.scheduler.function_finish_inputs(function_call_id=call_id)
        rec = await self.scheduler.function_wait_output(call_id, 0, timeout=300)
        if rec.status != GENERIC_STATUS_SUCCESS:
            return web.Response(status=500, text=rec.exc_repr or "function failed")
        from .._serialization import deserialize

        out = deserialize(rec.output)
        headers = {k: v for k, v in out.get("headers", [])}
        headers.pop("Content-Length", None)
        headers.pop("content-length", None)
        return web.Response(
            status=out.get("status", 200), body=out.get("body", b""), headers=headers
        )
