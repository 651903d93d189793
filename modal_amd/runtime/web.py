"""Worker-side web wrappers: FastAPI / ASGI / WSGI / web_server bridges.

Parity: /root/reference/py/modal/_runtime/asgi.py — ``asgi_app_wrapper``
(:99), ``wsgi_app_wrapper`` (:233), web_server proxy (:280-470),
``magic_fastapi_app`` (:240). A request arrives as a dict (method, path,
query_string, headers, body) and the wrapper returns
{status, headers, body}.
"""

from __future__ import annotations

import asyncio
import inspect
from typing import Any, Callable


def magic_fastapi_app(fn: Callable, method: str, docs: bool) -> Any:
    from fastapi import FastAPI

    app = FastAPI(docs_url="/docs" if docs else None)
    app.add_api_route("/", fn, methods=[method.upper()])
    return app


def _build_scope(req: dict) -> dict:
    return {
        "type": "http",
        "asgi": {"version": "3.0", "spec_version": "2.3"},
        "http_version": "1.1",
        "method": req["method"],
        "scheme": "http",
        "path": req.get("path") or "/",
        "raw_path": (req.get("path") or "/").encode(),
        "query_string": (req.get("query_string") or "").encode(),
        "root_path": "",
        "headers": [(k.lower().encode(), v.encode()) for k, v in req.get("headers", [])],
        "server": ("127.0.0.1", 80),
        "client": ("127.0.0.1", 0),
    }


async def run_asgi_streaming(app: Any, req: dict, emit: Callable) -> None:
    """Run the ASGI app, forwarding the response head then each body chunk
    through ``emit`` as they are produced (parity: the reference streams
    bodies via data_out chunks, asgi.py:140)."""
    scope = _build_scope(req)
    incoming = [{"type": "http.request", "body": req.get("body") or b"", "more_body": False}]
    done = asyncio.Event()
    state = {"status": 500, "headers": [], "head_sent": False}

    async def receive() -> dict:
        if incoming:
            return incoming.pop(0)
        await done.wait()
        return {"type": "http.disconnect"}

    async def send_head_once() -> None:
        if not state["head_sent"]:
            state["head_sent"] = True
            await emit({"status": state["status"], "headers": state["headers"]})

    async def send(message: dict) -> None:
        if message.get("type") == "http.response.start":
            state["status"] = message["status"]
            state["headers"] = [[k.decode(), v.decode()] for k, v in message.get("headers", [])]
        elif message.get("type") == "http.response.body":
            await send_head_once()
            body = message.get("body", b"")
            if body:
                await emit(bytes(body))
            if not message.get("more_body"):
                done.set()

    await app(scope, receive, send)
    await send_head_once()  # head even for empty-bodied responses


async def run_asgi(app: Any, req: dict) -> dict:
    scope = _build_scope(req)
    incoming = [{"type": "http.request", "body": req.get("body") or b"", "more_body": False}]
    sent: list[dict] = []
    done = asyncio.Event()

    async def receive() -> dict:
        if incoming:
            return incoming.pop(0)
        await done.wait()
        return {"type": "http.disconnect"}

    async def send(message: dict) -> None:
        sent.append(message)
        if message.get("type") == "http.response.body" and not message.get("more_body"):
            done.set()

    await app(scope, receive, send)
    status = 500
    headers: list = []
    body = b""
    for message in sent:
        if message["type"] == "http.response.start":
            status = message["status"]
            headers = [
                [k.decode(), v.decode()] for k, v in message.get("headers", [])
            ]
        elif message["type"] == "http.response.body":
            body += message.get("body", b"")
    return {"status": status, "headers": headers, "body": body}


def run_wsgi(app: Any, req: dict) -> dict:
    environ = _wsgi_environ(req)
    captured: dict = {"status": 500, "headers": []}

    def start_response(status: str, headers: list, exc_info: Any = None) -> Any:
        captured["status"] = int(status.split()[0])
        captured["headers"] = [[k, v] for k, v in headers]

    chunks = app(environ, start_response)
    body = b"".join(chunks)
    if hasattr(chunks, "close"):
        chunks.close()
    return {"status": captured["status"], "headers": captured["headers"], "body": body}


def _wsgi_environ(req: dict) -> dict:
    import io

    environ = {
        "REQUEST_METHOD": req["method"],
        "SCRIPT_NAME": "",
        "PATH_INFO": req.get("path") or "/",
        "QUERY_STRING": req.get("query_string") or "",
        "SERVER_NAME": "127.0.0.1",
        "SERVER_PORT": "80",
        "SERVER_PROTOCOL": "HTTP/1.1",
        "wsgi.version": (1, 0),
        "wsgi.url_scheme": "http",
        "wsgi.input": io.BytesIO(req.get("body") or b""),
        "wsgi.errors": io.StringIO(),
        "wsgi.multithread": True,
        "wsgi.multiprocess": False,
        "wsgi.run_once": False,
    }
    for key, value in req.get("headers", []):
        if key.lower() == "content-type":
            environ["CONTENT_TYPE"] = value
        elif key.lower() == "content-length":
            environ["CONTENT_LENGTH"] = value
        else:
            environ["HTTP_" + key.upper().replace("-", "_")] = value
    return environ


def run_wsgi_streaming(app: Any, req: dict, emit_threadsafe: Callable) -> None:
    """Iterate the WSGI body on this (executor) thread, emitting the head
    before the first chunk (WSGI apps may call start_response lazily)."""
    environ = _wsgi_environ(req)
    captured: dict = {"status": 500, "headers": [], "head_sent": False}

    def start_response(status: str, headers: list, exc_info: Any = None) -> Any:
        captured["status"] = int(status.split()[0])
        captured["headers"] = [[k, v] for k, v in headers]

    def head_once() -> None:
        if not captured["head_sent"]:
            captured["head_sent"] = True
            emit_threadsafe({"status": captured["status"], "headers": captured["headers"]})

    chunks = app(environ, start_response)
    try:
        for chunk in chunks:
            head_once()
            if chunk:
                emit_threadsafe(bytes(chunk))
    finally:
        head_once()
        if hasattr(chunks, "close"):
            chunks.close()


class WebEndpointRuntime:
    """Built once per web function in the worker; callable per request."""

    def __init__(self, web_config: dict, raw_fn: Callable):
        self.config = web_config
        self.kind = web_config["type"]
        self.raw_fn = raw_fn
        self._app: Any = None
        self._started = False

    def _ensure_app(self) -> Any:
        if self._app is None:
            if self.kind == "fastapi":
                self._app = magic_fastapi_app(
                    self.raw_fn, self.config.get("method", "GET"), self.config.get("docs", False)
                )
            elif self.kind == "asgi":
                self._app = self.raw_fn()
            elif self.kind == "wsgi":
                self._app = self.raw_fn()
        return self._app

    async def handle_streaming(self, req: dict, emit: Callable) -> None:
        """Streaming counterpart of :meth:`handle`: calls ``await emit(head)``
        with ``{"status", "headers"}`` then ``await emit(chunk_bytes)`` per
        body chunk, returning when the response is complete."""
        if self.kind in ("fastapi", "asgi"):
            await run_asgi_streaming(self._ensure_app(), req, emit)
            return
        if self.kind == "wsgi":
            loop = asyncio.get_running_loop()

            def emit_threadsafe(obj: Any) -> None:
                asyncio.run_coroutine_threadsafe(emit(obj), loop).result()

            await loop.run_in_executor(
                None, run_wsgi_streaming, self._ensure_app(), req, emit_threadsafe
            )
            return
        if self.kind == "web_server":
            if not self._started:
                self._started = True
                result = self.raw_fn()
                if inspect.iscoroutine(result):
                    await result
                await asyncio.sleep(min(self.config.get("startup_timeout", 5.0), 0.5))
            await self._proxy_streaming(req, emit)
            return
        raise ValueError(f"Unknown web endpoint type {self.kind}")

    async def _proxy_streaming(self, req: dict, emit: Callable) -> None:
        import aiohttp

        port = self.config["port"]
        url = f"http://127.0.0.1:{port}{req.get('path') or '/'}"
        if req.get("query_string"):
            url += f"?{req['query_string']}"
        async with aiohttp.ClientSession() as session:
            async with session.request(
                req["method"], url, data=req.get("body") or None,
                headers={k: v for k, v in req.get("headers", [])},
            ) as resp:
                await emit({
                    "status": resp.status,
                    "headers": [[k, v] for k, v in resp.headers.items()],
                })
                async for chunk in resp.content.iter_chunked(1 << 16):
                    if chunk:
                        await emit(chunk)

    async def handle(self, req: dict) -> dict:
        if self.kind in ("fastapi", "asgi"):
            return await run_asgi(self._ensure_app(), req)
        if self.kind == "wsgi":
            loop = asyncio.get_running_loop()
            return await loop.run_in_executor(None, run_wsgi, self._ensure_app(), req)
        if self.kind == "web_server":
            if not self._started:
                self._started = True
                result = self.raw_fn()
                if inspect.iscoroutine(result):
                    await result
                # give the server a beat to bind (parity: startup_timeout)
                await asyncio.sleep(min(self.config.get("startup_timeout", 5.0), 0.5))
            return await self._proxy(req)
        raise ValueError(f"Unknown web endpoint type {self.kind}")

    async def _proxy(self, req: dict) -> dict:
        import aiohttp

        port = self.config["port"]
        url = f"http://127.0.0.1:{port}{req.get('path') or '/'}"
        if req.get("query_string"):
            url += f"?{req['query_string']}"
        async with aiohttp.ClientSession() as session:
            async with session.request(
                req["method"], url, data=req.get("body") or None,
                headers={k: v for k, v in req.get("headers", [])},
            ) as resp:
                body = await resp.read()
                return {
                    "status": resp.status,
                    "headers": [[k, v] for k, v in resp.headers.items()],
                    "body": body,
                }
