"""Web endpoints: fastapi_endpoint / asgi_app / wsgi_app through the gateway."""

from __future__ import annotations

import json
import urllib.request

import pytest

import modal_amd as modal


def _get(url: str, data: bytes = None, method: str = "GET") -> tuple[int, bytes]:
    req = urllib.request.Request(url, data=data, method=method)
    try:
        with urllib.request.urlopen(req, timeout=30) as resp:
            return resp.status, resp.read()
    except urllib.error.HTTPError as e:
        return e.code, e.read()


def test_fastapi_endpoint(client):
    app = modal.App("web-app")

    @app.function()
    @modal.fastapi_endpoint(method="GET")
    def hello(name: str = "world"):
        return {"greeting": f"hello {name}"}

    with app.run(client=client):
        url = hello.web_url
        assert url and url.startswith("http://127.0.0.1:")
        status, body = _get(url + "/?name=amd")
        assert status == 200, body
        assert json.loads(body) == {"greeting": "hello amd"}


def test_asgi_app_endpoint(client):
    app = modal.App("asgi-app")

    @app.function()
    @modal.asgi_app()
    def my_asgi():
        async def app_impl(scope, receive, send):
            assert scope["type"] == "http"
            await send(
                {"type": "http.response.start", "status": 201,
                 "headers": [(b"x-custom", b"yes")]}
            )
            await send({"type": "http.response.body", "body": b"asgi-body"})

        return app_impl

    with app.run(client=client):
        status, body = _get(my_asgi.web_url + "/any/path")
        assert status == 201
        assert body == b"asgi-body"


def test_wsgi_app_endpoint(client):
    app = modal.App("wsgi-app")

    @app.function()
    @modal.wsgi_app()
    def my_wsgi():
        def app_impl(environ, start_response):
            start_response("200 OK", [("Content-Type", "text/plain")])
            return [b"wsgi says ", environ["REQUEST_METHOD"].encode()]

        return app_impl

    with app.run(client=client):
        status, body = _get(my_wsgi.web_url + "/", data=b"x", method="POST")
        assert status == 200
        assert body == b"wsgi says POST"


def test_fastapi_post_body(client):
    app = modal.App("web-post")

    @app.function()
    @modal.fastapi_endpoint(method="POST")
    def echo(payload: dict):
        return {"got": payload}

    with app.run(client=client):
        status, body = _get(
            echo.web_url, data=json.dumps({"a": 1}).encode(), method="POST"
        )
        # fastapi parses the dict body from JSON
        assert status in (200, 422), body
        if status == 200:
            assert json.loads(body) == {"got": {"a": 1}}


def test_streaming_response_arrives_incrementally(client):
    """Body chunks reach the HTTP client BEFORE the handler finishes
    (true streaming through the generator data channel, not buffering)."""
    import time

    app = modal.App("stream-app")

    @app.function()
    @modal.asgi_app()
    def streamer():
        import asyncio

        async def app_impl(scope, receive, send):
            await send({"type": "http.response.start", "status": 200, "headers": []})
            await send({"type": "http.response.body", "body": b"early|", "more_body": True})
            await asyncio.sleep(1.5)
            await send({"type": "http.response.body", "body": b"late", "more_body": False})

        return app_impl

    with app.run(client=client):
        t0 = time.monotonic()
        with urllib.request.urlopen(streamer.web_url, timeout=30) as resp:
            first = resp.read(6)
            first_latency = time.monotonic() - t0
            rest = resp.read()
        assert first == b"early|"
        assert rest == b"late"
        # the first chunk must not have waited for the 1.5 s sleep
        assert first_latency < 1.2, f"first chunk took {first_latency:.2f}s (buffered?)"


def test_streaming_wsgi_chunks(client):
    app = modal.App("wsgi-stream-app")

    @app.function()
    @modal.wsgi_app()
    def wsgi_streamer():
        def app_impl(environ, start_response):
            start_response("200 OK", [("Content-Type", "text/plain")])
            for i in range(5):
                yield f"chunk{i};".encode()

        return app_impl

    with app.run(client=client):
        status, body = _get(wsgi_streamer.web_url)
        assert status == 200
        assert body == b"".join(f"chunk{i};".encode() for i in range(5))
