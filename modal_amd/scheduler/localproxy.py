"""Local HTTP forward proxy backing ``modal.Proxy``.

The reference's Proxy (reference proxy.py:57) gives functions a static
outbound IP by routing their egress through a managed proxy host. The
single-node analog is a real forward proxy on 127.0.0.1: functions declared
with ``proxy=`` get ``HTTP_PROXY``/``HTTPS_PROXY`` pointed at it, so their
HTTP clients route exactly like they would in the reference — through one
shared egress point (useful for auditing/throttling, and it makes the proxy
path testable without egress).

Supports the two proxy request forms:
* ``CONNECT host:port`` — open a TCP tunnel, reply ``200 Connection
  Established``, then pipe bytes both ways (what https clients use).
* absolute-form requests (``GET http://host:port/path``) — connect to the
  origin, rewrite the request line to origin-form, pipe.
"""

from __future__ import annotations

import asyncio
from typing import Optional


class LocalForwardProxy:
    def __init__(self) -> None:
        self._server: Optional[asyncio.AbstractServer] = None
        self.port: int = 0
        self.bytes_relayed = 0
        self.connections = 0

    async def start(self) -> int:
        if self._server is not None:
            return self.port
        self._server = await asyncio.start_server(self._handle, "127.0.0.1", 0)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    async def _pipe(self, src: asyncio.StreamReader, dst: asyncio.StreamWriter) -> None:
        try:
            while True:
                data = await src.read(65536)
                if not data:
                    break
                self.bytes_relayed += len(data)
                dst.write(data)
                await dst.drain()
        except (ConnectionError, asyncio.CancelledError):
            pass
        finally:
            try:
                dst.close()
            except Exception:
                pass

    async def _handle(
        self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter
    ) -> None:
        self.connections += 1
        try:
            request_line = await reader.readline()
            parts = request_line.decode("latin-1").split()
            if len(parts) != 3:
                writer.close()
                return
            method, target, version = parts
            # drain request headers (kept for absolute-form forwarding)
            headers = []
            while True:
                line = await reader.readline()
                if line in (b"\r\n", b"\n", b""):
                    break
                headers.append(line)

            if method.upper() == "CONNECT":
                host, _, port_s = target.partition(":")
                try:
                    upstream_r, upstream_w = await asyncio.open_connection(
                        host, int(port_s or 443)
                    )
                except OSError:
                    writer.write(b"HTTP/1.1 502 Bad Gateway\r\n\r\n")
                    await writer.drain()
                    writer.close()
                    return
                writer.write(f"{version} 200 Connection Established\r\n\r\n".encode())
                await writer.drain()
                await asyncio.gather(
                    self._pipe(reader, upstream_w), self._pipe(upstream_r, writer)
                )
                return

            # absolute-form: GET http://host:port/path
            if "://" not in target:
                writer.write(b"HTTP/1.1 400 Bad Request\r\n\r\n")
                await writer.drain()
                writer.close()
                return
            rest = target.split("://", 1)[1]
            hostport, _, path = rest.partition("/")
            host, _, port_s = hostport.partition(":")
            path = "/" + path
            try:
                upstream_r, upstream_w = await asyncio.open_connection(
                    host, int(port_s or 80)
                )
            except OSError:
                writer.write(b"HTTP/1.1 502 Bad Gateway\r\n\r\n")
                await writer.drain()
                writer.close()
                return
            upstream_w.write(f"{method} {path} {version}\r\n".encode())
            drop = (b"proxy-connection:", b"proxy-authorization:")
            for h in headers:
                if not h.lower().startswith(drop):
                    upstream_w.write(h)
            upstream_w.write(b"\r\n")
            await upstream_w.drain()
            await asyncio.gather(
                self._pipe(reader, upstream_w), self._pipe(upstream_r, writer)
            )
        except (ConnectionError, asyncio.IncompleteReadError):
            try:
                writer.close()
            except Exception:
                pass
