"""Worker-plane transport: length-prefixed msgpack frames over Unix sockets.

This replaces the reference's gRPC/HTTP2 control plane
(/root/reference/py/modal/_utils/grpc_utils.py) with a single-node design:
4-byte little-endian length + msgpack body, no TLS, no HTTP framing. Payload
bytes ride inside msgpack bin values (zero re-encoding). An RPC layer on top
gives workers access to scheduler services (queues, dicts, volumes, blobs)
with the same semantics user code sees in the client process.
"""

from __future__ import annotations

import asyncio
import os
import itertools
import struct
from typing import Any, Awaitable, Callable, Optional

import msgpack

_LEN = struct.Struct("<I")

MAX_FRAME = 1 << 31  # 2 GiB guard

# frames above this ride the shared-memory ring (modal_amd._core.ShmRing)
# with only a doorbell on the socket; below it the socket round-trip is
# lower-latency than a ring poll
RING_MIN_FRAME = int(os.environ.get("MODAL_AMD_RING_MIN", 32 * 1024))
RING_CAPACITY = 64 << 20  # 64 MiB per direction per worker

try:
    from .. import _core  # C++ native core (built in-tree)
except ImportError:  # pragma: no cover - source-only checkout
    _core = None


def pack(msg: dict) -> bytes:
    body = msgpack.packb(msg, use_bin_type=True)
    return _LEN.pack(len(body)) + body


async def read_frame(reader: asyncio.StreamReader) -> Optional[dict]:
    try:
        header = await reader.readexactly(4)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    (length,) = _LEN.unpack(header)
    if length > MAX_FRAME:
        raise ValueError(f"Frame too large: {length}")
    try:
        body = await reader.readexactly(length)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    return msgpack.unpackb(body, raw=False, strict_map_key=False)


class Connection:
    """One framed connection with an RPC request/response layer.

    Both directions can send one-way messages (dispatched to ``handler``) and
    RPCs (``call``), multiplexed by message type:
      {"t": <kind>, ...}                     one-way
      {"t": "rpc", "i": id, "m": method, "p": params}  request
      {"t": "rpc_r", "i": id, "r": result} / {"t": "rpc_e", "i": id, "e": msg}
    """

    def __init__(
        self,
        reader: asyncio.StreamReader,
        writer: asyncio.StreamWriter,
        handler: Callable[[dict], Awaitable[None]],
        rpc_target: Optional[Any] = None,
    ):
        self.reader = reader
        self.writer = writer
        self.handler = handler
        self.rpc_target = rpc_target  # object whose async methods serve inbound RPCs
        self._rpc_seq = itertools.count(1)
        self._pending: dict[int, asyncio.Future] = {}
        self._send_lock = asyncio.Lock()
        self._closed = asyncio.Event()
        self._reader_task: Optional[asyncio.Task] = None
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        # shm ring pair: our producer side and our consumer side
        self.ring_out: Any = None
        self.ring_in: Any = None
        self.ring_frames_sent = 0
        self.ring_frames_received = 0

    def attach_rings(self, out_path: Optional[str], in_path: Optional[str], create: bool) -> bool:
        """Attach the shared-memory bulk channel (both sides call this)."""
        if _core is None or not out_path or not in_path:
            return False
        try:
            self.ring_out = _core.ShmRing(out_path, RING_CAPACITY, create)
            self.ring_in = _core.ShmRing(in_path, RING_CAPACITY, create)
            return True
        except Exception:
            self.ring_out = None
            self.ring_in = None
            return False

    def start(self) -> None:
        self._loop = asyncio.get_running_loop()
        self._reader_task = self._loop.create_task(self._read_loop())

    @property
    def closed(self) -> bool:
        return self._closed.is_set()

    async def wait_closed(self) -> None:
        await self._closed.wait()

    async def send(self, msg: dict) -> None:
        body = msgpack.packb(msg, use_bin_type=True)
        if self.ring_out is not None and len(body) >= RING_MIN_FRAME:
            # bulk path: payload through the shm ring, doorbell on the socket
            pushed = self.ring_out.push(body)
            if pushed:
                self.ring_frames_sent += 1
                async with self._send_lock:
                    self.writer.write(pack({"t": "rb"}))
                    await self.writer.drain()
                return
            # ring full: fall through to the socket
        data = _LEN.pack(len(body)) + body
        async with self._send_lock:
            self.writer.write(data)
            await self.writer.drain()

    async def call(self, method: str, params: Any = None, timeout: Optional[float] = None) -> Any:
        # RPCs may originate on another event loop (e.g. the synchronizer loop
        # driving user code's handles inside a worker); bridge to the loop that
        # owns this connection's streams.
        running = asyncio.get_running_loop()
        if self._loop is not None and running is not self._loop:
            fut = asyncio.run_coroutine_threadsafe(self.call(method, params, timeout), self._loop)
            return await asyncio.wrap_future(fut)
        rpc_id = next(self._rpc_seq)
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[rpc_id] = fut
        try:
            await self.send({"t": "rpc", "i": rpc_id, "m": method, "p": params})
            if timeout is not None:
                return await asyncio.wait_for(fut, timeout)
            return await fut
        finally:
            self._pending.pop(rpc_id, None)

    async def _dispatch(self, msg: dict) -> None:
        kind = msg.get("t")
        if kind == "rpc":
            asyncio.get_running_loop().create_task(self._serve_rpc(msg))
        elif kind == "rpc_r":
            fut = self._pending.get(msg["i"])
            if fut is not None and not fut.done():
                fut.set_result(msg.get("r"))
        elif kind == "rpc_e":
            fut = self._pending.get(msg["i"])
            if fut is not None and not fut.done():
                fut.set_exception(RemoteRPCError(msg.get("e", "remote error"), msg.get("c")))
        else:
            await self.handler(msg)

    async def _serve_rpc(self, msg: dict) -> None:
        rpc_id = msg["i"]
        method = msg["m"]
        try:
            if self.rpc_target is None:
                raise RuntimeError("No RPC target on this end")
            fn = getattr(self.rpc_target, method, None)
            if fn is None or method.startswith("_"):
                raise RuntimeError(f"Unknown RPC method {method!r}")
            params = msg.get("p") or {}
            result = await fn(**params)
            await self.send({"t": "rpc_r", "i": rpc_id, "r": result})
        except asyncio.CancelledError:
            raise
        except BaseException as exc:  # report, keep connection alive
            await self.send(
                {"t": "rpc_e", "i": rpc_id, "e": f"{type(exc).__name__}: {exc}", "c": type(exc).__name__}
            )

    async def _read_loop(self) -> None:
        try:
            while True:
                msg = await read_frame(self.reader)
                if msg is None:
                    break
                if msg.get("t") == "rb":
                    # doorbell: drain the inbound ring and dispatch its frames
                    if self.ring_in is not None:
                        for body in self.ring_in.pop_all():
                            self.ring_frames_received += 1
                            await self._dispatch(
                                msgpack.unpackb(body, raw=False, strict_map_key=False)
                            )
                    continue
                await self._dispatch(msg)
        except asyncio.CancelledError:
            pass
        except Exception:
            pass
        finally:
            self._closed.set()
            for fut in self._pending.values():
                if not fut.done():
                    fut.set_exception(ConnectionError("connection closed"))
            try:
                self.writer.close()
            except Exception:
                pass

    async def close(self) -> None:
        if self._reader_task is not None:
            self._reader_task.cancel()
        try:
            self.writer.close()
            await self.writer.wait_closed()
        except Exception:
            pass
        self._closed.set()


class RemoteRPCError(Exception):
    def __init__(self, message: str, code: Optional[str] = None):
        super().__init__(message)
        self.code = code
