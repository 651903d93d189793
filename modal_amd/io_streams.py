"""Sandbox/exec stdio streams: offset-resumable readers and writers.

Parity: /root/reference/py/modal/io_streams.py — ``_StreamReader`` (:56),
by-line splitting, ``StreamType``; resumable offsets per the command-router
contract (task_command_router_client.py:431-614).
"""

from __future__ import annotations

import enum
from typing import Any, AsyncGenerator

from ._sync import synchronize_api


class StreamType(enum.Enum):
    PIPE = "pipe"
    STDOUT = "stdout"
    DEVNULL = "devnull"


class _StreamReader:
    """Reads one fd (1=stdout, 2=stderr) of a sandbox or exec'd process."""

    def __init__(self, client: Any, target_id: str, fd: int, text: bool = True):
        self._client = client
        self._target_id = target_id
        self._fd = fd
        self._text = text
        self._offset = 0

    @property
    def file_descriptor(self) -> int:
        return self._fd

    async def read(self) -> Any:
        """Read everything until EOF (parity: reference read() semantics)."""
        chunks = []
        while True:
            resp = await self._client.svc.sandbox_stdio_read(
                target_id=self._target_id, fd=self._fd, offset=self._offset, timeout=55.0
            )
            data = resp["data"]
            if data:
                chunks.append(data)
                self._offset = resp["next_offset"]
            if resp["eof"]:
                break
        blob = b"".join(chunks)
        return blob.decode("utf-8", errors="replace") if self._text else blob

    async def read_chunk(self, timeout: float = 55.0) -> tuple[bytes, bool]:
        resp = await self._client.svc.sandbox_stdio_read(
            target_id=self._target_id, fd=self._fd, offset=self._offset, timeout=timeout
        )
        if resp["data"]:
            self._offset = resp["next_offset"]
        return resp["data"], resp["eof"]

    async def __aiter__(self) -> AsyncGenerator[Any, None]:
        """Iterate by line (parity: reference by-line splitting)."""
        pending = b""
        while True:
            data, eof = await self.read_chunk()
            pending += data
            while b"\n" in pending:
                line, _, pending = pending.partition(b"\n")
                yield (line.decode("utf-8", errors="replace") + "\n") if self._text else line + b"\n"
            if eof:
                if pending:
                    yield pending.decode("utf-8", errors="replace") if self._text else pending
                return


class _StreamWriter:
    """Writes stdin with resumable offsets; flushed on drain()."""

    def __init__(self, client: Any, target_id: str):
        self._client = client
        self._target_id = target_id
        self._buffer = bytearray()
        self._offset = 0
        self._eof = False

    def write(self, data: Any) -> None:
        if self._eof:
            raise ValueError("Stdin is closed")
        if isinstance(data, str):
            data = data.encode("utf-8")
        self._buffer.extend(data)

    def write_eof(self) -> None:
        self._eof = True

    async def drain(self) -> None:
        data = bytes(self._buffer)
        self._buffer.clear()
        self._offset = await self._client.svc.sandbox_stdin_write(
            target_id=self._target_id, offset=self._offset, data=data, eof=self._eof
        )


StreamReader = synchronize_api(_StreamReader, "StreamReader")
StreamWriter = synchronize_api(_StreamWriter, "StreamWriter")
