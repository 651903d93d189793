"""Tunnels: expose a port through a relay (parity:
/root/reference/py/modal/_tunnel.py:18,61 — TunnelStart/TunnelStop give the
container a public host:port relayed to its local port).

The single-node equivalent is a REAL TCP relay: ``forward(port)`` starts a
listener on a fresh port and pumps bytes bidirectionally to the target
port. That preserves the reference's observable semantics — the tunnel
address is distinct from the service address, connections through it are
proxied, and closing the tunnel kills the listener while the service stays
up — rather than handing back the original port unchanged (the round-1
placeholder the review called out).
"""

from __future__ import annotations

import contextlib
import socket
import threading
from dataclasses import dataclass, field
from typing import Any, Iterator, Optional


@dataclass(frozen=True)
class Tunnel:
    host: str
    port: int
    unencrypted_host: str = ""
    unencrypted_port: int = 0
    _relay: Optional["_TcpRelay"] = field(default=None, compare=False, repr=False)

    @property
    def url(self) -> str:
        value = f"https://{self.host}"
        if self.port != 443:
            value += f":{self.port}"
        return value

    @property
    def tls_socket(self) -> tuple[str, int]:
        return (self.host, self.port)

    @property
    def tcp_socket(self) -> tuple[str, int]:
        if not self.unencrypted_host:
            raise ValueError("Tunnel was not created with unencrypted=True")
        return (self.unencrypted_host, self.unencrypted_port)


class _TcpRelay:
    """Byte-pump relay: accept on an ephemeral port, connect to the target,
    shuttle both directions until either side closes."""

    def __init__(self, target_host: str, target_port: int):
        self.target = (target_host, target_port)
        self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._listener.bind(("127.0.0.1", 0))
        self._listener.listen(64)
        self.port = self._listener.getsockname()[1]
        self._closed = threading.Event()
        self._accept_thread = threading.Thread(target=self._accept_loop, daemon=True)
        self._accept_thread.start()
        self.connections_served = 0

    def _accept_loop(self) -> None:
        while not self._closed.is_set():
            try:
                conn, _addr = self._listener.accept()
            except OSError:
                return
            threading.Thread(target=self._serve, args=(conn,), daemon=True).start()

    def _serve(self, conn: socket.socket) -> None:
        try:
            upstream = socket.create_connection(self.target, timeout=10)
        except OSError:
            conn.close()
            return
        self.connections_served += 1

        def pump(src: socket.socket, dst: socket.socket) -> None:
            try:
                while True:
                    data = src.recv(65536)
                    if not data:
                        break
                    dst.sendall(data)
            except OSError:
                pass
            finally:
                for s in (src, dst):
                    try:
                        s.shutdown(socket.SHUT_RDWR)
                    except OSError:
                        pass

        t1 = threading.Thread(target=pump, args=(conn, upstream), daemon=True)
        t2 = threading.Thread(target=pump, args=(upstream, conn), daemon=True)
        t1.start()
        t2.start()
        t1.join()
        t2.join()
        for s in (conn, upstream):
            try:
                s.close()
            except OSError:
                pass

    def close(self) -> None:
        self._closed.set()
        # order matters: shutdown() unblocks the accept() WITHOUT freeing
        # the fd (closing first lets the number be reused by an unrelated
        # socket while the thread is still blocked on it); join, then close
        try:
            self._listener.shutdown(socket.SHUT_RDWR)
        except OSError:
            pass
        self._accept_thread.join(timeout=2)
        try:
            self._listener.close()
        except OSError:
            pass


@contextlib.contextmanager
def forward(port: int, *, unencrypted: bool = False, client: Any = None) -> Iterator[Tunnel]:
    """Expose a port through a relay listener on a fresh local port."""
    relay = _TcpRelay("127.0.0.1", port)
    try:
        yield Tunnel(
            host="127.0.0.1",
            port=relay.port,
            unencrypted_host="127.0.0.1" if unencrypted else "",
            unencrypted_port=relay.port if unencrypted else 0,
            _relay=relay,
        )
    finally:
        relay.close()
