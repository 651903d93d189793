"""Tunnels: expose a local port (parity: /root/reference/py/modal/_tunnel.py:18,61).

On a single node there is no relay: ``forward(port)`` yields a Tunnel whose
URL points at 127.0.0.1 — the port is already reachable. The API shape
(``Tunnel.url/.tls_socket/.tcp_socket``, ``forward`` context manager) matches
the reference so code moves over unchanged.
"""

from __future__ import annotations

import contextlib
from dataclasses import dataclass
from typing import Any, Iterator


@dataclass(frozen=True)
class Tunnel:
    host: str
    port: int
    unencrypted_host: str = ""
    unencrypted_port: int = 0

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


@contextlib.contextmanager
def forward(port: int, *, unencrypted: bool = False, client: Any = None) -> Iterator[Tunnel]:
    """Expose a port: locally an identity mapping on 127.0.0.1."""
    yield Tunnel(
        host="127.0.0.1",
        port=port,
        unencrypted_host="127.0.0.1" if unencrypted else "",
        unencrypted_port=port if unencrypted else 0,
    )
