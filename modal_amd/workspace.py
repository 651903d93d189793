"""Workspace: account-level metadata (parity: /root/reference/py/modal/_workspace.py:532).

Single-node: one implicit workspace named by config, no auth.
"""

from __future__ import annotations

from dataclasses import dataclass

from .config import config


@dataclass
class Workspace:
    name: str
    username: str = "local"

    @classmethod
    def current(cls) -> "Workspace":
        return cls(name=config.get("workspace") or "local")
