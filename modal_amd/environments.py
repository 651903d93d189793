"""Environments: namespaces for deployed objects.

Parity: /root/reference/py/modal/_environments.py:637 — environments scope
deployed names; the default is "main". Locally they are plain namespace keys
in the scheduler's name tables.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from ._sync import dual_function


@dataclass
class Environment:
    name: str
    webhook_suffix: str = ""


_ENVIRONMENTS: dict[str, Environment] = {"main": Environment("main")}


@dual_function
async def create_environment(name: str) -> Environment:
    env = Environment(name)
    _ENVIRONMENTS[name] = env
    return env


@dual_function
async def delete_environment(name: str) -> None:
    _ENVIRONMENTS.pop(name, None)


@dual_function
async def list_environments() -> list[Environment]:
    return list(_ENVIRONMENTS.values())


def ensure_env(name: Optional[str] = None) -> str:
    return name or "main"
