"""Secrets: named environment-variable bundles.

Parity: /root/reference/py/modal/secret.py — ``_Secret`` (:234),
``from_dict`` (:276), ``from_dotenv`` (:341), ``from_name``; applied to the
worker env when a function that references them loads
(scheduler/core.py:resolve_function_env).
"""

from __future__ import annotations

import os
from typing import Any, Optional

from ._object import _Object, live_method
from ._sync import synchronize_api
from .exception import InvalidError


class _Secret(_Object, type_kind="secret"):
    @classmethod
    def from_dict(cls, env_dict: Optional[dict[str, str]] = None) -> "_Secret":
        env_dict = env_dict or {}
        for key, value in env_dict.items():
            if value is not None and not isinstance(value, str):
                raise InvalidError(f"Secret values must be strings (key {key!r})")

        async def _load(obj: "_Secret", resolver: Any, existing: Any) -> None:
            sid = await resolver.client.svc.secret_get_or_create(
                name=None, environment=resolver.environment_name or "main", env=env_dict
            )
            obj._hydrate(sid, resolver.client, None)

        return cls._from_loader(_load, rep=f"Secret.from_dict([{', '.join(env_dict)}])")

    @classmethod
    def from_local_environ(cls, env_keys: list[str]) -> "_Secret":
        env_dict = {}
        for key in env_keys:
            if key not in os.environ:
                raise InvalidError(f"Environment variable {key!r} is not set locally")
            env_dict[key] = os.environ[key]
        return cls.from_dict(env_dict)

    @classmethod
    def from_dotenv(cls, path: Optional[str] = None, *, filename: str = ".env") -> "_Secret":
        import inspect

        if path is None:
            caller = inspect.stack()[1]
            path = os.path.dirname(os.path.abspath(caller.filename))
        if os.path.isdir(path):
            path = os.path.join(path, filename)
        env_dict: dict[str, str] = {}
        if os.path.exists(path):
            with open(path) as f:
                for line in f:
                    line = line.strip()
                    if not line or line.startswith("#") or "=" not in line:
                        continue
                    key, _, value = line.partition("=")
                    env_dict[key.strip()] = value.strip().strip("'\"")
        return cls.from_dict(env_dict)

    @classmethod
    def from_name(
        cls,
        name: str,
        *,
        environment_name: str = "",
        required_keys: Optional[list[str]] = None,
    ) -> "_Secret":
        async def _load(obj: "_Secret", resolver: Any, existing: Any) -> None:
            sid = await resolver.client.svc.secret_get_or_create(
                name=name,
                environment=environment_name or resolver.environment_name or "main",
                env=None,
                required_keys=required_keys or [],
            )
            obj._hydrate(sid, resolver.client, {"name": name})

        return cls._from_loader(_load, rep=f"Secret.from_name({name!r})")

    @property
    def name(self) -> Any:
        return (getattr(self, "_metadata", None) or {}).get("name")

    @live_method
    async def info(self) -> dict:
        """Name + key names — never values (parity: reference info())."""
        return await self._client.svc.object_info(object_id=self.object_id)

    @live_method
    async def update(self, env_dict: dict) -> None:
        """Merge new entries into the stored secret (parity: Secret.update)."""
        await self._client.svc.secret_update(secret_id=self.object_id, env=dict(env_dict))

    @classmethod
    async def create_deployed(
        cls,
        deployment_name: str,
        env_dict: dict[str, str],
        *,
        environment_name: str = "",
        overwrite: bool = False,
    ) -> str:
        from .client import _Client

        client = await _Client.from_env()
        sid = await client.svc.secret_get_or_create(
            name=deployment_name,
            environment=environment_name or "main",
            env=env_dict,
            overwrite=overwrite,
        )
        return sid

    @live_method
    async def env(self) -> dict[str, str]:
        """Inspect the env bundle (local extension; handy for tests/CLI)."""
        return await self._client.svc.secret_env(secret_id=self.object_id)


Secret = synchronize_api(_Secret, "Secret")

from .object_manager import install as _install_manager  # noqa: E402

_install_manager(_Secret, Secret, "secret")
