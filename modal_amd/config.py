"""Layered configuration: defaults -> ~/.modal_amd.toml profile -> MODAL_* env.

Parity with the reference config system (/root/reference/py/modal/config.py:301-344):
typed settings with transforms, profile switching, env-var override (env wins),
and ``config.override_locally`` for programmatic overrides (used by snapshot
restore). Env vars accept both ``MODAL_AMD_<KEY>`` and ``MODAL_<KEY>`` so code
written for the reference keeps working.
"""

from __future__ import annotations

import os
import threading
from typing import Any, Callable, Optional

try:
    import tomli as _toml
except ImportError:  # pragma: no cover
    _toml = None

from .exception import InvalidError

USER_CONFIG_PATH = os.environ.get("MODAL_AMD_CONFIG_PATH", os.path.expanduser("~/.modal_amd.toml"))


def _to_boolean(value: Any) -> bool:
    return str(value).lower() not in ("", "0", "false", "no", "none")


class _Setting:
    def __init__(self, default: Any = None, transform: Callable[[str], Any] = lambda x: x):
        self.default = default
        self.transform = transform


_SETTINGS: dict[str, _Setting] = {
    "profile": _Setting("default"),
    "loglevel": _Setting("WARNING", lambda s: s.upper()),
    "log_format": _Setting("STRING", lambda s: s.upper()),
    # scheduler / runtime
    "run_dir": _Setting(None),  # scratch dir for sockets, CAS, logs (default: tmpdir)
    "worker_count": _Setting(None, lambda s: int(s) if s else None),  # default: #GPUs or 1
    "gpu_workers": _Setting(None, lambda s: _to_boolean(s) if s != "" else None),
    "heartbeat_interval": _Setting(15.0, float),  # parity: config.py:318
    "function_call_timeout": _Setting(None, lambda s: float(s) if s else None),
    "sync_client_retries_enabled": _Setting(True, _to_boolean),
    # serialization / data plane
    "payload_format": _Setting("pickle"),  # pickle | cbor (parity: config.py:336)
    "max_inline_payload": _Setting(2 * 1024 * 1024, int),  # parity: blob_utils.py:36
    "blob_dir": _Setting(None),  # content-addressed store root
    "gpu_hash_threshold": _Setting(8 * 1024 * 1024, int),  # below this, CPU hashes
    # ux
    "traceback": _Setting(False, _to_boolean),
    "automount": _Setting(True, _to_boolean),
    "async_warnings": _Setting(True, _to_boolean),
    "image_builder_version": _Setting("LOCAL.1"),
    "strict_parameters": _Setting(False, _to_boolean),
    "snapshot_debug": _Setting(False, _to_boolean),
    # telemetry
    "telemetry_socket": _Setting(None),
    "runtime_perf_record": _Setting(False, _to_boolean),
    # identity (kept for API parity; unused by the local control plane)
    "token_id": _Setting(None),
    "token_secret": _Setting(None),
    "workspace": _Setting("local"),
    "environment": _Setting(None),
    "server_url": _Setting(None),
}


def _read_user_config() -> dict[str, dict[str, Any]]:
    if _toml is None or not os.path.exists(USER_CONFIG_PATH):
        return {}
    try:
        with open(USER_CONFIG_PATH, "rb") as f:
            return _toml.load(f)
    except Exception:
        return {}


_user_config = _read_user_config()


def config_profiles() -> list[str]:
    return list(_user_config.keys())


def _config_active_profile() -> str:
    env = os.environ.get("MODAL_AMD_PROFILE") or os.environ.get("MODAL_PROFILE")
    if env:
        return env
    for name, section in _user_config.items():
        if isinstance(section, dict) and section.get("active"):
            return name
    return "default"


class Config:
    """Read-only view over the layered settings with local override support."""

    def __init__(self) -> None:
        self._local_overrides: dict[str, Any] = {}
        self._lock = threading.Lock()

    def get(self, key: str, profile: Optional[str] = None, use_env: bool = True) -> Any:
        if key not in _SETTINGS:
            raise InvalidError(f"Unknown config key: {key}")
        s = _SETTINGS[key]
        with self._lock:
            if key in self._local_overrides:
                return self._local_overrides[key]
        if use_env:
            for env_key in (f"MODAL_AMD_{key.upper()}", f"MODAL_{key.upper()}"):
                if env_key in os.environ:
                    return s.transform(os.environ[env_key])
        profile = profile or _config_active_profile()
        section = _user_config.get(profile, {})
        if isinstance(section, dict) and key in section:
            raw = section[key]
            return s.transform(raw) if isinstance(raw, str) else raw
        return s.default

    def __getitem__(self, key: str) -> Any:
        return self.get(key)

    def override_locally(self, key: str, value: Any) -> None:
        """Programmatic override (parity: config.override_locally, used by restore)."""
        with self._lock:
            self._local_overrides[key] = value

    def clear_override(self, key: str) -> None:
        with self._lock:
            self._local_overrides.pop(key, None)

    def to_dict(self) -> dict[str, Any]:
        return {k: self.get(k) for k in _SETTINGS}


config = Config()


def reload_user_config() -> None:
    global _user_config
    _user_config = _read_user_config()
