"""Object ID scheme.

Parity: every resource is keyed by a short ID prefix
(/root/reference/py/modal/_object.py:101, _get_type_from_id :229):
``fu-`` Function, ``fc-`` FunctionCall, ``im-`` Image, ``sb-`` Sandbox,
``vo-`` Volume, ``qu-`` Queue, ``di-`` Dict, ``st-`` Secret, ``mo-`` Mount,
``cs-`` Cls, ``sn-`` SandboxSnapshot, plus ``ap-`` App, ``in-`` input,
``ta-`` task (worker), ``en-`` environment.
"""

from __future__ import annotations

import secrets

ID_PREFIXES = {
    "app": "ap",
    "function": "fu",
    "function_call": "fc",
    "image": "im",
    "sandbox": "sb",
    "volume": "vo",
    "queue": "qu",
    "dict": "di",
    "secret": "st",
    "mount": "mo",
    "cls": "cs",
    "sandbox_snapshot": "sn",
    "input": "in",
    "task": "ta",
    "environment": "en",
    "blob": "bl",
    "server": "sr",
    "tunnel": "tn",
    "nfs": "sv",
}

_PREFIX_TO_TYPE = {v: k for k, v in ID_PREFIXES.items()}


def new_id(kind: str) -> str:
    return f"{ID_PREFIXES[kind]}-{secrets.token_hex(8)}"


def id_type(object_id: str) -> str:
    prefix = object_id.split("-", 1)[0]
    try:
        return _PREFIX_TO_TYPE[prefix]
    except KeyError:
        raise ValueError(f"Unknown object id prefix: {object_id!r}") from None


def is_id(value: str, kind: str) -> bool:
    return isinstance(value, str) and value.startswith(ID_PREFIXES[kind] + "-")
