"""Import-reference parsing for CLI commands.

Parity: /root/reference/py/modal/cli/import_refs.py:1-401 — references of the
form ``file.py``, ``file.py::app``, ``file.py::app.function_name``,
``module.path::app`` resolve to an App and optionally a function or local
entrypoint.
"""

from __future__ import annotations

import importlib
import importlib.util
import os
import sys
from dataclasses import dataclass
from typing import Any, Optional

from ..app import App
from ..exception import InvalidError


@dataclass
class ImportRef:
    file_or_module: str
    object_path: str  # "" | "app" | "app.fn"


def parse_import_ref(ref: str) -> ImportRef:
    if "::" in ref:
        file_or_module, object_path = ref.split("::", 1)
    else:
        file_or_module, object_path = ref, ""
    return ImportRef(file_or_module, object_path)


def import_target(ref: ImportRef) -> Any:
    """Import the module of a reference (file path or dotted module)."""
    if ref.file_or_module.endswith(".py") or os.path.sep in ref.file_or_module:
        path = os.path.abspath(ref.file_or_module)
        if not os.path.exists(path):
            raise InvalidError(f"No such file: {ref.file_or_module}")
        module_name = os.path.splitext(os.path.basename(path))[0]
        sys.path.insert(0, os.path.dirname(path))
        spec = importlib.util.spec_from_file_location(module_name, path)
        module = importlib.util.module_from_spec(spec)
        sys.modules[module_name] = module
        spec.loader.exec_module(module)
        return module
    return importlib.import_module(ref.file_or_module)


def find_app(module: Any, object_path: str) -> App:
    first = object_path.split(".", 1)[0] if object_path else ""
    if first:
        obj = getattr(module, first, None)
        if isinstance(obj, App):
            return obj
        raise InvalidError(f"{first!r} in {module.__name__} is not an App")
    apps = [v for v in vars(module).values() if isinstance(v, App)]
    if len(apps) == 1:
        return apps[0]
    if not apps:
        raise InvalidError(f"No App found in {module.__name__}")
    named = [a for a in apps if a.name]
    if len(named) == 1:
        return named[0]
    raise InvalidError(
        f"Multiple Apps in {module.__name__}; use ::app_variable to disambiguate"
    )


def find_callable(module: Any, app: App, object_path: str) -> Optional[Any]:
    """Resolve the function / local entrypoint named by the reference."""
    parts = object_path.split(".") if object_path else []
    fn_name = parts[1] if len(parts) > 1 else (parts[0] if parts and not isinstance(getattr(module, parts[0], None), App) else None)
    if fn_name is None:
        # default: sole local entrypoint, else sole function
        if len(app.registered_entrypoints) == 1:
            return next(iter(app.registered_entrypoints.values()))
        if len(app.registered_functions) == 1:
            return next(iter(app.registered_functions.values()))
        if not app.registered_entrypoints and not app.registered_functions:
            raise InvalidError("App has no functions or entrypoints")
        raise InvalidError(
            "App has multiple functions; specify one with ::app.function_name"
        )
    if fn_name in app.registered_entrypoints:
        return app.registered_entrypoints[fn_name]
    if fn_name in app.registered_functions:
        return app.registered_functions[fn_name]
    raise InvalidError(f"No function or entrypoint named {fn_name!r} on the app")
