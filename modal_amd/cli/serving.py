"""``modal-amd serve``: run an app and redeploy on source change.

Parity: /root/reference/py/modal/serving.py:92 (_serve_app) + _watcher.py —
the reference uses watchfiles; offline we poll mtimes (0.5 s cadence).
"""

from __future__ import annotations

import os
import sys
import time
from typing import Any, Optional


def _watched_files(module: Any) -> dict[str, float]:
    files = {}
    path = getattr(module, "__file__", None)
    if path and os.path.exists(path):
        files[path] = os.path.getmtime(path)
    base_dir = os.path.dirname(path) if path else "."
    for dirpath, _dn, filenames in os.walk(base_dir):
        if "__pycache__" in dirpath:
            continue
        for fn in filenames:
            if fn.endswith(".py"):
                full = os.path.join(dirpath, fn)
                try:
                    files[full] = os.path.getmtime(full)
                except OSError:
                    pass
    return files


def serve_app(import_ref: Any, timeout: Optional[float] = None) -> None:
    from ..output import enable_output
    from .import_refs import find_app, import_target

    deadline = None if timeout is None else time.time() + timeout
    with enable_output():
        while True:
            module = import_target(import_ref)
            app = find_app(module, import_ref.object_path)
            ctx = app.run()
            ctx.__enter__()
            print(f"Serving app {app.name or app.app_id} (ctrl-c to stop)...")
            watched = _watched_files(module)
            try:
                while True:
                    time.sleep(0.5)
                    if deadline is not None and time.time() > deadline:
                        ctx.__exit__(None, None, None)
                        return
                    changed = False
                    for path, mtime in watched.items():
                        try:
                            if os.path.getmtime(path) != mtime:
                                changed = True
                                break
                        except OSError:
                            changed = True
                            break
                    if changed:
                        print("Change detected, reloading...")
                        break
            except KeyboardInterrupt:
                ctx.__exit__(None, None, None)
                return
            ctx.__exit__(None, None, None)
            # reload modules touched by the app file
            name = module.__name__
            sys.modules.pop(name, None)
