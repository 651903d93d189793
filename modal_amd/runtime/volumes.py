"""Worker-side volume mounting.

The reference mounts committed Volumes into the container filesystem; the
single-node equivalent materializes each volume as a directory under the run
dir (the volume service keeps the authoritative tree there) and exposes it at
the requested mount path via a symlink. Commit/reload become no-ops on the
data (same filesystem) but keep API semantics.
"""

from __future__ import annotations

import os
from typing import Any


def volume_root(run_dir: str, volume_id: str) -> str:
    return os.path.join(run_dir, "volumes", volume_id)


def mount_volumes(mounts: dict[str, str], runtime: Any) -> None:
    """mounts: {mount_path: volume_id}; create symlinks into the shared tree."""
    run_dir = os.path.dirname(runtime.socket_path)
    for mount_path, volume_id in mounts.items():
        target = volume_root(run_dir, volume_id)
        os.makedirs(target, exist_ok=True)
        if os.path.islink(mount_path):
            if os.readlink(mount_path) == target:
                continue
            os.unlink(mount_path)
        elif os.path.exists(mount_path):
            continue  # refuse to clobber an existing path
        parent = os.path.dirname(mount_path)
        try:
            os.makedirs(parent, exist_ok=True)
            os.symlink(target, mount_path)
        except OSError:
            pass  # unwritable mount point: volume still reachable via the API
