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


def mount_spec(spec: Any) -> tuple[str, bool, str]:
    """Normalize a mount value: volume_id str, or
    {"volume_id", "read_only", "sub_path"} (Volume.with_mount_options)."""
    if isinstance(spec, dict):
        return (
            spec["volume_id"],
            bool(spec.get("read_only")),
            (spec.get("sub_path") or "").strip("/"),
        )
    return spec, False, ""


def mount_volumes(mounts: dict[str, Any], runtime: Any) -> None:
    """mounts: {mount_path: spec}; create symlinks into the shared tree."""
    run_dir = os.path.dirname(runtime.socket_path)
    for mount_path, spec in mounts.items():
        volume_id, _ro, sub_path = mount_spec(spec)
        target = volume_root(run_dir, volume_id)
        if sub_path:
            target = os.path.join(target, sub_path)
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
