"""Image build service: hashed layer recipes -> cached local environments.

Single-node re-implementation of the reference's server-side image builder
(/root/reference/py/modal/_image.py:433-466 ImageGetOrCreate +
ImageJoinStreaming): an image is an ordered recipe of layers; its identity is
the SHA-256 of the canonical recipe (content-addressed like the reference's
layer dedup); building materializes a directory with
  env.json         (env vars, workdir, entrypoint)
  site-packages/   (pip_install --target, when wheels are available offline)
  files/           (added local files)
  build.log
Workers and sandboxes apply an image by extending PYTHONPATH/env.
"""

from __future__ import annotations

import hashlib
import json
import os
import subprocess
import sys
import time
from typing import Any, Optional

from ..exception import ExecutionError, NotFoundError
from ..utils.ids import new_id


class ImageState:
    def __init__(self, image_id: str, recipe_hash: str, root: str):
        self.image_id = image_id
        self.recipe_hash = recipe_hash
        self.root = root
        self.built = False
        self.build_log = ""
        self.env: dict[str, str] = {}
        self.workdir: Optional[str] = None
        self.entrypoint: list[str] = []
        self.cmd: list[str] = []
        self.python_paths: list[str] = []

    @property
    def fsroot(self) -> str:
        """The image's filesystem layer: absolute-path writes made by
        run_commands land here (overlay upper); sandboxes on this image
        stack it as an extra overlay lower (parity: the reference's
        server-built layered filesystems, _image.py:592)."""
        return os.path.join(self.root, "fsdiff")


def recipe_hash(recipe: list[dict]) -> str:
    canon = json.dumps(recipe, sort_keys=True, separators=(",", ":"))
    return hashlib.sha256(canon.encode()).hexdigest()


class ImageService:
    def __init__(self, run_dir: str, blob_store: Any):
        self.root = os.path.join(run_dir, "images")
        os.makedirs(self.root, exist_ok=True)
        self.blob_store = blob_store
        self.by_hash: dict[str, ImageState] = {}
        self.by_id: dict[str, ImageState] = {}

    async def get_or_create(self, recipe: list[dict], build: bool = True) -> dict:
        rh = recipe_hash(recipe)
        state = self.by_hash.get(rh)
        if state is None:
            image_id = new_id("image")
            root = os.path.join(self.root, image_id)
            os.makedirs(root, exist_ok=True)
            state = ImageState(image_id, rh, root)
            self.by_hash[rh] = state
            self.by_id[image_id] = state
            if build:
                await self._build(state, recipe)
        return {
            "image_id": state.image_id,
            "built": state.built,
            "env": state.env,
            "workdir": state.workdir,
            "python_paths": state.python_paths,
        }

    async def info(self, image_id: str) -> dict:
        state = self.by_id.get(image_id)
        if state is None:
            raise NotFoundError(f"Image {image_id} not found")
        return {
            "image_id": state.image_id,
            "built": state.built,
            "env": state.env,
            "workdir": state.workdir,
            "entrypoint": state.entrypoint,
            "python_paths": state.python_paths,
            "build_log": state.build_log[-4000:],
        }

    async def _build(self, state: ImageState, recipe: list[dict]) -> None:
        """Execute layers in order. Layer kinds mirror the reference DSL
        (_image.py:1100 pip_install, :2400 from_dockerfile, ...)."""
        import asyncio

        log: list[str] = []
        site = os.path.join(state.root, "site-packages")

        shell_argv: list = []  # Image.shell() override (SHELL json-array form)

        def run_shell(cmd: str, env: Optional[dict] = None, isolated: bool = False) -> None:
            full_env = dict(os.environ)
            full_env.update(state.env)
            if env:
                full_env.update(env)
            argv = [*shell_argv, cmd] if shell_argv else ["/bin/sh", "-c", cmd]
            if isolated:
                # run_commands execute INSIDE the image root: PID+mount
                # namespaces with an overlayfs whose upper is the image's
                # fs layer — absolute-path writes become image content
                from .isolation import isolation_argv

                wrapped = isolation_argv(
                    argv,
                    sandbox_dir=state.root,
                    run_dir=state.root,
                    workdir=state.workdir or state.root,
                )
                if wrapped is not None:
                    argv = wrapped[0]
            proc = subprocess.run(
                argv, cwd=state.root, env=full_env,
                capture_output=True, text=True, timeout=600,
            )
            log.append(f"$ {cmd}\n{proc.stdout}{proc.stderr}")
            if proc.returncode != 0:
                raise ExecutionError(f"Image build step failed ({cmd!r}):\n{proc.stderr[-2000:]}")

        loop = asyncio.get_running_loop()
        for layer in recipe:
            kind = layer.get("kind")
            if kind == "base":
                log.append(f"base: {layer.get('name', 'local')}")
            elif kind == "env":
                state.env.update(layer.get("vars", {}))
            elif kind == "workdir":
                state.workdir = layer["path"]
            elif kind == "entrypoint":
                state.entrypoint = list(layer.get("args", []))
            elif kind == "cmd":
                state.cmd = list(layer.get("args", []))
            elif kind == "shell":
                shell_argv[:] = layer.get("args", [])
            elif kind == "run_commands":
                for cmd in layer.get("commands", []):
                    await loop.run_in_executor(
                        None, lambda c=cmd: run_shell(c, isolated=True)
                    )
            elif kind == "pip_install":
                pkgs = layer.get("packages", [])
                if pkgs:
                    os.makedirs(site, exist_ok=True)
                    find_links = layer.get("find_links") or os.environ.get("MODAL_AMD_WHEELHOUSE")
                    flags = f"--no-index --find-links {find_links}" if find_links else "--no-index"
                    try:
                        await loop.run_in_executor(
                            None,
                            run_shell,
                            f"{sys.executable} -m pip install --target {site} {flags} "
                            + " ".join(f"'{p}'" for p in pkgs),
                        )
                    except ExecutionError:
                        # offline node: packages already importable from the base
                        # interpreter satisfy the layer; anything else surfaces at
                        # import time in the worker
                        missing = [p for p in pkgs if not _importable(p)]
                        if missing:
                            raise
                        log.append(f"pip_install satisfied by base interpreter: {pkgs}")
                    if os.path.isdir(site):
                        state.python_paths = [site]
            elif kind == "apt_install":
                log.append(f"apt_install recorded (no package manager offline): {layer.get('packages')}")
            elif kind == "micromamba_install":
                log.append(f"micromamba_install recorded: {layer.get('packages')}")
            elif kind == "add_local_file":
                dest_dir = os.path.join(state.root, "files")
                os.makedirs(dest_dir, exist_ok=True)
                blob_id = layer["blob_id"]
                dest = os.path.join(dest_dir, layer["remote_path"].lstrip("/"))
                os.makedirs(os.path.dirname(dest), exist_ok=True)
                with open(dest, "wb") as f:
                    f.write(self.blob_store.get(blob_id))
            elif kind == "dockerfile_commands":
                for cmd in layer.get("commands", []):
                    log.append(f"dockerfile: {cmd}")
                    stripped = cmd.strip()
                    if stripped.upper().startswith("RUN "):
                        await loop.run_in_executor(None, run_shell, stripped[4:])
                    elif stripped.upper().startswith("ENV "):
                        parts = stripped[4:].replace("=", " ").split()
                        for i in range(0, len(parts) - 1, 2):
                            state.env[parts[i]] = parts[i + 1]
                    elif stripped.upper().startswith("WORKDIR "):
                        state.workdir = stripped.split(None, 1)[1]
            elif kind == "run_function":
                # build-time function execution in a child process rooted in
                # the image dir (parity: reference Image.run_function builds)
                payload_hex = layer.get("payload_hex", "")
                script = (
                    "import cloudpickle, sys\n"
                    "fn, args, kwargs = cloudpickle.loads(bytes.fromhex(sys.argv[1]))\n"
                    "fn(*args, **kwargs)\n"
                )

                def run_fn_layer() -> None:
                    full_env = dict(os.environ)
                    full_env.update(state.env)
                    proc = subprocess.run(
                        [sys.executable, "-c", script, payload_hex],
                        cwd=state.root, env=full_env, capture_output=True, text=True,
                        timeout=600,
                    )
                    log.append(f"run_function {layer.get('name')}:\n{proc.stdout}{proc.stderr}")
                    if proc.returncode != 0:
                        raise ExecutionError(
                            f"Image.run_function({layer.get('name')}) failed:\n{proc.stderr[-2000:]}"
                        )

                await loop.run_in_executor(None, run_fn_layer)
            else:
                log.append(f"unknown layer kind {kind!r} (ignored)")
        state.build_log = "\n".join(log)
        with open(os.path.join(state.root, "build.log"), "w") as f:
            f.write(state.build_log)
        with open(os.path.join(state.root, "env.json"), "w") as f:
            json.dump({"env": state.env, "workdir": state.workdir}, f)
        state.built = True


def _importable(spec: str) -> bool:
    import importlib.util
    import re

    name = re.split(r"[<>=!\[~;]", spec, 1)[0].strip().replace("-", "_")
    try:
        return importlib.util.find_spec(name) is not None
    except (ImportError, ValueError, ModuleNotFoundError):
        return False
