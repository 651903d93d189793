"""Image: layered environment recipes (the Dockerfile-DSL surface).

Parity: /root/reference/py/modal/_image.py — ``_Image`` (:488), layer chain
``_from_args`` (:592), DSL ``pip_install`` (:1100), ``uv_pip_install``
(:1454), ``from_registry`` (:2203), ``from_dockerfile`` (:2400),
``micromamba`` (:2055), ``run_commands``, ``env``, ``workdir``,
``add_local_file/dir/python_source`` (:542-590), ``run_function`` build-time
exec. Locally an image is a hashed recipe built once into a cached directory
(scheduler/images.py); identical recipes share one build (content-addressed,
like the reference's layer dedup).
"""

from __future__ import annotations

import os
from typing import Any, Callable, Optional, Sequence, Union

from ._object import _Object, live_method
from ._sync import synchronize_api
from .exception import InvalidError


def _flatten_pkgs(packages: tuple) -> list[str]:
    out: list[str] = []
    for p in packages:
        if isinstance(p, (list, tuple)):
            out.extend(p)
        else:
            out.append(p)
    return out


class _Image(_Object, type_kind="image"):
    _recipe: list

    def _init_attrs(self) -> None:
        self._recipe = []
        self._env_cache: dict = {}
        self._python_paths: list[str] = []
        self._workdir: Optional[str] = None

    # -- construction ----------------------------------------------------
    @classmethod
    def _from_recipe(cls, recipe: list) -> "_Image":
        async def _load(obj: "_Image", resolver: Any, existing: Any) -> None:
            # stage added local files into the CAS before the build sees them
            for layer in obj._recipe:
                if layer.get("kind") == "add_local_file" and layer.get("local_path"):
                    with open(layer["local_path"], "rb") as f:
                        put = await resolver.client.svc.blob_put(data=f.read())
                    layer["blob_id"] = put["blob_id"]
            resp = await resolver.client.svc.image_get_or_create(recipe=obj._recipe)
            obj._hydrate(
                resp["image_id"],
                resolver.client,
                {
                    "env": resp.get("env"),
                    "workdir": resp.get("workdir"),
                    "python_paths": resp.get("python_paths"),
                },
            )

        obj = cls._from_loader(_load, rep=f"Image({len(recipe)} layers)")
        obj._recipe = recipe
        return obj

    def _extend(self, layer: dict) -> "_Image":
        return _Image._from_recipe([*self._recipe, layer])

    def _hydrate_metadata(self, metadata: dict) -> None:
        if metadata:
            self._env_cache = metadata.get("env") or {}
            self._python_paths = metadata.get("python_paths") or []
            self._workdir = metadata.get("workdir")

    # -- bases -----------------------------------------------------------
    @classmethod
    def debian_slim(cls, python_version: Optional[str] = None) -> "_Image":
        """The standard base. Locally: this node's interpreter/rootfs."""
        return cls._from_recipe(
            [{"kind": "base", "name": "debian-slim", "python_version": python_version}]
        )

    @classmethod
    def micromamba(cls, python_version: Optional[str] = None) -> "_Image":
        return cls._from_recipe(
            [{"kind": "base", "name": "micromamba", "python_version": python_version}]
        )

    @classmethod
    def from_registry(
        cls, tag: str, *, secret: Any = None, add_python: Optional[str] = None, **_: Any
    ) -> "_Image":
        return cls._from_recipe([{"kind": "base", "name": f"registry:{tag}", "add_python": add_python}])

    @classmethod
    def from_aws_ecr(cls, tag: str, secret: Any = None, **kwargs: Any) -> "_Image":
        return cls.from_registry(tag, secret=secret, **kwargs)

    @classmethod
    def from_gcp_artifact_registry(cls, tag: str, secret: Any = None, **kwargs: Any) -> "_Image":
        return cls.from_registry(tag, secret=secret, **kwargs)

    @classmethod
    def from_dockerfile(cls, path: Union[str, os.PathLike], **_: Any) -> "_Image":
        with open(path) as f:
            commands = [ln.rstrip() for ln in f if ln.strip() and not ln.strip().startswith("#")]
        return cls._from_recipe(
            [{"kind": "base", "name": "dockerfile"}, {"kind": "dockerfile_commands", "commands": commands}]
        )

    @classmethod
    def from_scratch(cls) -> "_Image":
        """An empty base image (parity: reference Image building blocks)."""
        return cls._from_recipe([{"kind": "base", "name": "scratch"}])

    @classmethod
    def from_name(cls, name: str, *, environment_name: str = "") -> "_Image":
        """Reference an Image previously published with .publish(); names may
        carry a ':tag' (':latest' implied) — parity: reference
        _image.py:2972 from_name."""

        async def _load(obj: "_Image", resolver: Any, existing: Any) -> None:
            resp = await resolver.client.svc.image_from_name(
                name=name, environment=environment_name
            )
            obj._hydrate(resp["image_id"], resolver.client, None)

        return cls._from_loader(_load, rep=f"Image.from_name({name!r})")

    @classmethod
    def from_id(cls, image_id: str, client: Any = None) -> "_Image":
        async def _load(obj: "_Image", resolver: Any, existing: Any) -> None:
            resp = await resolver.client.svc.image_info(image_id=image_id)
            obj._hydrate(image_id, resolver.client, resp)

        return cls._from_loader(_load, rep=f"Image.from_id({image_id})")

    # -- layers ----------------------------------------------------------
    def pip_install(
        self,
        *packages: Union[str, list[str]],
        find_links: Optional[str] = None,
        index_url: Optional[str] = None,
        extra_index_url: Optional[str] = None,
        pre: bool = False,
        extra_options: str = "",
        gpu: Any = None,
        secrets: Sequence[Any] = (),
        force_build: bool = False,
    ) -> "_Image":
        pkgs = _flatten_pkgs(packages)
        if not pkgs:
            return self
        return self._extend({"kind": "pip_install", "packages": sorted(pkgs), "find_links": find_links})

    def uv_pip_install(self, *packages: Union[str, list[str]], **kwargs: Any) -> "_Image":
        return self.pip_install(*packages, **{k: v for k, v in kwargs.items() if k == "find_links"})

    def pip_install_from_requirements(self, requirements_txt: str, **kwargs: Any) -> "_Image":
        with open(requirements_txt) as f:
            pkgs = [ln.strip() for ln in f if ln.strip() and not ln.startswith("#")]
        return self.pip_install(*pkgs, **kwargs)

    def uv_sync(self, uv_project_dir: str = "./", **kwargs: Any) -> "_Image":
        return self._extend({"kind": "run_commands", "commands": [f"cd {uv_project_dir} && uv sync --offline || true"]})

    def pip_install_from_pyproject(
        self, pyproject_toml: str, optional_dependencies: list[str] = [], **kwargs: Any
    ) -> "_Image":
        """Parity: reference image.py Image.pip_install_from_pyproject —
        record the pyproject's dependency list as a pip layer."""
        return self._extend(
            {
                "kind": "pip_install_from_pyproject",
                "path": str(pyproject_toml),
                "optional_dependencies": list(optional_dependencies),
                **{k: v for k, v in kwargs.items() if v},
            }
        )

    def poetry_install_from_file(self, poetry_pyproject_toml: str, **kwargs: Any) -> "_Image":
        return self._extend({"kind": "run_commands", "commands": ["poetry install || true"]})

    def apt_install(self, *packages: Union[str, list[str]], **kwargs: Any) -> "_Image":
        return self._extend({"kind": "apt_install", "packages": sorted(_flatten_pkgs(packages))})

    def micromamba_install(self, *packages: Union[str, list[str]], **kwargs: Any) -> "_Image":
        return self._extend({"kind": "micromamba_install", "packages": sorted(_flatten_pkgs(packages))})

    def run_commands(self, *commands: Union[str, list[str]], secrets: Sequence[Any] = (), gpu: Any = None, force_build: bool = False) -> "_Image":
        cmds = _flatten_pkgs(commands)
        return self._extend({"kind": "run_commands", "commands": cmds})

    def dockerfile_commands(self, *commands: Union[str, list[str]], **kwargs: Any) -> "_Image":
        return self._extend({"kind": "dockerfile_commands", "commands": _flatten_pkgs(commands)})

    def env(self, vars: dict[str, str]) -> "_Image":  # noqa: A002 - parity
        return self._extend({"kind": "env", "vars": dict(vars)})

    def workdir(self, path: Union[str, os.PathLike]) -> "_Image":
        return self._extend({"kind": "workdir", "path": str(path)})

    def entrypoint(self, entrypoint_commands: list[str]) -> "_Image":
        return self._extend({"kind": "entrypoint", "args": list(entrypoint_commands)})

    def cmd(self, cmd: list[str]) -> "_Image":
        return self._extend({"kind": "cmd", "args": list(cmd)})

    def shell(self, shell_commands: list[str]) -> "_Image":
        """Overwrite the default shell used by later run_commands
        (parity: reference _image.py:1990)."""
        return self._extend({"kind": "shell", "args": list(shell_commands)})

    def add_local_file(self, local_path: Union[str, os.PathLike], remote_path: str, *, copy: bool = False) -> "_Image":
        import hashlib

        with open(local_path, "rb") as f:
            digest = hashlib.sha256(f.read()).hexdigest()
        return self._extend(
            {"kind": "add_local_file", "remote_path": str(remote_path), "blob_id": digest,
             "local_path": str(local_path)}
        )

    def add_local_dir(
        self, local_path: Union[str, os.PathLike], remote_path: str, *, copy: bool = False, ignore: Any = None
    ) -> "_Image":
        img = self
        for dirpath, _dn, filenames in os.walk(local_path):
            for fn in filenames:
                full = os.path.join(dirpath, fn)
                rel = os.path.relpath(full, local_path)
                img = img.add_local_file(full, os.path.join(remote_path, rel))
        return img

    def add_local_python_source(self, *modules: str, copy: bool = False, ignore: Any = None) -> "_Image":
        import importlib.util

        img = self
        for mod in modules:
            spec = importlib.util.find_spec(mod)
            if spec is None or not spec.origin:
                raise InvalidError(f"Can't find local module {mod!r}")
            if spec.submodule_search_locations:
                img = img.add_local_dir(os.path.dirname(spec.origin), f"/pysource/{mod}")
            else:
                img = img.add_local_file(spec.origin, f"/pysource/{mod}.py")
        return img

    # deprecated aliases kept for API parity
    copy_local_file = add_local_file
    copy_local_dir = add_local_dir

    def run_function(self, raw_f: Callable, *args: Any, **kwargs: Any) -> "_Image":
        """Build-time function execution (reference run_function).
        Recorded as a layer; the runner executes it against the scheduler on
        first build."""
        import cloudpickle

        payload = cloudpickle.dumps((raw_f, args, kwargs)).hex()
        return self._extend({"kind": "run_function", "payload_hex": payload, "name": getattr(raw_f, "__name__", "f")})

    # -- introspection ----------------------------------------------------
    @live_method
    async def build_log(self) -> str:
        info = await self._client.svc.image_info(image_id=self.object_id)
        return info.get("build_log", "")

    def pipe(self, func: Callable, *args: Any, **kwargs: Any) -> "_Image":
        """Apply a local recipe-expanding function: func(image, *args)
        (parity: reference _image.py:2889)."""
        from ._sync import unwrap as _unwrap, wrap as _wrap

        return _unwrap(func(_wrap(self), *args, **kwargs))

    def pip_install_private_repos(
        self, *repositories: str, git_user: str = "", **kwargs: Any
    ) -> "_Image":
        """Recorded layer (no egress here — like registry pulls, the clone
        runs when connectivity exists); parity: _image.py:1187."""
        return self._extend(
            {
                "kind": "pip_install_private_repos",
                "repositories": list(repositories),
                "git_user": git_user,
            }
        )

    async def publish(
        self, name: str, *, environment_name: str = "", client: Any = None
    ) -> None:
        """Publish this built image under a workspace name (parity:
        reference _image.py:3010)."""
        await self.hydrate(client)
        await self._client.svc.image_publish(
            image_id=self.object_id, name=name, environment=environment_name
        )

    async def build(self, client: Any = None) -> "_Image":
        """Eagerly build this image (parity: reference Image.build — normally
        builds happen lazily at app start; this forces hydration now)."""
        await self.hydrate(client)
        return self

    @live_method
    async def logs(self) -> "Any":
        """Yield build-log lines (parity: reference Image.logs stream)."""
        info = await self._client.svc.image_info(image_id=self.object_id)
        return (info.get("build_log", "") or "").splitlines(keepends=True)

    def imports(self) -> Any:
        """Context manager that suppresses ImportError locally
        (parity: reference Image.imports)."""
        import contextlib

        @contextlib.contextmanager
        def ctx() -> Any:
            try:
                yield
            except ImportError as exc:
                import warnings

                warnings.warn(f"Deferred import failure (ok inside image): {exc}")

        return ctx()


Image = synchronize_api(_Image, "Image")
