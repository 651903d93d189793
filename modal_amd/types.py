"""Public typing helpers (parity: /root/reference/py/modal/types.py)."""

from __future__ import annotations

from typing import Union

#: GPU request spellings accepted by @app.function(gpu=...)
GPUSpec = Union[None, bool, int, str]

#: path-like accepted by volume/mount APIs
PathLike = Union[str, "os.PathLike[str]"]  # noqa: F821

from .sandbox import ContainerProcess, FileIO  # noqa: E402,F401  (parity re-exports)
from .volume import FileEntry  # noqa: E402,F401
