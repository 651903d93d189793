"""Output-chunk serialization for the worker fast path.

One C-pickler pass over a whole frame's result values (the mirror of input
chunking). Values containing tensors or live handles need the hook-aware
pickler per item, so those frames fall back to per-item bytes.
"""

from __future__ import annotations

import pickle
import sys
from typing import Optional


def serialize_value_chunk(values: list) -> tuple[Optional[bytes], Optional[list]]:
    """Returns (chunk_bytes, None) or (None, per_item_bytes) on fallback."""
    torch = sys.modules.get("torch")
    if torch is not None:
        from .._serialization import _walk_for_tensors

        if _walk_for_tensors(values, torch.Tensor):
            from .._serialization import serialize

            return None, [serialize(v) for v in values]
    try:
        data = pickle.dumps(values, 4)
    except Exception:
        from .._serialization import serialize

        return None, [serialize(v) for v in values]
    if len(data) > 2 * 1024 * 1024:
        # oversized frames go per-item so each output can blob-offload
        return None, [pickle.dumps(v, 4) for v in values]
    return data, None
