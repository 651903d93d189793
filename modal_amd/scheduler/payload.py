"""Payload envelope: inline bytes below the threshold, CAS blob above.

Parity: payloads >2 MiB go to blob storage instead of inline
(/root/reference/py/modal/_utils/blob_utils.py:36-39; decision at
function_utils.py:586). Single-node twist: workers share the filesystem with
the scheduler, so a "blob upload" is one write into the content-addressed
store and the id crosses the wire.
"""

from __future__ import annotations

from typing import Optional

from .blobs import INLINE_LIMIT, BlobStore


def encode_payload(data: bytes, store: Optional[BlobStore], limit: int = INLINE_LIMIT) -> dict:
    if store is not None and len(data) > limit:
        return {"payload": b"", "payload_blob": store.put(data)}
    return {"payload": data}


def decode_payload(item: dict, store: Optional[BlobStore]) -> bytes:
    blob_id = item.get("payload_blob")
    if blob_id:
        if store is None:
            raise RuntimeError("blob payload received but no blob store available")
        return store.get(blob_id)
    return item.get("payload") or b""
