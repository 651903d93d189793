"""CloudBucketMount: S3/R2/GCS bucket mount configuration.

Parity: /root/reference/py/modal/cloud_bucket_mount.py (config-only). With no
egress on this node, a bucket mount maps to a local directory prefix
(``MODAL_AMD_BUCKET_ROOT/<bucket_name>``) so code paths depending on the
mount-point layout still run.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Optional


@dataclass
class CloudBucketMount:
    bucket_name: str
    bucket_endpoint_url: Optional[str] = None
    key_prefix: Optional[str] = None
    secret: Optional[Any] = None
    oidc_auth_role_arn: Optional[str] = None
    read_only: bool = False
    requester_pays: bool = False

    def local_dir(self, root: Optional[str] = None) -> str:
        root = root or os.environ.get("MODAL_AMD_BUCKET_ROOT", "/tmp/modal-amd-buckets")
        path = os.path.join(root, self.bucket_name, self.key_prefix or "")
        return path
