"""Experimental APIs: clustered (gang) functions + cluster info.

Parity: /root/reference/py/modal/experimental + _clustered_functions.py —
``@modal.experimental.clustered(size)`` runs one invocation as a gang of
``size`` containers with rank/world bootstrap (``ClusterInfo{rank,
cluster_id, container_ips, fabric_ids}``, reference
_clustered_functions.py:13-19; NCCL env tuning :59-71; TaskClusterHello
rendezvous :74-86).

MI355X-native shape: the gang is ``size`` worker processes on this node, one
per GPU; rendezvous is an in-process barrier in the scheduler; the RCCL
communicator forms over xGMI (``torch.distributed`` backend "nccl" IS RCCL
on ROCm; MASTER_ADDR=127.0.0.1). ``fabric_ids`` carry the xGMI hive id.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from ..exception import InvalidError
from ..partial_function import PartialFunction


@dataclass
class ClusterInfo:
    rank: int
    cluster_id: str
    container_ips: list[str] = field(default_factory=list)
    fabric_ids: list[int] = field(default_factory=list)


_cluster_info: Optional[ClusterInfo] = None


def get_cluster_info() -> ClusterInfo:
    if _cluster_info is None:
        raise InvalidError(
            "get_cluster_info() only works inside a @clustered function"
        )
    return _cluster_info


def _set_cluster_info(info: Optional[ClusterInfo]) -> None:
    global _cluster_info
    _cluster_info = info


def clustered(
    size: int, broadcast_inputs: bool = True, rdma: bool = False, fabric_size: Optional[int] = None
) -> Callable:
    """Gang-schedule an invocation across ``size`` GPU workers.

    Inside the function, ``torch.distributed.init_process_group("nccl")``
    picks up the env this runtime sets (RANK/WORLD_SIZE/MASTER_ADDR/
    MASTER_PORT) and forms the RCCL communicator over xGMI.
    """
    if size < 1:
        raise InvalidError("cluster size must be >= 1")

    def wrapper(raw_f: Any) -> PartialFunction:
        if isinstance(raw_f, PartialFunction):
            raw, flags = raw_f.raw_f, dict(raw_f.flags)
        else:
            raw, flags = raw_f, {}
        flags.update({"cluster_size": size, "rdma": rdma})
        return PartialFunction(raw, flags)

    return wrapper
