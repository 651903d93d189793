"""SchedulerPlacement: placement constraint hints.

Parity: /root/reference/py/modal/scheduler_placement.py:1-43. On one node the
meaningful constraint is GPU affinity — which of the 8 MI355X GPUs (or which
xGMI-adjacent set) an invocation should land on.
"""

from __future__ import annotations

from typing import Optional, Sequence, Union


class SchedulerPlacement:
    def __init__(
        self,
        region: Union[str, Sequence[str], None] = None,
        zone: Optional[str] = None,
        spot: Optional[bool] = None,
        gpu_index: Optional[int] = None,
        gpu_set: Optional[Sequence[int]] = None,
    ):
        self.region = region
        self.zone = zone
        self.spot = spot
        # MI355X-native extension: pin to a GPU or an xGMI-proximal set
        self.gpu_index = gpu_index
        self.gpu_set = list(gpu_set) if gpu_set else None

    def to_dict(self) -> dict:
        return {
            "region": self.region,
            "zone": self.zone,
            "spot": self.spot,
            "gpu_index": self.gpu_index,
            "gpu_set": self.gpu_set,
        }
