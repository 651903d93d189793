"""Flash: direct-ingress service registration + custom autoscaler.

Parity: /root/reference/py/modal/experimental/flash.py — ``_FlashManager``
register/deregister (:31-249) and ``_FlashPrometheusAutoscaler`` (:281-641):
scrape per-replica metrics, compute a desired replica count from a target
metric value, and apply it with asymmetric smoothing (fast scale-up window,
slow scale-down window). Locally replicas are worker processes per GPU, and
the decision engine is reusable verbatim — it only needs a metrics callable
and an apply callable.
"""

from __future__ import annotations

import math
import time
from dataclasses import dataclass, field
from typing import Callable, Optional


@dataclass
class FlashEndpoint:
    name: str
    url: str
    registered_at: float = field(default_factory=time.time)


class FlashManager:
    """Registry of directly-ingressable service replicas."""

    def __init__(self) -> None:
        self.endpoints: dict[str, FlashEndpoint] = {}

    def register(self, name: str, url: str) -> FlashEndpoint:
        ep = FlashEndpoint(name, url)
        self.endpoints[name] = ep
        return ep

    def deregister(self, name: str) -> None:
        self.endpoints.pop(name, None)

    def list(self) -> list[FlashEndpoint]:
        return list(self.endpoints.values())


class FlashAutoscaler:
    """Metric-driven replica autoscaler with up/down smoothing windows.

    decision(): desired = ceil(current * metric / target), clamped to
    [min_replicas, max_replicas]; scale-ups apply after the value persists
    for ``scale_up_stabilization`` seconds, scale-downs after
    ``scale_down_stabilization`` (parity: reference up/down windows).
    """

    def __init__(
        self,
        get_metric: Callable[[], float],
        target_value: float,
        *,
        min_replicas: int = 1,
        max_replicas: int = 8,
        scale_up_stabilization: float = 0.0,
        scale_down_stabilization: float = 300.0,
        tolerance: float = 0.1,
    ):
        self.get_metric = get_metric
        self.target_value = target_value
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas
        self.scale_up_stabilization = scale_up_stabilization
        self.scale_down_stabilization = scale_down_stabilization
        self.tolerance = tolerance
        self._pending_desired: Optional[int] = None
        self._pending_since: float = 0.0

    def compute_desired(self, current_replicas: int, metric_value: float) -> int:
        if current_replicas == 0:
            return self.min_replicas if metric_value > 0 else 0
        ratio = metric_value / self.target_value
        if abs(ratio - 1.0) <= self.tolerance:
            return current_replicas
        desired = math.ceil(current_replicas * ratio)
        return max(self.min_replicas, min(self.max_replicas, desired))

    def decide(self, current_replicas: int, now: Optional[float] = None) -> int:
        """Stateful decision with stabilization windows. Returns the replica
        count to apply right now."""
        now = now if now is not None else time.time()
        desired = self.compute_desired(current_replicas, self.get_metric())
        if desired == current_replicas:
            self._pending_desired = None
            return current_replicas
        window = (
            self.scale_up_stabilization
            if desired > current_replicas
            else self.scale_down_stabilization
        )
        if self._pending_desired is None or (
            (self._pending_desired > current_replicas) != (desired > current_replicas)
        ):
            self._pending_desired = desired
            self._pending_since = now
        # keep the most conservative pending value within the window
        if desired > current_replicas:
            self._pending_desired = min(self._pending_desired, desired)
        else:
            self._pending_desired = max(self._pending_desired, desired)
        if now - self._pending_since >= window:
            applied = self._pending_desired
            self._pending_desired = None
            return applied
        return current_replicas
