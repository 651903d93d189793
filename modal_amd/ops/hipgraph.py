"""hipGraph capture for launch-bound repeated callables.

The MI355X design brief: "capture launch-bound inner loops in hipGraphs".
A k-kernel eager callable pays k CPU dispatches (~15-20 us each under
ROCm eager torch) every invocation; captured as a hipGraph (torch's
``torch.cuda.CUDAGraph`` IS hipGraph on ROCm) the whole sequence replays
with one dispatch. Worth it when k is large and shapes repeat — exactly
the per-item inner loop of a serving function.

Semantics and constraints (standard graph-capture rules):
* the callable must be shape-stable and side-effect free on host state;
* its tensor inputs are STAGED — callers pass new values, we copy them
  into the captured input buffers and replay;
* the captured output buffers are overwritten by every replay, so each
  call copies results out into a ring of ``ring_depth`` slots — results
  remain valid until ``ring_depth`` further calls (deep enough to cover a
  chunk-batched D2H flush; raise it if you hold results longer).

Usage::

    graphed = GraphedCall(fn, example_args=(torch.zeros(4096, device="cuda"),))
    y = graphed(x)           # one replay instead of k eager launches

Reference hook: the reference has no equivalent (its runtime never touches
kernels); this is MI355X-native machinery per the north-star brief.
"""

from __future__ import annotations

from typing import Any, Callable, Optional

__all__ = ["GraphedCall", "graphs_supported"]


def graphs_supported() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


class GraphedCall:
    """Capture ``fn(*tensors_and_scalars)`` into a hipGraph on first call.

    Positional args may be CUDA tensors (staged by copy into the captured
    input buffers; shapes/dtypes must match the first call) or Python
    numbers (staged through a 0-d float32 device tensor each). The return
    value must be a tensor or a flat tuple/list of tensors.
    """

    def __init__(
        self,
        fn: Callable,
        ring_depth: int = 64,
        warmups: int = 3,
    ):
        self._fn = fn
        self._ring_depth = max(ring_depth, 2)
        self._warmups = warmups
        self._graph: Any = None
        self._inputs: Optional[list] = None
        self._is_scalar: Optional[list] = None
        self._outputs: Any = None
        self._single: bool = True
        self._ring: Optional[list] = None
        self._slot = 0
        self.replays = 0

    def _stage_args(self, args: tuple) -> None:
        for buf, scalar, value in zip(self._inputs, self._is_scalar, args):
            if scalar:
                buf.fill_(float(value))
            else:
                buf.copy_(value, non_blocking=True)

    def _build(self, args: tuple) -> None:
        import torch

        self._is_scalar = [not torch.is_tensor(a) for a in args]
        self._inputs = [
            torch.full((), float(a), device="cuda", dtype=torch.float32)
            if not torch.is_tensor(a)
            else a.detach().clone()
            for a in args
        ]
        # warm up on a side stream so capture sees settled allocations
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self._warmups):
                out = self._fn(*self._inputs)
        torch.cuda.current_stream().wait_stream(side)

        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            out = self._fn(*self._inputs)
        if isinstance(out, (tuple, list)):
            self._single = False
            self._outputs = list(out)
        else:
            self._single = True
            self._outputs = [out]
        for o in self._outputs:
            if not torch.is_tensor(o):
                raise TypeError(
                    "GraphedCall callables must return tensors "
                    f"(got {type(o).__name__})"
                )
        self._ring = [
            [torch.empty_like(o) for o in self._outputs]
            for _ in range(self._ring_depth)
        ]

    def __call__(self, *args: Any) -> Any:
        if self._graph is None:
            self._build(args)
        if len(args) != len(self._inputs):
            raise TypeError(
                f"GraphedCall captured {len(self._inputs)} args, got {len(args)}"
            )
        self._stage_args(args)
        self._graph.replay()
        self.replays += 1
        slots = self._ring[self._slot]
        self._slot = (self._slot + 1) % self._ring_depth
        for slot, out in zip(slots, self._outputs):
            slot.copy_(out)
        return slots[0] if self._single else tuple(slots)
