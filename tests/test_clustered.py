"""@clustered gang functions: rank/world bootstrap, multi-process collective.

CPU variant runs torch.distributed over gloo with world_size 2 (two worker
processes); the GPU variant (marked gpu) uses the RCCL backend.
"""

from __future__ import annotations

import pytest

import modal_amd as modal
from modal_amd.experimental import clustered


def _make_allreduce_fn(app, backend: str):
    @app.function()
    @clustered(size=2)
    def allsum(x):
        import torch
        import torch.distributed as dist

        from modal_amd.experimental import get_cluster_info

        info = get_cluster_info()
        assert info.rank in (0, 1)
        assert len(info.container_ips) == 2
        dist.init_process_group(backend)
        t = torch.tensor([float(x + info.rank)])
        if backend == "nccl":
            t = t.cuda()
        dist.all_reduce(t)
        dist.destroy_process_group()
        return float(t.item())

    return allsum


def test_clustered_gloo_allreduce(client):
    app = modal.App("clustered-cpu")
    allsum = _make_allreduce_fn(app, "gloo")
    with app.run(client=client):
        # ranks contribute x+0 and x+1; rank 0's output is the result
        assert allsum.remote(5) == 11.0


def test_cluster_info_outside_raises():
    from modal_amd.exception import InvalidError
    from modal_amd.experimental import get_cluster_info

    with pytest.raises(InvalidError):
        get_cluster_info()


@pytest.mark.gpu
def test_clustered_rccl_allreduce(client):
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    if torch.cuda.device_count() < 2:
        pytest.skip("RCCL gang needs >= 2 GPUs (one rank per device)")
    app = modal.App("clustered-gpu")
    allsum = _make_allreduce_fn(app, "nccl")
    with app.run(client=client):
        assert allsum.remote(3) == 7.0


def test_gang_member_failure_fails_gang(client):
    """A failed gang member terminates its whole gang instead of retrying
    solo (a lone rank can never rendezvous; gang fate-sharing)."""
    import time

    app = modal.App("gang-fail-app")

    @app.function()
    @clustered(size=2)
    def failer():
        import time as _t

        from modal_amd.experimental import get_cluster_info

        if get_cluster_info().rank == 1:
            raise RuntimeError("boom-rank1")
        _t.sleep(30)  # rank 0 parked, as if waiting in a collective
        return "rank0-done"

    with app.run(client=client):
        t0 = time.time()
        with pytest.raises(Exception) as excinfo:
            failer.remote()
        elapsed = time.time() - t0
        assert elapsed < 20, "gang should be torn down, not awaited to completion"
        msg = str(excinfo.value)
        assert "boom-rank1" in msg or "gang member" in msg or "cancel" in msg.lower()
