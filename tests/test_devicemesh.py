"""Device-tensor plane: worker->worker transfers (gloo on CPU; RCCL on GPUs).

MODAL_AMD_MESH_ALL_TENSORS=1 routes even CPU tensors through the mesh so the
whole coordination protocol (export, device_transfer RPC, p2p send/recv,
host-staged pull fallback) runs without hardware.
"""

from __future__ import annotations

import time

import pytest

import modal_amd as modal

MESH_SECRET = None


def _mesh_secret():
    return modal.Secret.from_dict({"MODAL_AMD_MESH_ALL_TENSORS": "1"})


def test_queue_tensor_between_workers(client):
    pytest.importorskip("torch")
    app = modal.App("mesh-app")

    @app.function(secrets=[_mesh_secret()])
    def producer(q, n):
        import torch

        t = torch.arange(n, dtype=torch.float32) * 2
        q.put(t)
        return "sent"

    @app.function(secrets=[_mesh_secret()])
    def consumer(q):
        tensor = q.get(timeout=30)
        return float(tensor.sum().item())

    with app.run(client=client):
        with modal.Queue.ephemeral() as q:
            # make sure two distinct workers exist (mesh needs >= 2)
            fc1 = producer.spawn(q, 100)
            assert fc1.get(timeout=60) == "sent"
            # sum(0,2,4,...,198) = 2 * 99*100/2 = 9900
            assert consumer.remote(q) == 9900.0


def test_tensor_pull_to_client(client):
    """A worker-exported tensor returned to the client uses the host-staged
    pull path (the client is not a mesh member)."""
    torch = pytest.importorskip("torch")
    app = modal.App("mesh-pull")

    @app.function(secrets=[_mesh_secret()])
    def make_tensor(n):
        import torch

        return torch.ones(n, dtype=torch.int64) * 7

    with app.run(client=client):
        t = make_tensor.remote(50)
        assert t.shape == (50,)
        assert int(t.sum().item()) == 350


@pytest.mark.gpu
def test_gpu_tensor_roundtrip_via_queue(client):
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    app = modal.App("mesh-gpu")

    @app.function(gpu=1)
    def producer(q):
        import torch

        t = torch.full((1024,), 3.0, device="cuda")
        q.put(t)
        return "ok"

    @app.function(gpu=1)
    def consumer(q):
        t = q.get(timeout=60)
        assert t.is_cuda
        return float(t.sum().item())

    with app.run(client=client):
        with modal.Queue.ephemeral() as q:
            assert producer.remote(q) == "ok"
            # 1-GPU box: both functions land on the same worker -> the
            # same-worker shortcut serves the tensor without any transfer
            assert consumer.remote(q) == 3.0 * 1024
