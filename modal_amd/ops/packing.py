"""Segment pack/unpack wrappers over csrc/pack.hip.

Used for batched-tensor staging: N variable-length buffers living in one GPU
allocation are gathered into a contiguous packed buffer (and scattered back)
at HBM copy speed, replacing per-item host-side bytes joins.
"""

from __future__ import annotations

from . import load_lib


def pack_gpu(src: "object", src_offsets: "object", lengths: "object") -> tuple:
    """Gather slices of a CUDA uint8 tensor into one contiguous tensor.

    Returns (packed_tensor, dst_offsets_cpu).
    """
    lib = load_lib(required=True)
    import torch

    assert src.dtype == torch.uint8 and src.is_cuda
    lengths_cpu = lengths.to("cpu", dtype=torch.int64)
    dst_offsets_cpu = torch.zeros_like(lengths_cpu)
    if len(lengths_cpu) > 1:
        dst_offsets_cpu[1:] = torch.cumsum(lengths_cpu[:-1], 0)
    total = int(lengths_cpu.sum().item())
    packed = torch.empty(max(total, 1), dtype=torch.uint8, device="cuda")
    n = len(lengths_cpu)
    if n == 0:
        return packed[:0], dst_offsets_cpu
    src_off_d = src_offsets.to(device="cuda", dtype=torch.int64)
    len_d = lengths_cpu.cuda()
    dst_off_d = dst_offsets_cpu.cuda()
    max_len = int(lengths_cpu.max().item())
    rc = lib.ma_pack_segments(
        src.data_ptr(), src_off_d.data_ptr(), len_d.data_ptr(),
        packed.data_ptr(), dst_off_d.data_ptr(), n, max_len,
        torch.cuda.current_stream().cuda_stream,
    )
    if rc != 0:
        raise RuntimeError(f"pack kernel failed: hipError {rc}")
    return packed[:total], dst_offsets_cpu


def unpack_gpu(packed: "object", dst: "object", dst_offsets: "object", lengths: "object") -> None:
    """Scatter a packed CUDA uint8 tensor back to (offset, length) slices of dst."""
    lib = load_lib(required=True)
    import torch

    assert packed.dtype == torch.uint8 and packed.is_cuda and dst.is_cuda
    lengths_cpu = lengths.to("cpu", dtype=torch.int64)
    src_offsets_cpu = torch.zeros_like(lengths_cpu)
    if len(lengths_cpu) > 1:
        src_offsets_cpu[1:] = torch.cumsum(lengths_cpu[:-1], 0)
    n = len(lengths_cpu)
    if n == 0:
        return
    # keep device copies referenced past the (async) launch: a freed temporary's
    # block is recycled by the caching allocator before the kernel reads it
    src_off_d = src_offsets_cpu.cuda()
    len_d = lengths_cpu.cuda()
    dst_off_d = dst_offsets.to(device="cuda", dtype=torch.int64)
    rc = lib.ma_unpack_segments(
        packed.data_ptr(), src_off_d.data_ptr(), len_d.data_ptr(),
        dst.data_ptr(), dst_off_d.data_ptr(),
        n, int(lengths_cpu.max().item()),
        torch.cuda.current_stream().cuda_stream,
    )
    _ = (src_off_d, len_d, dst_off_d)  # alive past the launch; stream order covers the rest
    if rc != 0:
        raise RuntimeError(f"unpack kernel failed: hipError {rc}")
