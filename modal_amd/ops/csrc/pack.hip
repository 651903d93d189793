// Segmented pack/unpack (gather/scatter of variable-length byte buffers)
// for gfx950 (MI355X, CDNA4).
//
// MI355X-native replacement for the reference's per-item payload copying:
// batched inputs/outputs (49-per-request batches, 8 MiB volume blocks,
// /root/reference/py/modal/parallel_map.py:82, blob_utils.py:63) are packed
// into one contiguous pinned staging buffer with an offsets table, moved with
// a single hipMemcpyAsync, and unpacked on the other side. One workgroup per
// segment-tile, dwordx4 (uint4) copies on the aligned body, scalar bytes on
// the edges. Memory-bound by design: the target is the HBM3E copy roofline
// (~6.3 TB/s read+write), not ALU.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define COPY_BLOCK 256
#define TILE_BYTES (64 * 1024)  // one workgroup moves one 64 KiB tile

// Copy len bytes from src to dst with a 256-thread workgroup.
__device__ void copy_tile(const uint8_t* __restrict__ src, uint8_t* __restrict__ dst,
                          int len) {
  const int tid = threadIdx.x;
  // head: align dst to 16
  uintptr_t dmis = reinterpret_cast<uintptr_t>(dst) & 15;
  int head = dmis ? (16 - static_cast<int>(dmis)) : 0;
  if (head > len) head = len;
  if (tid < head) dst[tid] = src[tid];
  src += head; dst += head; len -= head;

  const bool src_aligned = (reinterpret_cast<uintptr_t>(src) & 15) == 0;
  const int n_vec = len >> 4;  // 16-byte packets
  if (src_aligned) {
    const uint4* s4 = reinterpret_cast<const uint4*>(src);
    uint4* d4 = reinterpret_cast<uint4*>(dst);
    for (int k = tid; k < n_vec; k += COPY_BLOCK) d4[k] = s4[k];
  } else {
    // misaligned source: dword loads still coalesce acceptably
    uint4* d4 = reinterpret_cast<uint4*>(dst);
    for (int k = tid; k < n_vec; k += COPY_BLOCK) {
      const uint8_t* s = src + (static_cast<int64_t>(k) << 4);
      uint4 v;
      uint32_t* vp = reinterpret_cast<uint32_t*>(&v);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        uint32_t b0 = s[j * 4 + 0], b1 = s[j * 4 + 1], b2 = s[j * 4 + 2], b3 = s[j * 4 + 3];
        vp[j] = b0 | (b1 << 8) | (b2 << 16) | (b3 << 24);
      }
      d4[k] = v;
    }
  }
  int tail_start = n_vec << 4;
  int tail = len - tail_start;
  if (tid < tail) dst[tail_start + tid] = src[tail_start + tid];
}

// tiles: flattened (segment, tile_in_segment) pairs precomputed on host would
// cost a host roundtrip; instead map grid tiles by binary search over the
// cumulative tile counts? For the segment-size mix we serve (packets of a few
// KiB up to MiBs), a simpler two-level scheme works: blocks are striped over
// segments round-robin; each block then loops over its segment's tiles.
extern "C" __global__ __launch_bounds__(COPY_BLOCK) void pack_segments_kernel(
    const uint8_t* __restrict__ src,
    const int64_t* __restrict__ src_off,
    const int64_t* __restrict__ lens,
    uint8_t* __restrict__ dst,
    const int64_t* __restrict__ dst_off,
    int n_segments,
    int blocks_per_segment) {
  const int seg = blockIdx.x / blocks_per_segment;
  const int sub = blockIdx.x % blocks_per_segment;
  if (seg >= n_segments) return;
  const int64_t len = lens[seg];
  const int64_t n_tiles = (len + TILE_BYTES - 1) / TILE_BYTES;
  for (int64_t t = sub; t < n_tiles; t += blocks_per_segment) {
    const int64_t start = t * TILE_BYTES;
    const int tlen = static_cast<int>(min(static_cast<int64_t>(TILE_BYTES), len - start));
    copy_tile(src + src_off[seg] + start, dst + dst_off[seg] + start, tlen);
  }
}

static int launch_segcopy(const void* src, const void* src_off, const void* lens, void* dst,
                          const void* dst_off, int n, long max_len, void* stream) {
  if (n <= 0) return 0;
  // enough blocks to fill 256 CUs even for few large segments
  int blocks_per_segment = 1;
  if (n < 1024) {
    long tiles = (max_len + TILE_BYTES - 1) / TILE_BYTES;
    long want = (2048 + n - 1) / n;  // ~2048 blocks total => 8 per CU
    blocks_per_segment = (int)(tiles < want ? (tiles > 0 ? tiles : 1) : want);
    if (blocks_per_segment < 1) blocks_per_segment = 1;
  }
  dim3 block(COPY_BLOCK);
  dim3 grid((unsigned)(n * blocks_per_segment));
  hipLaunchKernelGGL(pack_segments_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)src, (const int64_t*)src_off, (const int64_t*)lens,
                     (uint8_t*)dst, (const int64_t*)dst_off, n, blocks_per_segment);
  return (int)hipGetLastError();
}

extern "C" int ma_pack_segments(const void* src, const void* src_off, const void* lens,
                                void* dst, const void* dst_off, int n, long max_len,
                                void* stream) {
  return launch_segcopy(src, src_off, lens, dst, dst_off, n, max_len, stream);
}

// unpack is the same copy with roles swapped (src offsets are the packed
// offsets, dst offsets the scattered ones) — one symmetric kernel serves both.
extern "C" int ma_unpack_segments(const void* src, const void* src_off, const void* lens,
                                  void* dst, const void* dst_off, int n, long max_len,
                                  void* stream) {
  return launch_segcopy(src, src_off, lens, dst, dst_off, n, max_len, stream);
}
