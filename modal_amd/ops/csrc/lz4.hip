// LZ4-block (de)compression for gfx950 (MI355X, CDNA4).
//
// The north-star data plane calls for GPU chunk (de)compression next to the
// SHA-256 and pack kernels. Design mirrors the tree hash: a buffer splits
// into 4 KiB segments, each segment is an INDEPENDENT standard LZ4 block
// (greedy parser, 128-entry hash table in the lane's LDS slot), so
// compression parallelism = segment count (an 8 MiB volume block = 2048
// lanes). Output is written at a fixed stride; the pack kernel compacts.
// Decompression is one lane per segment into its fixed 4 KiB home — fully
// parallel, no coordination.
//
// Format per segment: standard LZ4 block (token | literals | 2B LE offset |
// extended lengths), interoperable with any LZ4 block decoder; the container
// framing (segment table) lives host-side in ops/compress.py.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define SEG_SIZE 4096
#define HASH_BITS 7
#define HASH_SIZE (1 << HASH_BITS)  // 128 u16 entries = 256 B LDS per lane
#define MIN_MATCH 4
// stop emitting matches near the end per the LZ4 spec (last 5 literals,
// no match within the last 12 bytes)
#define MFLIMIT 12
#define LASTLITERALS 5
#define COMP_BLOCK 128  // threads per workgroup (LDS: 128*256B = 32 KiB)

__device__ __forceinline__ uint32_t load32(const uint8_t* p) {
  uint32_t v;
  __builtin_memcpy(&v, p, 4);
  return v;
}

__device__ __forceinline__ uint32_t lz4_hash(uint32_t v) {
  return (v * 2654435761u) >> (32 - HASH_BITS);
}

// Compress one segment (<= SEG_SIZE bytes) into dst; returns compressed size,
// or 0 when the result would not fit in max_out (caller stores raw).
__device__ int lz4_compress_segment(const uint8_t* __restrict__ src, int src_len,
                                    uint8_t* __restrict__ dst, int max_out,
                                    uint16_t* __restrict__ table) {
#pragma unroll 8
  for (int i = 0; i < HASH_SIZE; ++i) table[i] = 0xFFFFu;

  int ip = 0;       // current position
  int anchor = 0;   // start of pending literals
  int op = 0;       // output position
  int misses = 0;   // LZ4-style skip acceleration through matchless regions
  const int mflimit = src_len - MFLIMIT;

  if (src_len >= MIN_MATCH + LASTLITERALS) {
    while (ip < mflimit) {
      uint32_t seq = load32(src + ip);
      uint32_t h = lz4_hash(seq);
      int cand = table[h];
      table[h] = (uint16_t)ip;
      if (cand != 0xFFFF && cand < ip && (ip - cand) <= 0xFFFF &&
          load32(src + cand) == seq) {
        misses = 0;
        // extend the match
        int mlen = MIN_MATCH;
        const int maxm = src_len - LASTLITERALS - ip;
        while (mlen < maxm && src[cand + mlen] == src[ip + mlen]) ++mlen;
        int lit = ip - anchor;
        // token + extended literal length + literals + offset + ext match len
        int need = 1 + (lit >= 15 ? 1 + lit / 255 : 0) + lit + 2 +
                   ((mlen - MIN_MATCH) >= 15 ? 1 + (mlen - MIN_MATCH) / 255 : 0);
        if (op + need + 8 > max_out) return 0;
        int ml = mlen - MIN_MATCH;
        uint8_t token = (uint8_t)((lit < 15 ? lit : 15) << 4) |
                        (uint8_t)(ml < 15 ? ml : 15);
        dst[op++] = token;
        if (lit >= 15) {
          int rest = lit - 15;
          while (rest >= 255) { dst[op++] = 255; rest -= 255; }
          dst[op++] = (uint8_t)rest;
        }
        {
          int i = 0;
          for (; i + 4 <= lit; i += 4) {  // word-wise literal copy
            uint32_t v;
            __builtin_memcpy(&v, src + anchor + i, 4);
            __builtin_memcpy(dst + op + i, &v, 4);
          }
          for (; i < lit; ++i) dst[op + i] = src[anchor + i];
          op += lit;
        }
        uint16_t off = (uint16_t)(ip - cand);
        dst[op++] = (uint8_t)(off & 0xFF);
        dst[op++] = (uint8_t)(off >> 8);
        if (ml >= 15) {
          int rest = ml - 15;
          while (rest >= 255) { dst[op++] = 255; rest -= 255; }
          dst[op++] = (uint8_t)rest;
        }
        ip += mlen;
        anchor = ip;
      } else {
        ip += 1 + (misses >> 6);  // accelerate through incompressible runs
        ++misses;
      }
    }
  }
  // trailing literals
  int lit = src_len - anchor;
  int need = 1 + (lit >= 15 ? 1 + lit / 255 : 0) + lit;
  if (op + need > max_out) return 0;
  dst[op++] = (uint8_t)((lit < 15 ? lit : 15) << 4);
  if (lit >= 15) {
    int rest = lit - 15;
    while (rest >= 255) { dst[op++] = 255; rest -= 255; }
    dst[op++] = (uint8_t)rest;
  }
  for (int i = 0; i < lit; ++i) dst[op++] = src[anchor + i];
  return op;
}

extern "C" __global__ __launch_bounds__(COMP_BLOCK) void lz4_compress_kernel(
    const uint8_t* __restrict__ src, int64_t src_len,
    uint8_t* __restrict__ dst,      // n_segments * out_stride
    int32_t* __restrict__ comp_lens,  // 0 => incompressible, store raw
    int out_stride, int n_segments) {
  __shared__ uint16_t tables[COMP_BLOCK][HASH_SIZE];
  int seg = blockIdx.x * blockDim.x + threadIdx.x;
  if (seg >= n_segments) return;
  int64_t start = (int64_t)seg * SEG_SIZE;
  int len = (int)min((int64_t)SEG_SIZE, src_len - start);
  int out = lz4_compress_segment(src + start, len, dst + (int64_t)seg * out_stride,
                                 out_stride, tables[threadIdx.x]);
  // only worth keeping if it actually shrank
  comp_lens[seg] = (out > 0 && out < len) ? out : 0;
}

// Decompress one standard LZ4 block; dst capacity dst_len (exact raw size).
// Returns bytes written or -1 on malformed input.
__device__ int lz4_decompress_segment(const uint8_t* __restrict__ src, int src_len,
                                      uint8_t* __restrict__ dst, int dst_len) {
  int ip = 0, op = 0;
  while (ip < src_len) {
    uint8_t token = src[ip++];
    int lit = token >> 4;
    if (lit == 15) {
      uint8_t b;
      do {
        if (ip >= src_len) return -1;
        b = src[ip++];
        lit += b;
      } while (b == 255);
    }
    if (ip + lit > src_len || op + lit > dst_len) return -1;
    {
      int i = 0;
      for (; i + 4 <= lit; i += 4) {
        uint32_t v;
        __builtin_memcpy(&v, src + ip + i, 4);
        __builtin_memcpy(dst + op + i, &v, 4);
      }
      for (; i < lit; ++i) dst[op + i] = src[ip + i];
    }
    ip += lit;
    op += lit;
    if (ip >= src_len) break;  // trailing-literal sequence
    if (ip + 2 > src_len) return -1;
    int off = src[ip] | (src[ip + 1] << 8);
    ip += 2;
    if (off == 0 || off > op) return -1;
    int mlen = (token & 0xF);
    if (mlen == 15) {
      uint8_t b;
      do {
        if (ip >= src_len) return -1;
        b = src[ip++];
        mlen += b;
      } while (b == 255);
    }
    mlen += MIN_MATCH;
    if (op + mlen > dst_len) return -1;
    const uint8_t* match = dst + op - off;
    if (off >= 8) {  // wide copy is overlap-safe at this distance
      int i = 0;
      for (; i + 8 <= mlen; i += 8) {
        uint64_t v;
        __builtin_memcpy(&v, match + i, 8);
        __builtin_memcpy(dst + op + i, &v, 8);
      }
      for (; i < mlen; ++i) dst[op + i] = match[i];
    } else {
      for (int i = 0; i < mlen; ++i) dst[op + i] = match[i];  // byte fwd copy
    }
    op += mlen;
  }
  return op;
}

extern "C" __global__ __launch_bounds__(256) void lz4_decompress_kernel(
    const uint8_t* __restrict__ comp,       // compacted segment data
    const int64_t* __restrict__ comp_offs,  // per-segment offset into comp
    const int32_t* __restrict__ comp_lens,  // 0 => raw (copy from comp as-is)
    uint8_t* __restrict__ dst, int64_t dst_len,
    int32_t* __restrict__ status, int n_segments) {
  int seg = blockIdx.x * blockDim.x + threadIdx.x;
  if (seg >= n_segments) return;
  int64_t start = (int64_t)seg * SEG_SIZE;
  int raw_len = (int)min((int64_t)SEG_SIZE, dst_len - start);
  const uint8_t* sp = comp + comp_offs[seg];
  int clen = comp_lens[seg];
  int written;
  if (clen == 0) {  // stored raw
    for (int i = 0; i < raw_len; ++i) dst[start + i] = sp[i];
    written = raw_len;
  } else {
    written = lz4_decompress_segment(sp, clen, dst + start, raw_len);
  }
  if (written != raw_len) atomicExch(status, seg + 1);
}

extern "C" int ma_lz4_compress(const void* src, long src_len, void* dst,
                               void* comp_lens, int out_stride, int n_segments,
                               void* stream) {
  if (n_segments <= 0) return 0;
  dim3 block(COMP_BLOCK);
  dim3 grid((n_segments + COMP_BLOCK - 1) / COMP_BLOCK);
  hipLaunchKernelGGL(lz4_compress_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)src, (int64_t)src_len, (uint8_t*)dst,
                     (int32_t*)comp_lens, out_stride, n_segments);
  return (int)hipGetLastError();
}

extern "C" int ma_lz4_decompress(const void* comp, const void* comp_offs,
                                 const void* comp_lens, void* dst, long dst_len,
                                 void* status, int n_segments, void* stream) {
  if (n_segments <= 0) return 0;
  dim3 block(256);
  dim3 grid((n_segments + 255) / 256);
  hipLaunchKernelGGL(lz4_decompress_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)comp, (const int64_t*)comp_offs,
                     (const int32_t*)comp_lens, (uint8_t*)dst, (int64_t)dst_len,
                     (int32_t*)status, n_segments);
  return (int)hipGetLastError();
}
