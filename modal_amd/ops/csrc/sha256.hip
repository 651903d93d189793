// Batched SHA-256 for gfx950 (MI355X, CDNA4).
//
// Data-plane hashing for the content-addressed store / Volume block store:
// the reference hashes every blob (MD5+SHA-256 streaming, 64 KiB chunks,
// /root/reference/py/modal/_utils/hash_utils.py:68) and every Volume v2
// 8 MiB block (per-block SHA-256, volume.py:1401) on the CPU. Here large
// buffers are hashed on-GPU: one lane per message (a 64 KiB leaf of the tree
// hash, or an arbitrary message of a batch), message chunks staged through
// LDS (per-lane 68-byte slots, conflict-free stride-17 banking), SHA-256
// rounds fully in registers.
//
// Wave width is 64 (CDNA4); a 256-thread workgroup hashes 256 messages and
// uses 256*68 B = 17 KiB of the CU's 160 KiB LDS, so occupancy is
// register-bound, not LDS-bound. A 1 GiB buffer = 16384 leaves fills the
// 256-CU chip with >64 workgroups of work.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define LANES_PER_BLOCK 256

__constant__ uint32_t K256[64] = {
    0x428a2f98u, 0x71374491u, 0xb5c0fbcfu, 0xe9b5dba5u, 0x3956c25bu, 0x59f111f1u,
    0x923f82a4u, 0xab1c5ed5u, 0xd807aa98u, 0x12835b01u, 0x243185beu, 0x550c7dc3u,
    0x72be5d74u, 0x80deb1feu, 0x9bdc06a7u, 0xc19bf174u, 0xe49b69c1u, 0xefbe4786u,
    0x0fc19dc6u, 0x240ca1ccu, 0x2de92c6fu, 0x4a7484aau, 0x5cb0a9dcu, 0x76f988dau,
    0x983e5152u, 0xa831c66du, 0xb00327c8u, 0xbf597fc7u, 0xc6e00bf3u, 0xd5a79147u,
    0x06ca6351u, 0x14292967u, 0x27b70a85u, 0x2e1b2138u, 0x4d2c6dfcu, 0x53380d13u,
    0x650a7354u, 0x766a0abbu, 0x81c2c92eu, 0x92722c85u, 0xa2bfe8a1u, 0xa81a664bu,
    0xc24b8b70u, 0xc76c51a3u, 0xd192e819u, 0xd6990624u, 0xf40e3585u, 0x106aa070u,
    0x19a4c116u, 0x1e376c08u, 0x2748774cu, 0x34b0bcb5u, 0x391c0cb3u, 0x4ed8aa4au,
    0x5b9cca4fu, 0x682e6ff3u, 0x748f82eeu, 0x78a5636fu, 0x84c87814u, 0x8cc70208u,
    0x90befffau, 0xa4506cebu, 0xbef9a3f7u, 0xc67178f2u};

__device__ __forceinline__ uint32_t rotr32(uint32_t x, int n) {
  return __builtin_rotateright32(x, n);
}

__device__ __forceinline__ uint32_t bswap32(uint32_t x) {
  return __builtin_bswap32(x);
}

// One compression round set over the 16 words sitting in the lane's LDS slot.
// Words in the slot are raw little-endian memory; byteswap on read.
__device__ void sha256_compress(uint32_t state[8], const uint32_t* slot) {
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) w[i] = bswap32(slot[i]);

  uint32_t a = state[0], b = state[1], c = state[2], d = state[3];
  uint32_t e = state[4], f = state[5], g = state[6], h = state[7];

#pragma unroll
  for (int t = 0; t < 64; ++t) {
    uint32_t wt;
    if (t < 16) {
      wt = w[t];
    } else {
      uint32_t w15 = w[(t - 15) & 15], w2 = w[(t - 2) & 15];
      uint32_t s0 = rotr32(w15, 7) ^ rotr32(w15, 18) ^ (w15 >> 3);
      uint32_t s1 = rotr32(w2, 17) ^ rotr32(w2, 19) ^ (w2 >> 10);
      wt = w[t & 15] + s0 + w[(t - 7) & 15] + s1;
      w[t & 15] = wt;
    }
    uint32_t S1 = rotr32(e, 6) ^ rotr32(e, 11) ^ rotr32(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = h + S1 + ch + K256[t] + wt;
    uint32_t S0 = rotr32(a, 2) ^ rotr32(a, 13) ^ rotr32(a, 22);
    uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + maj;
    h = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  state[0] += a; state[1] += b; state[2] += c; state[3] += d;
  state[4] += e; state[5] += f; state[6] += g; state[7] += h;
}

// Two interleaved rounds streams: SHA-256's per-round dependency chain is
// ~6 cycles deep with ~2 issue slots used, so a second independent chain per
// lane roughly doubles ALU utilisation at low occupancy (the common case:
// one 8 MiB volume block = 512 leaves = 2 workgroups).
__device__ void sha256_compress_dual(uint32_t* sA, const uint32_t* slotA,
                                     uint32_t* sB, const uint32_t* slotB) {
  uint32_t wA[16], wB[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    wA[i] = bswap32(slotA[i]);
    wB[i] = bswap32(slotB[i]);
  }
  uint32_t aA = sA[0], bA = sA[1], cA = sA[2], dA = sA[3];
  uint32_t eA = sA[4], fA = sA[5], gA = sA[6], hA = sA[7];
  uint32_t aB = sB[0], bB = sB[1], cB = sB[2], dB = sB[3];
  uint32_t eB = sB[4], fB = sB[5], gB = sB[6], hB = sB[7];

#pragma unroll
  for (int t = 0; t < 64; ++t) {
    uint32_t wtA, wtB;
    if (t < 16) {
      wtA = wA[t];
      wtB = wB[t];
    } else {
      uint32_t w15, w2, s0, s1;
      w15 = wA[(t - 15) & 15]; w2 = wA[(t - 2) & 15];
      s0 = rotr32(w15, 7) ^ rotr32(w15, 18) ^ (w15 >> 3);
      s1 = rotr32(w2, 17) ^ rotr32(w2, 19) ^ (w2 >> 10);
      wtA = wA[t & 15] + s0 + wA[(t - 7) & 15] + s1;
      wA[t & 15] = wtA;
      w15 = wB[(t - 15) & 15]; w2 = wB[(t - 2) & 15];
      s0 = rotr32(w15, 7) ^ rotr32(w15, 18) ^ (w15 >> 3);
      s1 = rotr32(w2, 17) ^ rotr32(w2, 19) ^ (w2 >> 10);
      wtB = wB[t & 15] + s0 + wB[(t - 7) & 15] + s1;
      wB[t & 15] = wtB;
    }
    uint32_t k = K256[t];
    uint32_t S1A = rotr32(eA, 6) ^ rotr32(eA, 11) ^ rotr32(eA, 25);
    uint32_t S1B = rotr32(eB, 6) ^ rotr32(eB, 11) ^ rotr32(eB, 25);
    uint32_t chA = (eA & fA) ^ (~eA & gA);
    uint32_t chB = (eB & fB) ^ (~eB & gB);
    uint32_t t1A = hA + S1A + chA + k + wtA;
    uint32_t t1B = hB + S1B + chB + k + wtB;
    uint32_t S0A = rotr32(aA, 2) ^ rotr32(aA, 13) ^ rotr32(aA, 22);
    uint32_t S0B = rotr32(aB, 2) ^ rotr32(aB, 13) ^ rotr32(aB, 22);
    uint32_t majA = (aA & bA) ^ (aA & cA) ^ (bA & cA);
    uint32_t majB = (aB & bB) ^ (aB & cB) ^ (bB & cB);
    uint32_t t2A = S0A + majA;
    uint32_t t2B = S0B + majB;
    hA = gA; gA = fA; fA = eA; eA = dA + t1A;
    dA = cA; cA = bA; bA = aA; aA = t1A + t2A;
    hB = gB; gB = fB; fB = eB; eB = dB + t1B;
    dB = cB; cB = bB; bB = aB; aB = t1B + t2B;
  }
  sA[0] += aA; sA[1] += bA; sA[2] += cA; sA[3] += dA;
  sA[4] += eA; sA[5] += fA; sA[6] += gA; sA[7] += hA;
  sB[0] += aB; sB[1] += bB; sB[2] += cB; sB[3] += dB;
  sB[4] += eB; sB[5] += fB; sB[6] += gB; sB[7] += hB;
}

// Stage chunk c of a message into the lane's LDS slot (data or padding).
__device__ void sha256_load_chunk(const uint8_t* __restrict__ src, int64_t len,
                                  int64_t n_full, int rem, int n_pad, bool aligned16,
                                  int64_t c, uint32_t* slot) {
  uint8_t* slot_bytes = reinterpret_cast<uint8_t*>(slot);
  if (c < n_full) {
    const uint8_t* p = src + (c << 6);
    if (aligned16) {
      const uint4* v = reinterpret_cast<const uint4*>(p);
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        uint4 x = v[k];
        slot[k * 4 + 0] = x.x;
        slot[k * 4 + 1] = x.y;
        slot[k * 4 + 2] = x.z;
        slot[k * 4 + 3] = x.w;
      }
    } else {
#pragma unroll
      for (int k = 0; k < 64; ++k) slot_bytes[k] = p[k];
    }
    return;
  }
  const int pc = static_cast<int>(c - n_full);
#pragma unroll
  for (int k = 0; k < 16; ++k) slot[k] = 0u;
  if (pc == 0) {
    const uint8_t* p = src + (n_full << 6);
    for (int k = 0; k < rem; ++k) slot_bytes[k] = p[k];
    slot_bytes[rem] = 0x80u;
  }
  if (pc == n_pad - 1) {
    const uint64_t bitlen = static_cast<uint64_t>(len) << 3;
    slot[14] = bswap32(static_cast<uint32_t>(bitlen >> 32));
    slot[15] = bswap32(static_cast<uint32_t>(bitlen & 0xffffffffu));
  }
}

// ILP x2 variant: each lane hashes TWO messages with interleaved rounds.
extern "C" __global__ __launch_bounds__(LANES_PER_BLOCK) void sha256_many2_kernel(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ offsets,
    const int64_t* __restrict__ lengths,
    uint8_t* __restrict__ out,
    int n) {
  __shared__ uint32_t tile[LANES_PER_BLOCK][2][17];
  const int pair = blockIdx.x * blockDim.x + threadIdx.x;
  const int iA = pair * 2;
  const int iB = iA + 1;
  if (iA >= n) return;

  uint32_t stateA[8] = {0x6a09e667u, 0xbb67ae85u, 0x3c6ef372u, 0xa54ff53au,
                        0x510e527fu, 0x9b05688cu, 0x1f83d9abu, 0x5be0cd19u};
  uint32_t stateB[8] = {0x6a09e667u, 0xbb67ae85u, 0x3c6ef372u, 0xa54ff53au,
                        0x510e527fu, 0x9b05688cu, 0x1f83d9abu, 0x5be0cd19u};
  uint32_t* slotA = tile[threadIdx.x][0];
  uint32_t* slotB = tile[threadIdx.x][1];

  const int64_t lenA = lengths[iA];
  const uint8_t* srcA = buf + offsets[iA];
  const int64_t nfA = lenA >> 6;
  const int remA = static_cast<int>(lenA & 63);
  const int npA = (remA + 9 <= 64) ? 1 : 2;
  const bool alA = ((reinterpret_cast<uintptr_t>(srcA)) & 15) == 0;
  const int64_t totalA = nfA + npA;

  const bool haveB = iB < n;
  const int64_t lenB = haveB ? lengths[iB] : 0;
  const uint8_t* srcB = haveB ? buf + offsets[iB] : srcA;
  const int64_t nfB = lenB >> 6;
  const int remB = static_cast<int>(lenB & 63);
  const int npB = (remB + 9 <= 64) ? 1 : 2;
  const bool alB = ((reinterpret_cast<uintptr_t>(srcB)) & 15) == 0;
  const int64_t totalB = haveB ? nfB + npB : 0;

  const int64_t total = totalA > totalB ? totalA : totalB;
  for (int64_t c = 0; c < total; ++c) {
    const bool aA = c < totalA;
    const bool aB = c < totalB;
    if (aA) sha256_load_chunk(srcA, lenA, nfA, remA, npA, alA, c, slotA);
    if (aB) sha256_load_chunk(srcB, lenB, nfB, remB, npB, alB, c, slotB);
    if (aA && aB) {
      sha256_compress_dual(stateA, slotA, stateB, slotB);
    } else if (aA) {
      sha256_compress(stateA, slotA);
    } else if (aB) {
      sha256_compress(stateB, slotB);
    }
  }
  uint32_t* digA = reinterpret_cast<uint32_t*>(out + static_cast<int64_t>(iA) * 32);
#pragma unroll
  for (int k = 0; k < 8; ++k) digA[k] = bswap32(stateA[k]);
  if (haveB) {
    uint32_t* digB = reinterpret_cast<uint32_t*>(out + static_cast<int64_t>(iB) * 32);
#pragma unroll
    for (int k = 0; k < 8; ++k) digB[k] = bswap32(stateB[k]);
  }
}

extern "C" __global__ __launch_bounds__(LANES_PER_BLOCK) void sha256_many_kernel(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ offsets,
    const int64_t* __restrict__ lengths,
    uint8_t* __restrict__ out,  // n * 32 bytes
    int n) {
  // Per-lane 17-word LDS slot: stride 17 (odd) => base bank = 17*tid mod 32
  // is a bijection within each 32-lane group => conflict-free b32 access.
  __shared__ uint32_t tile[LANES_PER_BLOCK][17];

  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;

  const int64_t off = offsets[i];
  const int64_t len = lengths[i];
  uint32_t* slot = tile[threadIdx.x];
  uint8_t* slot_bytes = reinterpret_cast<uint8_t*>(slot);

  uint32_t state[8] = {0x6a09e667u, 0xbb67ae85u, 0x3c6ef372u, 0xa54ff53au,
                       0x510e527fu, 0x9b05688cu, 0x1f83d9abu, 0x5be0cd19u};

  const int64_t n_full = len >> 6;               // full 64-byte chunks
  const int rem = static_cast<int>(len & 63);
  // one padding chunk if rem+1+8 <= 64, else two
  const int n_pad = (rem + 9 <= 64) ? 1 : 2;

  const uint8_t* src = buf + off;
  const bool aligned16 = ((reinterpret_cast<uintptr_t>(src)) & 15) == 0;

  for (int64_t c = 0; c < n_full; ++c) {
    const uint8_t* p = src + (c << 6);
    if (aligned16) {
      const uint4* v = reinterpret_cast<const uint4*>(p);
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        uint4 x = v[k];  // global -> regs -> LDS stage
        slot[k * 4 + 0] = x.x;
        slot[k * 4 + 1] = x.y;
        slot[k * 4 + 2] = x.z;
        slot[k * 4 + 3] = x.w;
      }
    } else {
#pragma unroll
      for (int k = 0; k < 64; ++k) slot_bytes[k] = p[k];
    }
    sha256_compress(state, slot);
  }

  // padding chunk(s)
  const uint64_t bitlen = static_cast<uint64_t>(len) << 3;
  for (int pc = 0; pc < n_pad; ++pc) {
#pragma unroll
    for (int k = 0; k < 16; ++k) slot[k] = 0u;
    if (pc == 0) {
      const uint8_t* p = src + (n_full << 6);
      for (int k = 0; k < rem; ++k) slot_bytes[k] = p[k];
      slot_bytes[rem] = 0x80u;
    }
    if (pc == n_pad - 1) {
      // big-endian 64-bit bit length in the last 8 bytes
      slot[14] = bswap32(static_cast<uint32_t>(bitlen >> 32));
      slot[15] = bswap32(static_cast<uint32_t>(bitlen & 0xffffffffu));
    }
    sha256_compress(state, slot);
  }

  uint32_t* dig = reinterpret_cast<uint32_t*>(out + static_cast<int64_t>(i) * 32);
#pragma unroll
  for (int k = 0; k < 8; ++k) dig[k] = bswap32(state[k]);
}

extern "C" int ma_sha256_many(const void* buf, const void* offsets, const void* lengths,
                              void* out, int n, void* stream) {
  if (n <= 0) return 0;
  dim3 block(LANES_PER_BLOCK);
  dim3 grid((n + LANES_PER_BLOCK - 1) / LANES_PER_BLOCK);
  hipLaunchKernelGGL(sha256_many_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)buf, (const int64_t*)offsets,
                     (const int64_t*)lengths, (uint8_t*)out, n);
  return (int)hipGetLastError();
}

extern "C" int ma_sha256_many2(const void* buf, const void* offsets, const void* lengths,
                               void* out, int n, void* stream) {
  if (n <= 0) return 0;
  const int pairs = (n + 1) / 2;
  dim3 block(LANES_PER_BLOCK);
  dim3 grid((pairs + LANES_PER_BLOCK - 1) / LANES_PER_BLOCK);
  hipLaunchKernelGGL(sha256_many2_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)buf, (const int64_t*)offsets,
                     (const int64_t*)lengths, (uint8_t*)out, n);
  return (int)hipGetLastError();
}
