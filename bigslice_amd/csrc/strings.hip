// K17: device hashing of variable-length byte rows (strings) —
// murmur3_x86_32 bit-identical to the host path
// (bigslice_amd/hashing.py murmur3_bytes) and to the reference's
// string hashing (frame/ops_builtin.go:143-150).  Row i is
// bytes[offsets[i]:offsets[i+1]].  The 64-bit variant packs two
// independently-seeded 32-bit hashes (dictionary ids with ~2^-64
// pairwise collision odds at distinct seeds).

#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "murmur3.h"

__device__ __forceinline__ uint32_t mm3_bytes_row(const uint8_t* p,
                                                  uint32_t len,
                                                  uint32_t seed) {
  uint32_t h = seed;
  uint32_t off = 0;
  for (; off + 4 <= len; off += 4) {
    uint32_t k = (uint32_t)p[off] | ((uint32_t)p[off + 1] << 8) |
                 ((uint32_t)p[off + 2] << 16) |
                 ((uint32_t)p[off + 3] << 24);
    h = mm3_mix(h, k);
  }
  if (off < len) {
    uint32_t k1 = 0;
    const uint32_t tail = len - off;
    if (tail >= 3) k1 ^= (uint32_t)p[off + 2] << 16;
    if (tail >= 2) k1 ^= (uint32_t)p[off + 1] << 8;
    k1 ^= (uint32_t)p[off];
    k1 *= 0xcc9e2d51u;
    k1 = mm3_rotl(k1, 15);
    k1 *= 0x1b873593u;
    h ^= k1;
  }
  return mm3_fmix(h, len);
}

extern "C" __global__ void k_hash_bytes(const uint8_t* bytes,
                                        const int64_t* offsets, int64_t n,
                                        uint32_t seed, uint32_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t s = offsets[i];
    out[i] = mm3_bytes_row(bytes + s, (uint32_t)(offsets[i + 1] - s),
                           seed);
  }
}

extern "C" __global__ void k_hash_bytes64(const uint8_t* bytes,
                                          const int64_t* offsets,
                                          int64_t n, uint32_t seed_hi,
                                          uint32_t seed_lo, int64_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t s = offsets[i];
    const uint8_t* p = bytes + s;
    const uint32_t len = (uint32_t)(offsets[i + 1] - s);
    const uint64_t hi = mm3_bytes_row(p, len, seed_hi);
    const uint64_t lo = mm3_bytes_row(p, len, seed_lo);
    out[i] = (int64_t)((hi << 32) | lo);
  }
}
