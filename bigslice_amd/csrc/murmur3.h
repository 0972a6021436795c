// Device murmur3_x86_32 — bit-identical to the engine's host path
// (bigslice_amd/hashing.py) and to the reference's key hashing
// (frame/ops_builtin.go:140-164: scalars hashed as 4/8 little-endian
// bytes; bool -> seed + {0,1}).  Keeping the hash bit-identical keeps
// partition assignment reproducible across CPU/GPU and vs the reference.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

__device__ __host__ __forceinline__ uint32_t mm3_rotl(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}

__device__ __host__ __forceinline__ uint32_t mm3_mix(uint32_t h, uint32_t k) {
  k *= 0xcc9e2d51u;
  k = mm3_rotl(k, 15);
  k *= 0x1b873593u;
  h ^= k;
  h = mm3_rotl(h, 13);
  return h * 5u + 0xe6546b64u;
}

__device__ __host__ __forceinline__ uint32_t mm3_fmix(uint32_t h, uint32_t len) {
  h ^= len;
  h ^= h >> 16;
  h *= 0x85ebca6bu;
  h ^= h >> 13;
  h *= 0xc2b2ae35u;
  h ^= h >> 16;
  return h;
}

// 4-byte value (reference hash32).
__device__ __host__ __forceinline__ uint32_t mm3_u32(uint32_t v, uint32_t seed) {
  return mm3_fmix(mm3_mix(seed, v), 4);
}

// 8-byte value (reference hash64): low dword first (little-endian).
__device__ __host__ __forceinline__ uint32_t mm3_u64(uint64_t v, uint32_t seed) {
  uint32_t h = seed;
  h = mm3_mix(h, (uint32_t)(v & 0xffffffffu));
  h = mm3_mix(h, (uint32_t)(v >> 32));
  return mm3_fmix(h, 8);
}
