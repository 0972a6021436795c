// Column descriptors passed by value into kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "murmur3.h"

// dtype codes shared with ext.cpp
enum DtypeCode : int32_t {
  DT_I8 = 0,
  DT_U8 = 1,
  DT_I16 = 2,
  DT_I32 = 3,
  DT_I64 = 4,
  DT_F32 = 5,
  DT_F64 = 6,
  DT_BOOL = 7,
  DT_U32 = 8,
  DT_U64 = 9,
};

constexpr int MAX_KEY_COLS = 8;
constexpr int MAX_COLS = 16;

struct ColDesc {
  const void* ptr;
  int32_t code;
};

struct MutColDesc {
  void* ptr;
  int32_t code;
};

__device__ __forceinline__ int elt_size(int32_t code) {
  switch (code) {
    case DT_I8: case DT_U8: case DT_BOOL: return 1;
    case DT_I16: return 2;
    case DT_I32: case DT_F32: case DT_U32: return 4;
    default: return 8;
  }
}

// Hash one element exactly as the reference does per type
// (frame/ops_builtin.go): <=4-byte ints via uint32 conversion
// (sign-extending signed types), 8-byte via 8 LE bytes, floats via their
// bit patterns, bool -> seed + {0,1}.
__device__ __forceinline__ uint32_t hash_elt(const ColDesc& c, int64_t i,
                                             uint32_t seed) {
  switch (c.code) {
    case DT_I8:
      return mm3_u32((uint32_t)(int32_t)((const int8_t*)c.ptr)[i], seed);
    case DT_U8:
      return mm3_u32((uint32_t)((const uint8_t*)c.ptr)[i], seed);
    case DT_I16:
      return mm3_u32((uint32_t)(int32_t)((const int16_t*)c.ptr)[i], seed);
    case DT_I32:
      return mm3_u32((uint32_t)((const int32_t*)c.ptr)[i], seed);
    case DT_U32:
      return mm3_u32(((const uint32_t*)c.ptr)[i], seed);
    case DT_F32:
      return mm3_u32(((const uint32_t*)c.ptr)[i], seed);  // bit pattern
    case DT_I64:
      return mm3_u64((uint64_t)((const int64_t*)c.ptr)[i], seed);
    case DT_U64:
      return mm3_u64(((const uint64_t*)c.ptr)[i], seed);
    case DT_F64:
      return mm3_u64(((const uint64_t*)c.ptr)[i], seed);  // bit pattern
    case DT_BOOL:
      return seed + (uint32_t)(((const uint8_t*)c.ptr)[i] != 0);
  }
  return seed;
}

struct KeyCols {
  ColDesc cols[MAX_KEY_COLS];
  int n;
};

__device__ __forceinline__ uint32_t hash_row(const KeyCols& k, int64_t i,
                                             uint32_t seed) {
  uint32_t h = 0;
  for (int c = 0; c < k.n; ++c) h ^= hash_elt(k.cols[c], i, seed);
  return h;
}
