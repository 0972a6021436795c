// K9: hash-aggregate group-by over device-resident columns.
//
// Role-parity with the reference's combiningFrame (exec/combiner.go:
// open-addressing, power-of-two capacity, grow x2 on pressure, seeded with
// hashSeed 0x9acb0442 so a previous partitioning step does not strip hash
// entropy).  The CDNA4 redesign: one pass of global-memory atomics into an
// open-addressing table in HBM/LLC; on table overflow the host doubles
// capacity and retries (the reference's rehash-x2, done batch-wise).
// Slot `cap` (one extra) is reserved for the sentinel key itself.

#include <hip/hip_runtime.h>

#include "columns.h"

#define THREADS 256

// Sentinel marking an empty slot; rows whose key equals the sentinel are
// accumulated in the reserved extra slot (index == capacity).
#define GB_SENTINEL 0x8000000000000000LL

enum AggCode : int32_t { AGG_SUM = 0, AGG_MIN = 1, AGG_MAX = 2, AGG_PROD = 3 };

struct ValCols {
  ColDesc src[MAX_COLS];
  MutColDesc tab[MAX_COLS];
  int32_t agg[MAX_COLS];
  int n;
};

template <typename T, typename U>
__device__ __forceinline__ void cas_loop(T* addr, T v, int agg) {
  U* a = (U*)addr;
  U old = *a, assumed;
  do {
    assumed = old;
    T cur = __builtin_bit_cast(T, assumed);
    T nv;
    switch (agg) {
      case AGG_MIN: nv = v < cur ? v : cur; break;
      case AGG_MAX: nv = v > cur ? v : cur; break;
      default: nv = cur * v; break;
    }
    if (nv == cur) return;
    old = atomicCAS(a, assumed, __builtin_bit_cast(U, nv));
  } while (old != assumed);
}

__device__ __forceinline__ void accum(const MutColDesc& t, int64_t slot,
                                      const ColDesc& s, int64_t i,
                                      int agg) {
  switch (t.code) {
    case DT_I32: {
      int32_t v = ((const int32_t*)s.ptr)[i];
      int32_t* a = (int32_t*)t.ptr + slot;
      if (agg == AGG_SUM) atomicAdd((int*)a, (int)v);
      else if (agg == AGG_MIN) atomicMin((int*)a, (int)v);
      else if (agg == AGG_MAX) atomicMax((int*)a, (int)v);
      else cas_loop<int32_t, unsigned int>(a, v, agg);
      break;
    }
    case DT_I64: {
      int64_t v = ((const int64_t*)s.ptr)[i];
      int64_t* a = (int64_t*)t.ptr + slot;
      if (agg == AGG_SUM)
        atomicAdd((unsigned long long*)a, (unsigned long long)v);
      else
        cas_loop<int64_t, unsigned long long>(a, v, agg);
      break;
    }
    case DT_F32: {
      float v = ((const float*)s.ptr)[i];
      float* a = (float*)t.ptr + slot;
      if (agg == AGG_SUM) atomicAdd(a, v);
      else cas_loop<float, unsigned int>(a, v, agg);
      break;
    }
    case DT_F64: {
      double v = ((const double*)s.ptr)[i];
      double* a = (double*)t.ptr + slot;
      if (agg == AGG_SUM) atomicAdd(a, v);
      else cas_loop<double, unsigned long long>(a, v, agg);
      break;
    }
  }
}

// Compaction: scan the table, appending used slots to the output arrays.
// Wave-aggregated cursor: one global atomicAdd per wave (64 lanes), not per
// element — a single hot counter saturates at ~88 atomics/us on this chip.
struct CompactCols {
  ColDesc tab[MAX_COLS];
  MutColDesc out[MAX_COLS];
  int n;
};

__device__ __forceinline__ void copy_elt(const MutColDesc& dst, int64_t di,
                                         const ColDesc& src, int64_t si) {
  switch (elt_size(src.code)) {
    case 1: ((uint8_t*)dst.ptr)[di] = ((const uint8_t*)src.ptr)[si]; break;
    case 2: ((uint16_t*)dst.ptr)[di] = ((const uint16_t*)src.ptr)[si]; break;
    case 4: ((uint32_t*)dst.ptr)[di] = ((const uint32_t*)src.ptr)[si]; break;
    default: ((uint64_t*)dst.ptr)[di] = ((const uint64_t*)src.ptr)[si];
  }
}

// Two-pass block-chunk compaction: each block owns a contiguous slot
// range; pass 1 counts its used slots (block reduce -> ONE cursor atomic
// per block), pass 2 writes them out at block_base + thread prefix.
// A single hot cursor saturates at ~88 atomics/us on this chip, so the
// atomic count must scale with blocks (~2k), not waves (~100k).
extern "C" __global__ void k_groupby_compact(
    const int64_t* tkeys, int64_t cap, int64_t slots_per_block,
    CompactCols cols, int64_t* out_keys, unsigned long long* cursor) {
  __shared__ uint32_t counts[THREADS];
  __shared__ unsigned long long block_base;
  int64_t start = (int64_t)blockIdx.x * slots_per_block;
  int64_t end = min(start + slots_per_block, cap);
  // pass 1: per-thread count over its strided slots
  uint32_t mine = 0;
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    mine += (tkeys[i] != GB_SENTINEL);
  counts[threadIdx.x] = mine;
  __syncthreads();
  // exclusive scan of per-thread counts (simple LDS scan, 256 wide)
  if (threadIdx.x == 0) {
    uint32_t run = 0;
    for (int t = 0; t < (int)blockDim.x; ++t) {
      uint32_t c = counts[t];
      counts[t] = run;
      run += c;
    }
    block_base = run ? atomicAdd(cursor, (unsigned long long)run) : 0;
  }
  __syncthreads();
  // pass 2: write used slots at base + prefix
  int64_t pos = (int64_t)block_base + counts[threadIdx.x];
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    int64_t k = tkeys[i];
    if (k == GB_SENTINEL) continue;
    out_keys[pos] = k;
    for (int c = 0; c < cols.n; ++c)
      copy_elt(cols.out[c], pos, cols.tab[c], i);
    ++pos;
  }
}

// Two-level insert for the single-int64-SUM shape (reference K10,
// exec/bigmachine.go:1144-1199 two-level combine): each workgroup first
// aggregates into a 1024-slot LDS table (16 KiB: small enough to keep
// full 32-wave/CU occupancy for the global-atomic path), absorbing hot
// keys entirely
// on-CU (LDS atomics), and only LDS-table misses and the end-of-block
// flush touch the global table.  At low key cardinality this removes
// the global atomic contention that serializes the one-level kernel
// (measured 4x); at high cardinality the LDS probe adds a small
// constant.
#define GB_LDS_SLOTS 1024

__device__ __forceinline__ void gb_global_insert_sum(
    int64_t k, long long v, int64_t* tkeys, long long* tab, uint64_t mask,
    int64_t cap, uint32_t seed, int32_t* sentinel_seen, int32_t* overflow,
    int64_t max_probes) {
  int64_t slot;
  if (k == GB_SENTINEL) {
    atomicOr(sentinel_seen, 1);
    slot = cap;
  } else {
    uint64_t h = mm3_u64((uint64_t)k, seed) & mask;
    int64_t probes = 0;
    for (;;) {
      long long cur = ((volatile long long*)tkeys)[h];
      if (cur == k) break;
      if (cur == GB_SENTINEL) {
        long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                   (unsigned long long)GB_SENTINEL,
                                   (unsigned long long)k);
        if (prev == GB_SENTINEL || prev == k) break;
      }
      h = (h + 1) & mask;
      if (++probes >= max_probes) {
        atomicOr(overflow, 1);
        return;
      }
    }
    slot = (int64_t)h;
  }
  atomicAdd((unsigned long long*)&tab[slot], (unsigned long long)v);
}

// One row through the LDS table; returns false on LDS-table miss
// (caller falls back to the global table).  hits counts rows absorbed
// by an EXISTING LDS entry (the duplicate-rate signal).
__device__ __forceinline__ bool gb_lds_try(
    int64_t k, long long v, long long* lk, long long* lv, uint32_t seed,
    uint32_t* hits) {
  uint32_t h = mm3_u64((uint64_t)k, seed) & (GB_LDS_SLOTS - 1);
  for (int p = 0; p < 4; ++p) {  // short LDS probe chain
    long long cur = lk[h];
    bool existed = true;
    if (cur == GB_SENTINEL) {
      long long prev = atomicCAS((unsigned long long*)&lk[h],
                                 (unsigned long long)GB_SENTINEL,
                                 (unsigned long long)k);
      existed = (prev != GB_SENTINEL);
      cur = existed ? prev : k;
    }
    if (cur == k) {
      atomicAdd((unsigned long long*)&lv[h], (unsigned long long)v);
      if (existed && hits) atomicAdd(hits, 1u);
      return true;
    }
    h = (h + 1) & (GB_LDS_SLOTS - 1);
  }
  return false;
}

// Grid-stride geometry like the one-level kernel (the block-chunked
// variant measured 2.6x slower in the global-atomic regime); each
// block's LDS table aggregates its strided rows and flushes once.
extern "C" __global__ void k_groupby_insert_sum_i64_lds(
    const int64_t* keys, const int64_t* vals, int64_t n, int64_t* tkeys,
    long long* tab, int64_t cap, uint32_t seed, int32_t* sentinel_seen,
    int32_t* overflow, int64_t max_probes, int64_t rows_per_block,
    int32_t force, uint32_t* global_hits) {
  __shared__ long long lk[GB_LDS_SLOTS];
  __shared__ long long lv[GB_LDS_SLOTS];
  __shared__ uint32_t lds_hits;
  for (int i = threadIdx.x; i < GB_LDS_SLOTS; i += blockDim.x) {
    lk[i] = GB_SENTINEL;
    lv[i] = 0;
  }
  if (threadIdx.x == 0) lds_hits = 0;
  __syncthreads();
  uint64_t gmask = (uint64_t)cap - 1;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t base = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;

  // Phase 1 (sample): the first SAMPLE_ITERS grid-stride iterations go
  // through the LDS table, counting rows absorbed by existing entries.
  // force: -1 = sample-adaptive, 0 = global-only, 1 = always-LDS.
  const int64_t SAMPLE_ITERS = 16;
  int64_t sampled = 0;
  int64_t it = 0;
  int64_t i = base;
  if (force == -1) {
    for (; i < n && it < SAMPLE_ITERS; i += stride, ++it) {
      int64_t k = keys[i];
      long long v = (long long)vals[i];
      ++sampled;
      if (k == GB_SENTINEL ||
          !gb_lds_try(k, v, lk, lv, seed, &lds_hits))
        gb_global_insert_sum(k, v, tkeys, tab, gmask, cap, seed,
                             sentinel_seen, overflow, max_probes);
    }
    __syncthreads();
  }
  // Duplicate-rate decision: keep the LDS tier only when >=1/8 of the
  // block's sampled rows hit an existing entry — otherwise the key
  // space is too wide for the LDS table and probing it costs latency.
  (void)sampled;
  bool use_lds = (force == -1)
      ? (lds_hits * 8u >= (uint32_t)(SAMPLE_ITERS * blockDim.x))
      : (force == 1);
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    long long v = (long long)vals[i];
    if (k == GB_SENTINEL ||
        !(use_lds && gb_lds_try(k, v, lk, lv, seed, nullptr)))
      gb_global_insert_sum(k, v, tkeys, tab, gmask, cap, seed,
                           sentinel_seen, overflow, max_probes);
  }
  __syncthreads();
  // flush the block's LDS table into the global one
  for (int i2 = threadIdx.x; i2 < GB_LDS_SLOTS; i2 += blockDim.x) {
    if (lk[i2] != GB_SENTINEL)
      gb_global_insert_sum(lk[i2], lv[i2], tkeys, tab, gmask, cap, seed,
                           sentinel_seen, overflow, max_probes);
  }
  // absorption feedback for the host-side mode decision
  if (global_hits && threadIdx.x == 0 && lds_hits)
    atomicAdd(global_hits, lds_hits);
}

// Packed-slot variant for the hottest shape (single int64 value, SUM):
// slot i = table[2i]=key, table[2i+1]=sum.  Key and accumulator share a
// 16-byte-aligned pair, so each row touches ONE cache line instead of
// two — the insert is atomic-latency-bound, so halving touched lines
// matters.  Table size 2*(cap+1); extra slot for the sentinel key.
// Partition ids from table-slot high bits (for the partition-first
// insert experiment: rows whose slots share a table range are inserted
// together, giving the TCC atomics temporal locality).
extern "C" __global__ void k_slot_pids(const int64_t* keys, int64_t n,
                                       int64_t cap, int32_t nparts,
                                       uint32_t seed, int32_t* pids) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t mask = (uint64_t)cap - 1;
  uint64_t shift = 0;
  while (((uint64_t)cap >> shift) > (uint64_t)nparts) ++shift;
  for (; i < n; i += stride) {
    uint64_t h = mm3_u64((uint64_t)keys[i], seed) & mask;
    pids[i] = (int32_t)(h >> shift);
  }
}

extern "C" __global__ void k_groupby_insert_packed_sum_i64(
    const int64_t* keys, const int64_t* vals, int64_t n, int64_t* table,
    int64_t cap, uint32_t seed, int32_t* sentinel_seen, int32_t* overflow,
    int64_t max_probes) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t mask = (uint64_t)cap - 1;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    int64_t slot;
    if (k == GB_SENTINEL) {
      atomicOr(sentinel_seen, 1);
      slot = cap;
    } else {
      uint64_t h = mm3_u64((uint64_t)k, seed) & mask;
      int64_t probes = 0;
      for (;;) {
        long long cur = ((volatile long long*)table)[2 * h];
        if (cur == k) break;
        if (cur == GB_SENTINEL) {
          long long prev = atomicCAS((unsigned long long*)&table[2 * h],
                                     (unsigned long long)GB_SENTINEL,
                                     (unsigned long long)k);
          if (prev == GB_SENTINEL || prev == k) break;
        }
        h = (h + 1) & mask;
        if (++probes >= max_probes) {
          atomicOr(overflow, 1);
          return;
        }
      }
      slot = (int64_t)h;
    }
    atomicAdd((unsigned long long*)&table[2 * slot + 1],
              (unsigned long long)vals[i]);
  }
}

// Sub-table replication experiment (NOTES.md item 1 / round-2 task:
// confirm or refute the fabric/atomic-throughput ceiling): REP
// replicas of the packed table, block b inserting into replica
// b % REP (blocks land on XCD b % 8, so replica ~= XCD).  Relieves
// per-line atomic contention at the cost of REP x the table footprint
// (pushes a presized table out of the 256 MiB L3).  Merge = compact
// each replica then re-aggregate (cheap at #distinct rows).
extern "C" __global__ void k_groupby_insert_packed_rep(
    const int64_t* keys, const int64_t* vals, int64_t n, int64_t* table,
    int64_t cap, int32_t nrep, uint32_t seed, int32_t* sentinel_seen,
    int32_t* overflow, int64_t max_probes) {
  int64_t* mytab = table + (int64_t)(blockIdx.x % nrep) * 2 * (cap + 1);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t mask = (uint64_t)cap - 1;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    int64_t slot;
    if (k == GB_SENTINEL) {
      atomicOr(sentinel_seen, 1);
      slot = cap;
    } else {
      uint64_t h = mm3_u64((uint64_t)k, seed) & mask;
      int64_t probes = 0;
      for (;;) {
        long long cur = ((volatile long long*)mytab)[2 * h];
        if (cur == k) break;
        if (cur == GB_SENTINEL) {
          long long prev = atomicCAS((unsigned long long*)&mytab[2 * h],
                                     (unsigned long long)GB_SENTINEL,
                                     (unsigned long long)k);
          if (prev == GB_SENTINEL || prev == k) break;
        }
        h = (h + 1) & mask;
        if (++probes >= max_probes) {
          atomicOr(overflow, 1);
          return;
        }
      }
      slot = (int64_t)h;
    }
    atomicAdd((unsigned long long*)&mytab[2 * slot + 1],
              (unsigned long long)vals[i]);
  }
}

extern "C" __global__ void k_groupby_compact_packed(
    const int64_t* table, int64_t cap, int64_t slots_per_block,
    int64_t* out_keys, int64_t* out_vals, unsigned long long* cursor) {
  __shared__ uint32_t counts[THREADS];
  __shared__ unsigned long long block_base;
  int64_t start = (int64_t)blockIdx.x * slots_per_block;
  int64_t end = min(start + slots_per_block, cap);
  uint32_t mine = 0;
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    mine += (table[2 * i] != GB_SENTINEL);
  counts[threadIdx.x] = mine;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t run = 0;
    for (int t = 0; t < (int)blockDim.x; ++t) {
      uint32_t c = counts[t];
      counts[t] = run;
      run += c;
    }
    block_base = run ? atomicAdd(cursor, (unsigned long long)run) : 0;
  }
  __syncthreads();
  int64_t pos = (int64_t)block_base + counts[threadIdx.x];
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    int64_t k = table[2 * i];
    if (k == GB_SENTINEL) continue;
    out_keys[pos] = k;
    out_vals[pos] = table[2 * i + 1];
    ++pos;
  }
}

extern "C" __global__ void k_fill_packed_slots(int64_t* table,
                                               int64_t nslots) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nslots; i += stride) {
    table[2 * i] = GB_SENTINEL;
    table[2 * i + 1] = 0;
  }
}

extern "C" __global__ void k_groupby_insert(
    const int64_t* keys, int64_t n, ValCols vals, int64_t* tkeys,
    int64_t cap, uint32_t seed, int32_t* sentinel_seen, int32_t* overflow,
    int64_t max_probes) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t mask = (uint64_t)cap - 1;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    int64_t slot;
    if (k == GB_SENTINEL) {
      // reserved extra slot
      if (atomicOr(sentinel_seen, 1) == 0) {}
      slot = cap;
    } else {
      uint64_t h = mm3_u64((uint64_t)k, seed) & mask;
      int64_t probes = 0;
      for (;;) {
        // Plain read first: once the table is warm most probes land on
        // an already-claimed matching key, and a load is far cheaper
        // than an atomicCAS.
        long long cur = ((volatile long long*)tkeys)[h];
        if (cur == k) break;
        if (cur == GB_SENTINEL) {
          long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                     (unsigned long long)GB_SENTINEL,
                                     (unsigned long long)k);
          if (prev == GB_SENTINEL || prev == k) break;
        }
        h = (h + 1) & mask;
        // A long probe chain means the table is too loaded: signal the
        // host to grow x2 and re-insert (reference combiner grow policy).
        if (++probes >= max_probes) {
          atomicOr(overflow, 1);
          return;
        }
      }
      slot = (int64_t)h;
    }
    for (int c = 0; c < vals.n; ++c)
      accum(vals.tab[c], slot, vals.src[c], i, vals.agg[c]);
  }
}

// Resolve one row whose FIRST probe was already loaded (cur0): the MLP
// insert batches first-probe loads of several rows before resolving,
// so the random-access latency of the table read is paid R-ways in
// parallel instead of serially per row.
__device__ __forceinline__ void gb_resolve_add(
    int64_t k, long long v, long long cur0, uint64_t h, int64_t* tkeys,
    long long* tab, uint64_t mask, int64_t cap, uint32_t /*seed*/,
    int32_t* sentinel_seen, int32_t* overflow, int64_t max_probes) {
  int64_t slot;
  if (k == GB_SENTINEL) {
    atomicOr(sentinel_seen, 1);
    slot = cap;
  } else {
    long long cur = cur0;
    int64_t probes = 0;
    for (;;) {
      if (cur == k) break;
      if (cur == GB_SENTINEL) {
        long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                   (unsigned long long)GB_SENTINEL,
                                   (unsigned long long)k);
        if (prev == GB_SENTINEL || prev == k) break;
      }
      h = (h + 1) & mask;
      if (++probes >= max_probes) {
        atomicOr(overflow, 1);
        return;
      }
      cur = ((volatile long long*)tkeys)[h];
    }
    slot = (int64_t)h;
  }
  atomicAdd((unsigned long long*)&tab[slot], (unsigned long long)v);
}

#define GB_MLP_R 4

// Multi-row software-pipelined one-level insert (single int64 SUM
// shape): R grid-stride rows per iteration, all R first-probe loads in
// flight together (NOTES.md headroom item 1).
extern "C" __global__ void k_groupby_insert_sum_i64_mlp(
    const int64_t* keys, const int64_t* vals, int64_t n, int64_t* tkeys,
    long long* tab, int64_t cap, uint32_t seed, int32_t* sentinel_seen,
    int32_t* overflow, int64_t max_probes) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t mask = (uint64_t)cap - 1;
  for (int64_t base = tid; base < n; base += stride * GB_MLP_R) {
    int64_t k[GB_MLP_R];
    long long v[GB_MLP_R];
    uint64_t h[GB_MLP_R];
    long long cur[GB_MLP_R];
    bool act[GB_MLP_R];
#pragma unroll
    for (int r = 0; r < GB_MLP_R; ++r) {
      int64_t i = base + (int64_t)r * stride;  // coalesced per r
      act[r] = i < n;
      if (act[r]) {
        k[r] = keys[i];
        v[r] = vals[i];
      }
    }
#pragma unroll
    for (int r = 0; r < GB_MLP_R; ++r)
      if (act[r] && k[r] != GB_SENTINEL)
        h[r] = mm3_u64((uint64_t)k[r], seed) & mask;
#pragma unroll
    for (int r = 0; r < GB_MLP_R; ++r)  // R loads in flight
      if (act[r] && k[r] != GB_SENTINEL)
        cur[r] = ((volatile long long*)tkeys)[h[r]];
#pragma unroll
    for (int r = 0; r < GB_MLP_R; ++r)
      if (act[r])
        gb_resolve_add(k[r], v[r], cur[r], h[r], tkeys, tab, mask, cap,
                       seed, sentinel_seen, overflow, max_probes);
  }
}
