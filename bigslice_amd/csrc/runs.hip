// K18: ordered run-boundary compaction over a SORTED key array.
// Emits (unique_keys, run_starts, count) in ONE streaming pass.
//
// Replaces rocPRIM reduce_by_key over a counting iterator (measured
// 637 us for 62.5M int64 keys = 0.8 TB/s — its generic carry
// propagation moves key+aggregate pairs through the lookback): a
// boundary flag here is just keys[r] != keys[r-1], so a tile's state
// granule is ONE {flag|count} word and the compaction is ballot
// arithmetic.  Geometry: 256-thread blocks, each thread IPT rows
// strided by BLOCK (coalesced); ordering inside a tile is
// (slab k, wave, lane), scanned in LDS once per tile.
//
// Reference parity: this is the assembly step of the cogroup reader
// (cogroup.go:150-214 groups equal keys per dep); the sort-merge
// itself is in ops/cogroup.py.

#include <hip/hip_runtime.h>

#include <cstdint>

#define RUNS_BLOCK 256
#define RUNS_IPT 32
#define RUNS_TILE (RUNS_BLOCK * RUNS_IPT)
#define RUNS_FLAG_AGG 1ull
#define RUNS_FLAG_PREFIX 2ull
#define RUNS_LOOKBACK_BATCH 8

// 4 waves/SIMD floor (<=128 VGPRs): the unrolled slab loops otherwise
// hold 175 VGPRs of live addresses and occupancy drops to 8 waves/CU.
__global__ __launch_bounds__(RUNS_BLOCK, 4) void k_runs_sorted(
    const int64_t* __restrict__ keys, int64_t n,
    int64_t* __restrict__ uniq_out, int64_t* __restrict__ starts_out,
    int64_t* __restrict__ count_out,
    unsigned long long* __restrict__ state) {
  constexpr int WAVES = RUNS_BLOCK / 64;
  __shared__ unsigned int cnt[RUNS_IPT * WAVES];  // slab-major
  __shared__ unsigned int wtot[RUNS_BLOCK / 64];  // per-wave totals
  __shared__ unsigned long long lds_base[1];
  const int tile = blockIdx.x;
  const int64_t base = (int64_t)tile * RUNS_TILE;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));

  // pass A: flags + per-(slab, wave) counts.  Keys are NOT kept in
  // registers: boundaries are ~1 row in 50+, so pass B re-reads
  // keys[r] for flagged lanes only (a few MB) instead of pinning
  // IPT int64 registers per thread.
  // Consecutive lanes hold consecutive rows, so keys[r-1] is a wave
  // shuffle; lane 0's real neighbor load is issued inside the same
  // burst.  Loads burst 8-deep before the shuffle/ballot chain — the
  // interleaved form serializes every load behind a waitcnt (same
  // fix as the radix scatter's burst loads).
  uint64_t my_flags = 0;  // bit k = row base + k*BLOCK + tid
  unsigned int my_wtot = 0;
#pragma unroll
  for (int k0 = 0; k0 < RUNS_IPT; k0 += 8) {
    int64_t kvb[8], kpb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int64_t r =
          base + (int64_t)(k0 + j) * RUNS_BLOCK + threadIdx.x;
      kvb[j] = (r < n) ? keys[r] : 0;
      if (lane == 0) kpb[j] = (r > 0 && r < n) ? keys[r - 1] : 0;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k0 + j;
      const int64_t r = base + (int64_t)k * RUNS_BLOCK + threadIdx.x;
      const bool in = r < n;
      int64_t kprev = __shfl_up(kvb[j], 1, 64);
      if (lane == 0) kprev = kpb[j];
      const bool flag = in && ((r == 0) || (kvb[j] != kprev));
      my_flags |= flag ? (1ull << k) : 0ull;
      const uint64_t b = __ballot(flag);
      if (lane == 0) {
        const unsigned int c = (unsigned int)__popcll(b);
        cnt[k * WAVES + wave] = c;
        my_wtot += c;
      }
    }
  }
  if (lane == 0) wtot[wave] = my_wtot;
  __syncthreads();
  // Split the inter-tile critical path across waves: wave 0 reduces
  // the counters (7 shuffle steps), publishes AGG IMMEDIATELY, then
  // walks 64 predecessors per round with lane-parallel loads; wave 1
  // meanwhile builds the exclusive scan of the counters.  The serial
  // form (one thread: 128-add scan, then a batched walk) left every
  // successor tile spinning on a publish that sat behind the scan.
  constexpr int NCNT = RUNS_IPT * WAVES;
  unsigned int run = 0;
  if (wave == 0) {
    // total from the per-wave accumulators (NOT from cnt — wave 1 is
    // overwriting cnt with its exclusive scan concurrently)
    unsigned int s = (lane < WAVES) ? wtot[lane] : 0;
    for (int off = 32; off; off >>= 1)
      s += (unsigned int)__shfl_xor((int)s, off, 64);
    run = s;  // wave-uniform total
    if (lane == 0) {
      const unsigned long long pub =
          ((tile == 0 ? RUNS_FLAG_PREFIX : RUNS_FLAG_AGG) << 62) |
          (unsigned long long)run;
      __hip_atomic_store(&state[tile], pub, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
    unsigned long long excl = 0;
    if (tile > 0) {
      int j0 = tile - 1;
      bool done = false;
      while (!done) {
        const int idx = j0 - lane;
        unsigned long long x = 0;
        if (idx >= 0) {
          x = __hip_atomic_load(&state[idx], __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
          while ((x >> 62) == 0) {
            __builtin_amdgcn_s_sleep(1);
            x = __hip_atomic_load(&state[idx], __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
          }
        }
        const uint64_t pmask =
            __ballot(idx >= 0 && (x >> 62) == RUNS_FLAG_PREFIX);
        unsigned long long c = x & ((1ull << 62) - 1);
        if (pmask) {
          const int lp = __ffsll((unsigned long long)pmask) - 1;
          if (lane > lp) c = 0;
          done = true;
        } else if (idx < 0) {
          c = 0;
        }
        for (int off = 32; off; off >>= 1)
          c += __shfl_xor((unsigned long long)c, off, 64);
        excl += c;
        j0 -= 64;
        if (j0 < 0) done = true;
      }
      if (lane == 0)
        __hip_atomic_store(
            &state[tile],
            (RUNS_FLAG_PREFIX << 62) | (excl + (unsigned long long)run),
            __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    }
    if (lane == 0) {
      lds_base[0] = excl;
      if (tile == (int)gridDim.x - 1)
        *count_out = (int64_t)(excl + run);
    }
  } else if (wave == 1) {
    // exclusive scan of cnt[NCNT] in place: shuffle scan per 64-chunk
    // with a running carry
    unsigned int carry = 0;
    for (int i0 = 0; i0 < NCNT; i0 += 64) {
      const unsigned int v = cnt[i0 + lane];
      unsigned int incl = v;
      for (int off = 1; off < 64; off <<= 1) {
        const unsigned int up =
            (unsigned int)__shfl_up((int)incl, off, 64);
        if (lane >= off) incl += up;
      }
      cnt[i0 + lane] = carry + incl - v;
      carry += (unsigned int)__shfl((int)incl, 63, 64);
    }
  }
  __syncthreads();
  const unsigned long long tb = lds_base[0];

  // pass B: compacted ordered writes
#pragma unroll
  for (int k = 0; k < RUNS_IPT; ++k) {
    const bool flag = (my_flags >> k) & 1u;
    const uint64_t b = __ballot(flag);
    if (flag) {
      const int64_t r = base + (int64_t)k * RUNS_BLOCK + threadIdx.x;
      const unsigned int pos = cnt[k * WAVES + wave] +
                               (unsigned int)__popcll(b & lt);
      uniq_out[tb + pos] = keys[r];
      starts_out[tb + pos] = r;
    }
  }
}
