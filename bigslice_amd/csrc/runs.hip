// K18: ordered run-boundary compaction over a SORTED key array.
// Emits (unique_keys, run_starts, count) in ONE streaming pass.
//
// Replaces rocPRIM reduce_by_key over a counting iterator (measured
// 637 us for 62.5M int64 keys = 0.8 TB/s — its generic carry
// propagation moves key+aggregate pairs through the lookback): a
// boundary flag here is just keys[r] != keys[r-1], so a tile's state
// granule is ONE {flag|count} word and the compaction is ballot
// arithmetic.
//
// Geometry notes (measured on the way here):
// * consecutive lanes hold consecutive rows, so keys[r-1] is a wave
//   shuffle; lane 0's real neighbor load rides in the same burst;
// * loads burst 8-deep ahead of the shuffle/ballot chain (interleaved
//   form serializes each load behind a waitcnt);
// * per-block fixed costs (two barriers, the lookback's L2 round
//   trips) were ~half the runtime of an 8K-row tile whose streaming
//   part is ~2.5 us, capping the kernel at 1.4 TB/s.  Large inputs
//   therefore run 64K-row tiles: the outer slab loop is NOT unrolled
//   (a fully unrolled 64-slab loop held 175 VGPRs of live addresses)
//   and pass B recomputes boundary flags from L2-warm keys instead of
//   pinning a >64-bit flag register per thread.
//
// Reference parity: this is the assembly step of the cogroup reader
// (cogroup.go:150-214 groups equal keys per dep); the sort-merge
// itself is in ops/cogroup.py.

#include <hip/hip_runtime.h>

#include <cstdint>

#define RUNS_BLOCK 256
#define RUNS_FLAG_AGG 1ull
#define RUNS_FLAG_PREFIX 2ull

template <int IPT>
__global__ __launch_bounds__(RUNS_BLOCK, 4) void k_runs_sorted(
    const int64_t* __restrict__ keys, int64_t n,
    int64_t* __restrict__ uniq_out, int64_t* __restrict__ starts_out,
    int64_t* __restrict__ count_out,
    unsigned long long* __restrict__ state) {
  constexpr int WAVES = RUNS_BLOCK / 64;
  constexpr int64_t TILE = (int64_t)RUNS_BLOCK * IPT;
  __shared__ unsigned int cnt[IPT * WAVES];  // slab-major
  __shared__ unsigned int wtot[WAVES];       // per-wave totals
  __shared__ unsigned char fbits[IPT / 8][RUNS_BLOCK];  // flag bytes
  __shared__ unsigned long long lds_base[1];
  const int tile = blockIdx.x;
  const int64_t base = (int64_t)tile * TILE;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));

  // pass A: boundary flags -> per-(slab, wave) counts
  unsigned int my_wtot = 0;
#pragma unroll 1
  for (int k0 = 0; k0 < IPT; k0 += 8) {
    int64_t kvb[8], kpb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int64_t r =
          base + (int64_t)(k0 + j) * RUNS_BLOCK + threadIdx.x;
      kvb[j] = (r < n) ? keys[r] : 0;
      if (lane == 0) kpb[j] = (r > 0 && r < n) ? keys[r - 1] : 0;
    }
    unsigned int fb = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k0 + j;
      const int64_t r = base + (int64_t)k * RUNS_BLOCK + threadIdx.x;
      const bool in = r < n;
      int64_t kprev = __shfl_up(kvb[j], 1, 64);
      if (lane == 0) kprev = kpb[j];
      const bool flag = in && ((r == 0) || (kvb[j] != kprev));
      fb |= flag ? (1u << j) : 0u;
      const uint64_t b = __ballot(flag);
      if (lane == 0) {
        const unsigned int c = (unsigned int)__popcll(b);
        cnt[k * WAVES + wave] = c;
        my_wtot += c;
      }
    }
    fbits[k0 / 8][threadIdx.x] = (unsigned char)fb;
  }
  if (lane == 0) wtot[wave] = my_wtot;
  __syncthreads();

  // Split the inter-tile critical path across waves: wave 0 reduces
  // the per-wave totals (7 shuffles), publishes AGG IMMEDIATELY, then
  // walks 64 predecessors per round with lane-parallel loads; wave 1
  // meanwhile builds the exclusive scan of the slab counters (totals
  // come from wtot, NOT cnt, which wave 1 is overwriting).
  constexpr int NCNT = IPT * WAVES;
  unsigned int run = 0;
  if (wave == 0) {
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

  // pass B: flags from LDS, compacted ordered writes.  Only flagged
  // lanes (~1 row in 50+) re-read their key from global.
#pragma unroll 1
  for (int k0 = 0; k0 < IPT; k0 += 8) {
    const unsigned int fb = fbits[k0 / 8][threadIdx.x];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k0 + j;
      const bool flag = (fb >> j) & 1u;
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
}

// K16 fast path: segmented reduction over SORTED values given the
// run boundaries from k_runs_sorted.  One THREAD per run: lanes of a
// wave own 64 ADJACENT runs, so the wave's serial walks collectively
// stream one contiguous region (every cache line fully used).  The
// host guards this kernel to max run length <= 4096 and falls back to
// rocPRIM reduce_by_key for skewed runs and for float sums (which
// need its fixed reduction tree for run-to-run determinism; integer
// adds are exact in any order).  codes: 0=sum 1=min 2=max.
__global__ __launch_bounds__(256) void k_segreduce_i64(
    const int64_t* __restrict__ vals,
    const int64_t* __restrict__ starts, int64_t m, int64_t n,
    int code, int64_t* __restrict__ out) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < m; i += stride) {
    const int64_t s = starts[i];
    const int64_t e = (i + 1 < m) ? starts[i + 1] : n;
    int64_t acc;
    if (code == 0) {
      acc = 0;
      for (int64_t j = s; j < e; ++j) acc += vals[j];
    } else if (code == 1) {
      acc = vals[s];
      for (int64_t j = s + 1; j < e; ++j) acc = min(acc, vals[j]);
    } else {
      acc = vals[s];
      for (int64_t j = s + 1; j < e; ++j) acc = max(acc, vals[j]);
    }
    out[i] = acc;
  }
}

// Guard helper: max run length + run count in one 2-word device
// scalar so the host pays a single readback.  cnt_dev is
// runs_sorted's count output (read on device — no extra sync).
__global__ void k_runs_guard(const int64_t* __restrict__ starts,
                             const int64_t* __restrict__ cnt_dev,
                             int64_t n,
                             unsigned long long* __restrict__ out_max) {
  const int64_t m = *cnt_dev;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  unsigned long long mx = 0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < m; i += stride) {
    const int64_t e = (i + 1 < m) ? starts[i + 1] : n;
    const unsigned long long len = (unsigned long long)(e - starts[i]);
    mx = mx > len ? mx : len;
  }
  for (int off = 32; off; off >>= 1) {
    const unsigned long long o = __shfl_xor(mx, off, 64);
    mx = mx > o ? mx : o;
  }
  if ((threadIdx.x % 64) == 0) atomicMax(out_max, mx);
}
