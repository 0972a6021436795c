// K6: hand-written LSD radix sort for gfx950 (CDNA4), (key, value) pairs.
//
// The reference's in-core sort (sortio/sort.go:31-76) runs here as a
// cheap AND/OR byte-constancy probe + one decoupled-lookback scatter
// kernel per NON-CONSTANT 8-bit digit; each scatter also tallies the
// next pass's histogram in its idle issue slots, so a pass is a single
// ~32 B/row sweep with no histogram prepass.
//
// Dispatch (ext.hip): the probe's executed-byte set drives both
// engines — a CONTIGUOUS set runs the in-tree rocPRIM onesweep with
// begin/end bits trimmed (measured 1.2x faster than this kernel on
// uniform full-width keys: 33.6 vs 40.6 ms for 500M int64 pairs); a
// set with interior constant bytes runs THIS kernel, which skips
// arbitrary bytes (measured 1.37x faster than rocPRIM there: 21.2 vs
// 29.0 ms on low16|random<<40 keys).  profiles/sort_variants.txt holds
// the measured matrix; BIGSLICE_SORT_{HAND,ROCPRIM}=1 force either.
//
// CDNA4 specifics (see /opt/skills/guides/MI355X_MICROARCH.md):
// * wave64 ballot match ranking (8 ballots -> peer mask) gives stable
//   in-wave digit ranks without LDS atomics;
// * a 1024-thread workgroup per CU stages the 8192-row tile in a
//   128 KiB LDS pair buffer (160 KiB LDS/CU) and rewrites the
//   scattered digit writes as digit-contiguous bursts (~32 rows/digit);
// * cross-tile prefixes are published as 8-byte {2-bit flag | 62-bit
//   count} granules with agent-scope relaxed atomics — per-XCD L2s are
//   not coherent, so plain loads would read stale flags (guide §G16);
//   blocks dispatch first-to-last, so looking back at lower tile ids
//   cannot deadlock, and speculative batched reads pipeline the walk;
// * global loads burst IPT-deep before the ranking's LDS chain (the
//   interleaved form serializes every load behind a waitcnt).

#include <hip/hip_runtime.h>

#include <cstdint>
#include <type_traits>

#define RDX_RADIX 256
#define RDX_TILE 8192   // default geometry (variant 0)
#define RDX_BLOCK 512
#define RDX_FLAG_AGG 1ull
#define RDX_FLAG_PREFIX 2ull
// Lookback batch width: with one workgroup per CU, ~256 tiles run
// concurrently, so a tile's lookback walks ~256 not-yet-PREFIX
// predecessors.  Walking them with dependent loads serializes ~256
// HBM/L2 latencies; reading B speculatively per round pipelines them.
#define RDX_LOOKBACK_BATCH 8

// ------------------------------------------------------------------ prepass

// Bitwise AND/OR reduction over keys: byte position p is constant
// across all keys (its pass can be SKIPPED) iff AND and OR agree on
// that byte.  Pure-bandwidth replacement for a full histogram prepass.
template <typename K>
__global__ void k_radix_andor(const K* __restrict__ keys, int64_t n,
                              unsigned long long* __restrict__ out_and,
                              unsigned long long* __restrict__ out_or) {
  using U = std::conditional_t<sizeof(K) == 8, uint64_t, uint32_t>;
  U a = ~(U)0, o = 0;
  // 8 independent strided loads per iteration: the reduction is pure
  // bandwidth, so keep many loads in flight per lane
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  const int64_t stride8 = step * 8;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + 7 * step < n; i += stride8) {
    U u0 = (U)keys[i];
    U u1 = (U)keys[i + step];
    U u2 = (U)keys[i + 2 * step];
    U u3 = (U)keys[i + 3 * step];
    U u4 = (U)keys[i + 4 * step];
    U u5 = (U)keys[i + 5 * step];
    U u6 = (U)keys[i + 6 * step];
    U u7 = (U)keys[i + 7 * step];
    a &= (u0 & u1) & (u2 & u3) & (u4 & u5) & (u6 & u7);
    o |= (u0 | u1) | (u2 | u3) | (u4 | u5) | (u6 | u7);
  }
  for (; i < n; i += step) {
    const U u = (U)keys[i];
    a &= u;
    o |= u;
  }
  for (int off = 32; off; off >>= 1) {
    a &= (U)__shfl_xor((unsigned long long)a, off, 64);
    o |= (U)__shfl_xor((unsigned long long)o, off, 64);
  }
  // Block-level LDS reduction -> ONE atomic pair per block.  The
  // earlier per-wave form issued 2 contended same-address atomics per
  // wave; at a 2048-block grid that tail alone ran ~0.4 ms (the whole
  // 0.5 GB read should take ~70 us), measured as 1.1 TB/s on a pure
  // streaming reduction.
  __shared__ unsigned long long red_a[8], red_o[8];
  const int wv = threadIdx.x / 64;
  if ((threadIdx.x % 64) == 0) {
    red_a[wv] = (unsigned long long)a |
                ~(unsigned long long)(U)~(U)0;
    red_o[wv] = (unsigned long long)o;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned long long ba = red_a[0], bo = red_o[0];
    for (int w = 1; w < (int)(blockDim.x / 64); ++w) {
      ba &= red_a[w];
      bo |= red_o[w];
    }
    atomicAnd(out_and, ba);
    atomicOr(out_or, bo);
  }
}

// Digit histogram of ONE byte position (the first executed pass; later
// passes' histograms are computed on the fly inside the scatter).
// Ballot-leader counting: one LDS add per distinct digit per wave.
template <typename K>
__global__ void k_radix_hist_one(const K* __restrict__ keys, int64_t n,
                                 int shift,
                                 unsigned long long* __restrict__ hist) {
  using U = std::conditional_t<sizeof(K) == 8, uint64_t, uint32_t>;
  __shared__ unsigned int lh[RDX_RADIX];
  for (int i = threadIdx.x; i < RDX_RADIX; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i - (threadIdx.x % 64) < n; i += stride) {
    const bool active = i < n;
    const U u = active ? (U)keys[i] : U(0);
    const unsigned int d = (unsigned int)((u >> shift) & 0xFF);
    uint64_t peers = __ballot(active);
    for (int b = 0; b < 8; ++b) {
      const uint64_t bb = __ballot((d >> b) & 1);
      peers &= ((d >> b) & 1) ? bb : ~bb;
    }
    if (active && (peers & lt) == 0)
      atomicAdd(&lh[d], (unsigned int)__popcll(peers));
  }
  __syncthreads();
  for (int i = threadIdx.x; i < RDX_RADIX; i += blockDim.x) {
    if (lh[i]) atomicAdd(&hist[i], (unsigned long long)lh[i]);
  }
}

// Exclusive scan of a 256-bin histogram into digit_base, applying the
// sign-bias bin permutation (xor_mask = 0x80 for the top byte of a
// signed key, else 0).  One block; runs on-device so no host sync sits
// between passes.
__global__ void k_radix_scan_hist(
    const unsigned long long* __restrict__ hist,
    unsigned long long* __restrict__ digit_base, int xor_mask) {
  __shared__ unsigned long long tmp[RDX_RADIX];
  const int d = threadIdx.x;  // 256 threads
  tmp[d] = hist[d ^ xor_mask];
  __syncthreads();
  // simple serial scan by thread 0 (256 adds, once per pass: trivial)
  if (d == 0) {
    unsigned long long run = 0;
    for (int i = 0; i < RDX_RADIX; ++i) {
      const unsigned long long v = tmp[i];
      tmp[i] = run;
      run += v;
    }
  }
  __syncthreads();
  digit_base[d] = tmp[d];
}

// ------------------------------------------------------------------ scatter

// One pass: stable scatter of (key, value) pairs by the digit at `shift`.
// HAS_VAL=0 sorts keys only.  Digit order is unsigned over the biased key
// (bias flips the sign bit of the top byte so signed ints sort correctly).
//
// SPLIT=1 reorders keys and values through ONE reused LDS buffer in two
// phases (key phase records each slot's digit in a byte array; the value
// phase re-loads values from global straight into their slots): the LDS
// footprint halves, so two 512-thread workgroups co-reside per CU
// (16 waves/CU) while keeping 8K-row tiles (32-row digit runs for
// coalesced stores).  SPLIT=0 stages both columns at once (bigger LDS,
// one workgroup).
// SPLIT runs multiple workgroups per CU; its launch bound asks the
// register allocator for the waves/SIMD the LDS footprint admits so
// the extra workgroups actually co-reside (160KB/CU / LDS-per-WG
// workgroups x BLOCK/64 waves each).
template <typename K, int HAS_VAL, int TILE, int BLOCK, int SPLIT,
          int NT = 0>
__launch_bounds__(BLOCK,
                  (SPLIT && TILE <= 4096 && BLOCK == 512)
                      ? 8
                      : ((SPLIT && BLOCK == 512) ? 4 : 1)) __global__
void k_radix_scatter(
    const K* __restrict__ keys_in, K* __restrict__ keys_out,
    const int64_t* __restrict__ vals_in, int64_t* __restrict__ vals_out,
    int64_t n, int shift, uint64_t bias,
    const unsigned long long* __restrict__ digit_base,  // [256] excl.
    unsigned long long* __restrict__ state,              // [ntiles*256]
    int next_shift,  // next pass's byte position, or -1
    unsigned long long* __restrict__ next_hist) {        // [256]
  using U = std::conditional_t<sizeof(K) == 8, uint64_t, uint32_t>;
  constexpr int WAVES = BLOCK / 64;
  constexpr int IPT = TILE / BLOCK;  // items per thread
  // one raw staging buffer: keys [+ values when both staged at once];
  // the SPLIT value phase reuses it as an int64 buffer
  constexpr size_t KBYTES = (size_t)TILE * sizeof(K);
  constexpr size_t VBYTES = (size_t)TILE * 8;
  constexpr size_t BUFBYTES =
      (HAS_VAL && !SPLIT) ? (KBYTES + VBYTES)
                          : ((HAS_VAL && SPLIT && VBYTES > KBYTES)
                                 ? VBYTES
                                 : KBYTES);
  __shared__ __align__(16) unsigned char lds_raw[BUFBYTES];
  K* const lds_keys = (K*)lds_raw;
  int64_t* const lds_vals = (int64_t*)(lds_raw + KBYTES);
  __shared__ unsigned char lds_dig[(HAS_VAL && SPLIT) ? TILE : 1];
  __shared__ unsigned short wavehist[WAVES][RDX_RADIX];
  // next pass's digit counts, tallied in this pass's idle issue slots
  // (the kernel is memory-wait bound) so no histogram prepass is
  // needed for passes after the first
  __shared__ unsigned short wavehist2[WAVES][RDX_RADIX];
  __shared__ unsigned int digit_start[RDX_RADIX];     // excl. in tile
  __shared__ unsigned int wave_tot[WAVES > 4 ? WAVES : 4];
  __shared__ unsigned long long tile_base[RDX_RADIX]; // global base

  const int tile = blockIdx.x;
  const int64_t base = (int64_t)tile * TILE;
  const int64_t rem = n - base;
  const int cnt = rem < TILE ? (int)rem : TILE;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));

  for (int i = threadIdx.x; i < WAVES * RDX_RADIX; i += BLOCK) {
    ((unsigned short*)wavehist)[i] = 0;
    ((unsigned short*)wavehist2)[i] = 0;
  }
  __syncthreads();

  // ---- load keys + stable in-wave ranking (ballot match) ----
  K my_keys[IPT];
  int64_t my_vals[(HAS_VAL && !SPLIT) ? IPT : 1];
  unsigned short my_rank[IPT];
  unsigned char my_dig[IPT];
  const int wrow0 = wave * (64 * IPT);
  // burst the loads first (all IPT in flight) — interleaving them with
  // the ranking's LDS chain makes each iteration block on its own
  // global load (observed in the ISA as GL,waitcnt pairs)
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    my_keys[i] = (r < cnt)
                     ? (NT ? __builtin_nontemporal_load(keys_in + base + r)
                           : keys_in[base + r])
                     : K(0);
  }
  if (HAS_VAL && !SPLIT) {
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      my_vals[i] = (r < cnt)
                       ? (NT ? __builtin_nontemporal_load(vals_in + base + r)
                             : vals_in[base + r])
                       : 0;
    }
  }
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    const bool active = r < cnt;
    K k = my_keys[i];
    const unsigned int d =
        (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
    my_dig[i] = (unsigned char)d;
    uint64_t peers = __ballot(active);
    for (int b = 0; b < 8; ++b) {
      const uint64_t bb = __ballot((d >> b) & 1);
      peers &= ((d >> b) & 1) ? bb : ~bb;
    }
    const unsigned int before = wavehist[wave][d];  // broadcast read
    const unsigned int rank_here = (unsigned int)__popcll(peers & lt);
    if (active && (peers & lt) == 0)  // leader: first active peer
      wavehist[wave][d] =
          (unsigned short)(before + (unsigned int)__popcll(peers));
    my_rank[i] = (unsigned short)(before + rank_here);
    if (next_shift >= 0) {
      // tally the NEXT pass's (raw) digit — hidden in idle VALU slots
      const unsigned int d2 =
          (unsigned int)(((U)k >> next_shift) & 0xFF);
      uint64_t p2 = __ballot(active);
      for (int b = 0; b < 8; ++b) {
        const uint64_t bb = __ballot((d2 >> b) & 1);
        p2 &= ((d2 >> b) & 1) ? bb : ~bb;
      }
      if (active && (p2 & lt) == 0)
        wavehist2[wave][d2] = (unsigned short)(
            wavehist2[wave][d2] + (unsigned int)__popcll(p2));
    }
  }
  __syncthreads();

  // ---- wave offsets + digit scan + early publish (one thread/digit;
  // the cross-wave carry resolves inside one extra barrier) ----
  unsigned int my_total = 0;  // valid for threadIdx.x < 256
  if (threadIdx.x < RDX_RADIX) {
    const int d = threadIdx.x;
    unsigned int sum = 0;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) {
      const unsigned int c = wavehist[w][d];
      wavehist[w][d] = (unsigned short)sum;
      sum += c;
    }
    my_total = sum;
    // publish tile totals early so successors' lookback resolves
    const unsigned long long pub =
        ((unsigned long long)(tile == 0 ? RDX_FLAG_PREFIX : RDX_FLAG_AGG)
         << 62) | (unsigned long long)sum;
    __hip_atomic_store(&state[(uint64_t)tile * RDX_RADIX + d], pub,
                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    // exclusive scan of digit totals across the 4 scan waves
    unsigned int incl = sum;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      const unsigned int up = __shfl_up(incl, off, 64);
      if (lane >= off) incl += up;
    }
    digit_start[d] = incl - sum;  // wave-local exclusive, fixed below
    if (lane == 63) wave_tot[d / 64] = incl;
  }
  __syncthreads();
  if (threadIdx.x < RDX_RADIX) {
    unsigned int carry = 0;
    const int sw = threadIdx.x / 64;
    for (int w = 0; w < sw; ++w) carry += wave_tot[w];
    digit_start[threadIdx.x] += carry;
  }
  __syncthreads();

  // ---- lookback (threads < 256) OVERLAPPED with key reorder (all
  // threads; the scan waves reorder after their walk resolves).  One
  // barrier then covers both tile_base and the reordered buffer. ----
  unsigned long long lb_excl = 0;
  if (threadIdx.x < RDX_RADIX && tile > 0) {
    const int d = threadIdx.x;
    int j = tile - 1;
    bool done = false;
    while (!done) {
      // speculative batch: independent loads pipeline the walk over
      // concurrently-running predecessors
      unsigned long long v[RDX_LOOKBACK_BATCH];
      const int m = (j + 1) < RDX_LOOKBACK_BATCH ? (j + 1)
                                                 : RDX_LOOKBACK_BATCH;
#pragma unroll
      for (int q = 0; q < RDX_LOOKBACK_BATCH; ++q) {
        if (q < m)
          v[q] = __hip_atomic_load(
              &state[(uint64_t)(j - q) * RDX_RADIX + d],
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      }
      for (int q = 0; q < m; ++q) {
        unsigned long long x = v[q];
        while ((x >> 62) == 0) {  // unpublished: re-poll this one
          __builtin_amdgcn_s_sleep(1);
          x = __hip_atomic_load(
              &state[(uint64_t)(j - q) * RDX_RADIX + d],
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        lb_excl += x & ((1ull << 62) - 1);
        if ((x >> 62) == RDX_FLAG_PREFIX) {
          done = true;
          break;
        }
      }
      j -= m;
      if (j < 0) done = true;
    }
    // publish our inclusive prefix for successors
    __hip_atomic_store(
        &state[(uint64_t)tile * RDX_RADIX + d],
        (RDX_FLAG_PREFIX << 62) | (lb_excl + my_total),
        __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
  if (threadIdx.x < RDX_RADIX)
    tile_base[threadIdx.x] = digit_base[threadIdx.x] + lb_excl;

  // ---- reorder keys into LDS (digit-contiguous) ----
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    if (r < cnt) {
      const int d = my_dig[i];
      const unsigned int pos =
          digit_start[d] + wavehist[wave][d] + my_rank[i];
      lds_keys[pos] = my_keys[i];
      if (HAS_VAL && !SPLIT) lds_vals[pos] = my_vals[i];
    }
  }
  __syncthreads();

  // ---- digit-contiguous key store-out ----
  for (int j = threadIdx.x; j < cnt; j += BLOCK) {
    const K k = lds_keys[j];
    const unsigned int d =
        (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
    const int64_t gpos = (int64_t)(tile_base[d] + j - digit_start[d]);
    if (NT) {
      __builtin_nontemporal_store(k, keys_out + gpos);
      if (HAS_VAL && !SPLIT)
        __builtin_nontemporal_store(lds_vals[j], vals_out + gpos);
    } else {
      keys_out[gpos] = k;
      if (HAS_VAL && !SPLIT) vals_out[gpos] = lds_vals[j];
    }
    if (HAS_VAL && SPLIT) lds_dig[j] = (unsigned char)d;
  }

  // ---- SPLIT value phase: reuse the key buffer for values ----
  if (HAS_VAL && SPLIT) {
    __syncthreads();  // key buffer fully drained
    int64_t* lds_v2 = (int64_t*)lds_raw;
    int64_t vv[IPT];
#pragma unroll
    for (int i = 0; i < IPT; ++i) {  // burst loads (see above)
      const int r = wrow0 + i * 64 + lane;
      vv[i] = (r < cnt) ? vals_in[base + r] : 0;
    }
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      if (r < cnt) {
        const int d = my_dig[i];
        const unsigned int pos =
            digit_start[d] + wavehist[wave][d] + my_rank[i];
        lds_v2[pos] = vv[i];
      }
    }
    __syncthreads();
    for (int j = threadIdx.x; j < cnt; j += BLOCK) {
      const unsigned int d = lds_dig[j];
      const int64_t gpos = (int64_t)(tile_base[d] + j - digit_start[d]);
      vals_out[gpos] = lds_v2[j];
    }
  }

  // flush the next pass's digit counts (final since before the first
  // barrier; ordering is covered by the earlier __syncthreads)
  if (next_shift >= 0 && threadIdx.x < RDX_RADIX) {
    unsigned int s = 0;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) s += wavehist2[w][threadIdx.x];
    if (s) atomicAdd(&next_hist[threadIdx.x], (unsigned long long)s);
  }
}

// Persistent-block variant: one workgroup per CU loops over tiles
// (block b handles tiles b, b+G, ...), issuing the NEXT tile's
// register loads right after this tile's reorder frees the registers —
// the ~900-cycle HBM load latency hides under the current tile's
// lookback + store-out instead of serializing at the head of each
// tile.  Geometry: (8192-row tiles, 1024 threads, both columns staged;
// 150 KiB LDS -> one workgroup of 16 waves per CU).
template <typename K, int HAS_VAL, int TILE, int BLOCK>
__launch_bounds__(BLOCK, 1) __global__ void k_radix_scatter_persist(
    const K* __restrict__ keys_in, K* __restrict__ keys_out,
    const int64_t* __restrict__ vals_in, int64_t* __restrict__ vals_out,
    int64_t n, int shift, uint64_t bias,
    const unsigned long long* __restrict__ digit_base,
    unsigned long long* __restrict__ state, int next_shift,
    unsigned long long* __restrict__ next_hist) {
  using U = std::conditional_t<sizeof(K) == 8, uint64_t, uint32_t>;
  constexpr int WAVES = BLOCK / 64;
  constexpr int IPT = TILE / BLOCK;
  __shared__ __align__(16) K lds_keys[TILE];
  __shared__ int64_t lds_vals[HAS_VAL ? TILE : 1];
  __shared__ unsigned short wavehist[WAVES][RDX_RADIX];
  // u32: accumulates across ALL of this block's tiles (flushed once)
  __shared__ unsigned int wavehist2[WAVES][RDX_RADIX];
  __shared__ unsigned int digit_start[RDX_RADIX];
  __shared__ unsigned int wave_tot[WAVES > 4 ? WAVES : 4];
  __shared__ unsigned long long tile_base[RDX_RADIX];

  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int G = gridDim.x;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  const int wrow0 = wave * (64 * IPT);

  for (int i = threadIdx.x; i < WAVES * RDX_RADIX; i += BLOCK) {
    ((unsigned short*)wavehist)[i] = 0;
    ((unsigned int*)wavehist2)[i] = 0;
  }

  K my_keys[IPT];
  int64_t my_vals[HAS_VAL ? IPT : 1];
  unsigned short my_rank[IPT];
  unsigned char my_dig[IPT];

  auto load_regs = [&](int64_t t) {
    const int64_t base = (int64_t)t * TILE;
    const int cnt = (n - base) < TILE ? (int)(n - base) : TILE;
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      my_keys[i] = (r < cnt) ? keys_in[base + r] : K(0);
    }
    if (HAS_VAL) {
#pragma unroll
      for (int i = 0; i < IPT; ++i) {
        const int r = wrow0 + i * 64 + lane;
        my_vals[i] = (r < cnt) ? vals_in[base + r] : 0;
      }
    }
  };

  int64_t t = blockIdx.x;
  if (t < ntiles) load_regs(t);
  __syncthreads();  // wavehist zeroed

  for (; t < ntiles; t += G) {
    const int64_t base = (int64_t)t * TILE;
    const int cnt = (n - base) < TILE ? (int)(n - base) : TILE;

    // ---- rank (stable ballot match) + next-pass digit tally ----
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      const bool active = r < cnt;
      const K k = my_keys[i];
      const unsigned int d =
          (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
      my_dig[i] = (unsigned char)d;
      uint64_t peers = __ballot(active);
      for (int b = 0; b < 8; ++b) {
        const uint64_t bb = __ballot((d >> b) & 1);
        peers &= ((d >> b) & 1) ? bb : ~bb;
      }
      const unsigned int before = wavehist[wave][d];
      const unsigned int rank_here = (unsigned int)__popcll(peers & lt);
      if (active && (peers & lt) == 0)
        wavehist[wave][d] =
            (unsigned short)(before + (unsigned int)__popcll(peers));
      my_rank[i] = (unsigned short)(before + rank_here);
      if (next_shift >= 0) {
        const unsigned int d2 =
            (unsigned int)(((U)k >> next_shift) & 0xFF);
        uint64_t p2 = __ballot(active);
        for (int b = 0; b < 8; ++b) {
          const uint64_t bb = __ballot((d2 >> b) & 1);
          p2 &= ((d2 >> b) & 1) ? bb : ~bb;
        }
        if (active && (p2 & lt) == 0)
          wavehist2[wave][d2] += (unsigned int)__popcll(p2);
      }
    }
    __syncthreads();

    // ---- wave offsets + digit scan + early publish ----
    unsigned int my_total = 0;
    if (threadIdx.x < RDX_RADIX) {
      const int d = threadIdx.x;
      unsigned int sum = 0;
#pragma unroll
      for (int w = 0; w < WAVES; ++w) {
        const unsigned int c = wavehist[w][d];
        wavehist[w][d] = (unsigned short)sum;
        sum += c;
      }
      my_total = sum;
      const unsigned long long pub =
          ((unsigned long long)(t == 0 ? RDX_FLAG_PREFIX : RDX_FLAG_AGG)
           << 62) | (unsigned long long)sum;
      __hip_atomic_store(&state[(uint64_t)t * RDX_RADIX + d], pub,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      unsigned int incl = sum;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        const unsigned int up = __shfl_up(incl, off, 64);
        if (lane >= off) incl += up;
      }
      digit_start[d] = incl - sum;
      if (lane == 63) wave_tot[d / 64] = incl;
    }
    __syncthreads();
    if (threadIdx.x < RDX_RADIX) {
      unsigned int carry = 0;
      const int sw = threadIdx.x / 64;
      for (int w = 0; w < sw; ++w) carry += wave_tot[w];
      digit_start[threadIdx.x] += carry;
    }
    __syncthreads();

    // ---- lookback (threads < 256) overlapped with reorder ----
    unsigned long long lb_excl = 0;
    if (threadIdx.x < RDX_RADIX && t > 0) {
      const int d = threadIdx.x;
      int64_t j = t - 1;
      bool done = false;
      while (!done) {
        unsigned long long v[RDX_LOOKBACK_BATCH];
        const int m = (j + 1) < RDX_LOOKBACK_BATCH
                          ? (int)(j + 1) : RDX_LOOKBACK_BATCH;
#pragma unroll
        for (int q = 0; q < RDX_LOOKBACK_BATCH; ++q) {
          if (q < m)
            v[q] = __hip_atomic_load(
                &state[(uint64_t)(j - q) * RDX_RADIX + d],
                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        for (int q = 0; q < m; ++q) {
          unsigned long long x = v[q];
          while ((x >> 62) == 0) {
            __builtin_amdgcn_s_sleep(1);
            x = __hip_atomic_load(
                &state[(uint64_t)(j - q) * RDX_RADIX + d],
                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
          }
          lb_excl += x & ((1ull << 62) - 1);
          if ((x >> 62) == RDX_FLAG_PREFIX) {
            done = true;
            break;
          }
        }
        j -= m;
        if (j < 0) done = true;
      }
      __hip_atomic_store(
          &state[(uint64_t)t * RDX_RADIX + d],
          (RDX_FLAG_PREFIX << 62) | (lb_excl + my_total),
          __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    }
    if (threadIdx.x < RDX_RADIX)
      tile_base[threadIdx.x] = digit_base[threadIdx.x] + lb_excl;

    // ---- reorder into LDS ----
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      if (r < cnt) {
        const int d = my_dig[i];
        const unsigned int pos =
            digit_start[d] + wavehist[wave][d] + my_rank[i];
        lds_keys[pos] = my_keys[i];
        if (HAS_VAL) lds_vals[pos] = my_vals[i];
      }
    }

    // ---- prefetch the NEXT tile's rows (registers are free now; the
    // loads fly under the store-out below) ----
    if (t + G < ntiles) load_regs(t + G);
    __syncthreads();  // reorder + tile_base visible

    // ---- digit-contiguous store-out + wavehist re-zero ----
    for (int j = threadIdx.x; j < cnt; j += BLOCK) {
      const K k = lds_keys[j];
      const unsigned int d =
          (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
      const int64_t gpos =
          (int64_t)(tile_base[d] + j - digit_start[d]);
      keys_out[gpos] = k;
      if (HAS_VAL) vals_out[gpos] = lds_vals[j];
    }
    if (threadIdx.x < RDX_RADIX) {
#pragma unroll
      for (int w = 0; w < WAVES; ++w)
        wavehist[w][threadIdx.x] = 0;
    }
    __syncthreads();  // LDS drained + wavehist clean for next tile
  }

  if (next_shift >= 0 && threadIdx.x < RDX_RADIX) {
    unsigned int s = 0;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) s += wavehist2[w][threadIdx.x];
    if (s) atomicAdd(&next_hist[threadIdx.x], (unsigned long long)s);
  }
}

// Chunked-reorder variant (MEASURED SLOWER — kept as the record of a
// tested hypothesis, variants 9/10): instead of staging the WHOLE
// tile's pairs in LDS (150 KiB -> one workgroup per CU), slots stream
// through a BLOCK-sized LDS window in TILE/BLOCK chunks (conditional
// writes into the window, one coalesced read per thread per chunk).
// LDS drops to ~30 KiB so 2-4 workgroups co-reside (up to 32
// waves/CU) — the occupancy structure of rocPRIM's onesweep.
// Outcome on 500 M int64 kv pairs (profiles/sort_variants.txt):
// 21-22 ms (20-bit) / 54-58 ms (full) vs the monolithic SPLIT
// default's 15.6 / 41.2 ms — the chunked window makes every thread
// re-scan all IPT items per chunk (IPT^2 = 64 conditional LDS writes
// instead of IPT) and pays 2*IPT barriers per tile, which costs more
// than the 2-4x occupancy buys on this bandwidth-bound kernel; a
// 6-waves/SIMD spill-free build was slower still (58-61 ms).  So
// rocPRIM's uniform-key edge is NOT its occupancy; the production
// dispatch (probe -> rocPRIM for contiguous byte sets, hand kernel
// for interior constant bytes) stands.
template <typename K, int HAS_VAL, int TILE, int BLOCK, int WPS = 8>
__launch_bounds__(BLOCK, WPS) __global__
void k_radix_scatter_chunked(
    const K* __restrict__ keys_in, K* __restrict__ keys_out,
    const int64_t* __restrict__ vals_in, int64_t* __restrict__ vals_out,
    int64_t n, int shift, uint64_t bias,
    const unsigned long long* __restrict__ digit_base,
    unsigned long long* __restrict__ state, int next_shift,
    unsigned long long* __restrict__ next_hist) {
  using U = std::conditional_t<sizeof(K) == 8, uint64_t, uint32_t>;
  constexpr int WAVES = BLOCK / 64;
  constexpr int IPT = TILE / BLOCK;
  __shared__ __align__(16) int64_t lds_win[BLOCK + 1];  // +1 spill slot
  __shared__ unsigned short wavehist[WAVES][RDX_RADIX];
  __shared__ unsigned short wavehist2[WAVES][RDX_RADIX];
  __shared__ unsigned int digit_start[RDX_RADIX];
  __shared__ unsigned int wave_tot[WAVES > 4 ? WAVES : 4];
  __shared__ unsigned long long win_base[RDX_RADIX];

  const int tile = blockIdx.x;
  const int64_t base = (int64_t)tile * TILE;
  const int64_t rem = n - base;
  const int cnt = rem < TILE ? (int)rem : TILE;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const uint64_t lt = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  const int wrow0 = wave * (64 * IPT);

  for (int i = threadIdx.x; i < WAVES * RDX_RADIX; i += BLOCK) {
    ((unsigned short*)wavehist)[i] = 0;
    ((unsigned short*)wavehist2)[i] = 0;
  }
  __syncthreads();

  // ---- burst loads + stable ballot-match rank + next-pass tally ----
  K my_keys[IPT];
  unsigned short my_rank[IPT];
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    my_keys[i] = (r < cnt) ? keys_in[base + r] : K(0);
  }
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    const bool active = r < cnt;
    const K k = my_keys[i];
    const unsigned int d =
        (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
    uint64_t peers = __ballot(active);
    for (int b = 0; b < 8; ++b) {
      const uint64_t bb = __ballot((d >> b) & 1);
      peers &= ((d >> b) & 1) ? bb : ~bb;
    }
    const unsigned int before = wavehist[wave][d];
    if (active && (peers & lt) == 0)
      wavehist[wave][d] =
          (unsigned short)(before + (unsigned int)__popcll(peers));
    my_rank[i] =
        (unsigned short)(before + (unsigned int)__popcll(peers & lt));
    if (next_shift >= 0) {
      const unsigned int d2 = (unsigned int)(((U)k >> next_shift) &
                                             0xFF);
      uint64_t p2 = __ballot(active);
      for (int b = 0; b < 8; ++b) {
        const uint64_t bb = __ballot((d2 >> b) & 1);
        p2 &= ((d2 >> b) & 1) ? bb : ~bb;
      }
      if (active && (p2 & lt) == 0)
        wavehist2[wave][d2] = (unsigned short)(
            wavehist2[wave][d2] + (unsigned int)__popcll(p2));
    }
  }
  __syncthreads();

  // ---- digit scan + early publish ----
  unsigned int my_total = 0;
  if (threadIdx.x < RDX_RADIX) {
    const int d = threadIdx.x;
    unsigned int sum = 0;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) {
      const unsigned int c = wavehist[w][d];
      wavehist[w][d] = (unsigned short)sum;
      sum += c;
    }
    my_total = sum;
    const unsigned long long pub =
        ((unsigned long long)(tile == 0 ? RDX_FLAG_PREFIX : RDX_FLAG_AGG)
         << 62) | (unsigned long long)sum;
    __hip_atomic_store(&state[(uint64_t)tile * RDX_RADIX + d], pub,
                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    unsigned int incl = sum;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      const unsigned int up = __shfl_up(incl, off, 64);
      if (lane >= off) incl += up;
    }
    digit_start[d] = incl - sum;
    if (lane == 63) wave_tot[d / 64] = incl;
  }
  __syncthreads();
  if (threadIdx.x < RDX_RADIX) {
    unsigned int carry = 0;
    const int sw = threadIdx.x / 64;
    for (int w = 0; w < sw; ++w) carry += wave_tot[w];
    digit_start[threadIdx.x] += carry;
  }
  __syncthreads();

  // full tile rank (position within the tile's digit-major order)
#pragma unroll
  for (int i = 0; i < IPT; ++i) {
    const int r = wrow0 + i * 64 + lane;
    if (r < cnt) {
      const unsigned int d =
          (unsigned int)((((U)my_keys[i] ^ (U)bias) >> shift) & 0xFF);
      my_rank[i] = (unsigned short)(
          digit_start[d] + wavehist[wave][d] + my_rank[i]);
    } else {
      my_rank[i] = (unsigned short)TILE;  // parked outside any window
    }
  }

  // ---- lookback: global exclusive tile prefix per digit ----
  unsigned long long lb_excl = 0;
  if (threadIdx.x < RDX_RADIX && tile > 0) {
    const int d = threadIdx.x;
    int j = tile - 1;
    bool done = false;
    while (!done) {
      unsigned long long v[RDX_LOOKBACK_BATCH];
      const int m = (j + 1) < RDX_LOOKBACK_BATCH ? (j + 1)
                                                 : RDX_LOOKBACK_BATCH;
#pragma unroll
      for (int q = 0; q < RDX_LOOKBACK_BATCH; ++q) {
        if (q < m)
          v[q] = __hip_atomic_load(
              &state[(uint64_t)(j - q) * RDX_RADIX + d],
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      }
      for (int q = 0; q < m; ++q) {
        unsigned long long x = v[q];
        while ((x >> 62) == 0) {
          __builtin_amdgcn_s_sleep(1);
          x = __hip_atomic_load(
              &state[(uint64_t)(j - q) * RDX_RADIX + d],
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        lb_excl += x & ((1ull << 62) - 1);
        if ((x >> 62) == RDX_FLAG_PREFIX) {
          done = true;
          break;
        }
      }
      j -= m;
      if (j < 0) done = true;
    }
    __hip_atomic_store(
        &state[(uint64_t)tile * RDX_RADIX + d],
        (RDX_FLAG_PREFIX << 62) | (lb_excl + my_total),
        __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
  if (threadIdx.x < RDX_RADIX)
    // store (global digit base - tile digit start): slot j of digit d
    // lands at win_base[d] + j directly
    win_base[threadIdx.x] = digit_base[threadIdx.x] + lb_excl -
                            digit_start[threadIdx.x];
  __syncthreads();

  // ---- chunked key reorder through the BLOCK-sized LDS window ----
  unsigned char my_dig2[IPT];  // digit per item, for the value pass
#pragma unroll
  for (int j = 0; j < IPT; ++j) {
    const unsigned int x = j * BLOCK;
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      // clamped write: ranks outside this window hit the spill slot
      const unsigned int off = (unsigned int)my_rank[i] - x;
      lds_win[off < BLOCK ? off : BLOCK] = (int64_t)my_keys[i];
    }
    __syncthreads();
    const unsigned int slot = x + threadIdx.x;
    if ((int)slot < cnt) {
      const K k = (K)lds_win[threadIdx.x];
      const unsigned int d =
          (unsigned int)((((U)k ^ (U)bias) >> shift) & 0xFF);
      keys_out[win_base[d] + slot] = k;
      if (HAS_VAL) my_dig2[j] = (unsigned char)d;
    }
    __syncthreads();
  }

  // ---- value pass through the same window ----
  if (HAS_VAL) {
    int64_t my_vals[IPT];
#pragma unroll
    for (int i = 0; i < IPT; ++i) {
      const int r = wrow0 + i * 64 + lane;
      my_vals[i] = (r < cnt) ? vals_in[base + r] : 0;
    }
#pragma unroll
    for (int j = 0; j < IPT; ++j) {
      const unsigned int x = j * BLOCK;
#pragma unroll
      for (int i = 0; i < IPT; ++i) {
        const unsigned int off = (unsigned int)my_rank[i] - x;
        lds_win[off < BLOCK ? off : BLOCK] = my_vals[i];
      }
      __syncthreads();
      const unsigned int slot = x + threadIdx.x;
      if ((int)slot < cnt) {
        vals_out[win_base[my_dig2[j]] + slot] = lds_win[threadIdx.x];
      }
      __syncthreads();
    }
  }

  // flush the next pass's digit tallies
  if (next_shift >= 0 && threadIdx.x < RDX_RADIX) {
    unsigned int s = 0;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) s += wavehist2[w][threadIdx.x];
    if (s) atomicAdd(&next_hist[threadIdx.x], (unsigned long long)s);
  }
}
