// K3/K4: murmur3 row hashing and the fused hash+histogram+scatter
// partitioner (the reference's defaultPartitioner + scatter loop,
// exec/compile.go:20-24 / exec/bigmachine.go:960-996, redesigned as
// whole-batch CDNA4 kernels).
//
// Partition pipeline (all device-resident):
//   1. k_part_hist: per-row partition id (hash % nparts) + per-block
//      LDS histogram -> global per-block histogram matrix.
//   2. (host) tiny torch cumsum over the (nblocks x nparts) matrix gives
//      each block a contiguous destination range per partition -- the
//      exact bucket layout an RCCL all-to-allv wants.
//   3. k_part_scatter: re-reads its row range and writes every column's
//      rows into the block's reserved ranges (one fused multi-column
//      pass: reads and writes each byte exactly once).
//
// Launch geometry: 256-thread blocks, ROWS_PER_BLOCK chosen so the grid
// is >> 256 workgroups to fill all 8 XCDs.

#include <hip/hip_runtime.h>

#include "columns.h"

#define THREADS 256

extern "C" __global__ void k_hash_rows(KeyCols keys, int64_t n,
                                       uint32_t seed, uint32_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = hash_row(keys, i, seed);
}

// Pass 1: partition ids + per-block histogram.
extern "C" __global__ void k_part_hist(KeyCols keys, int64_t n,
                                       int32_t nparts, uint32_t seed,
                                       int64_t rows_per_block,
                                       int32_t* pids, uint32_t* block_hist) {
  extern __shared__ uint32_t lhist[];  // nparts counters
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) lhist[p] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * rows_per_block;
  int64_t end = min(start + rows_per_block, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    uint32_t h = hash_row(keys, i, seed);
    int32_t p = (int32_t)(h % (uint32_t)nparts);
    pids[i] = p;
    atomicAdd(&lhist[p], 1u);
  }
  __syncthreads();
  uint32_t* gh = block_hist + (int64_t)blockIdx.x * nparts;
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) gh[p] = lhist[p];
}

// Pass 1 variant for a user-supplied partitioner: pids precomputed.
extern "C" __global__ void k_pids_hist(const int32_t* pids, int64_t n,
                                       int32_t nparts,
                                       int64_t rows_per_block,
                                       uint32_t* block_hist) {
  extern __shared__ uint32_t lhist[];
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) lhist[p] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * rows_per_block;
  int64_t end = min(start + rows_per_block, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    atomicAdd(&lhist[pids[i]], 1u);
  __syncthreads();
  uint32_t* gh = block_hist + (int64_t)blockIdx.x * nparts;
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) gh[p] = lhist[p];
}

// Destination-offset computation: one small kernel replaces a chain of
// torch cumsum/sum launches.  block_off[b][p] = start of block b's
// reserved range in partition p; part_counts[p] = total rows.
// Single workgroup; each thread owns partition columns (strided).
extern "C" __global__ void k_part_offsets(const uint32_t* block_hist,
                                          int64_t nblocks, int32_t nparts,
                                          int64_t* block_off,
                                          int64_t* part_counts) {
  // One wave per partition strip, shuffle-based prefix scan across the
  // block axis.  A per-thread serial walk here is latency-bound at
  // small fan-outs (nparts=8 left 8 lanes chasing 4096 strided loads,
  // ~1.5 ms per call); the wave scan runs the same work in ~10 us.
  __shared__ int64_t totals[4096];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  for (int p = wave; p < nparts; p += nwaves) {
    int64_t carry = 0;
    for (int64_t b0 = 0; b0 < nblocks; b0 += 64) {
      const int64_t b = b0 + lane;
      const int64_t v =
          (b < nblocks) ? (int64_t)block_hist[b * nparts + p] : 0;
      int64_t run = v;  // inclusive wave scan
      for (int off = 1; off < 64; off <<= 1) {
        const int64_t up = __shfl_up(run, off);
        if (lane >= off) run += up;
      }
      if (b < nblocks)  // within-partition exclusive offset
        block_off[b * nparts + p] = carry + run - v;
      carry += __shfl(run, 63);
    }
    totals[p] = carry;
    part_counts[p] = carry;
  }
  __syncthreads();
  if (wave == 0) {  // exclusive scan of partition starts, wave-chunked
    int64_t carry = 0;
    for (int p0 = 0; p0 < nparts; p0 += 64) {
      const int p = p0 + lane;
      const int64_t v = (p < nparts) ? totals[p] : 0;
      int64_t run = v;
      for (int off = 1; off < 64; off <<= 1) {
        const int64_t up = __shfl_up(run, off);
        if (lane >= off) run += up;
      }
      if (p < nparts) totals[p] = carry + run - v;
      carry += __shfl(run, 63);
    }
  }
  __syncthreads();
  const int64_t entries = nblocks * nparts;
  for (int64_t i = threadIdx.x; i < entries; i += blockDim.x)
    block_off[i] += totals[i % nparts];
}

struct ScatterCols {
  ColDesc src[MAX_COLS];
  MutColDesc dst[MAX_COLS];
  int n;
};

// Pass 2: fused multi-column scatter into reserved ranges.
extern "C" __global__ void k_part_scatter(ScatterCols cols,
                                          const int32_t* pids, int64_t n,
                                          int32_t nparts,
                                          int64_t rows_per_block,
                                          const int64_t* block_off) {
  extern __shared__ uint32_t lcnt[];  // per-partition cursor
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) lcnt[p] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * rows_per_block;
  int64_t end = min(start + rows_per_block, n);
  const int64_t* boff = block_off + (int64_t)blockIdx.x * nparts;
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    int32_t p = pids[i];
    uint32_t r = atomicAdd(&lcnt[p], 1u);
    int64_t dsti = boff[p] + r;
    for (int c = 0; c < cols.n; ++c) {
      switch (elt_size(cols.src[c].code)) {
        case 1:
          ((uint8_t*)cols.dst[c].ptr)[dsti] =
              ((const uint8_t*)cols.src[c].ptr)[i];
          break;
        case 2:
          ((uint16_t*)cols.dst[c].ptr)[dsti] =
              ((const uint16_t*)cols.src[c].ptr)[i];
          break;
        case 4:
          ((uint32_t*)cols.dst[c].ptr)[dsti] =
              ((const uint32_t*)cols.src[c].ptr)[i];
          break;
        default:
          ((uint64_t*)cols.dst[c].ptr)[dsti] =
              ((const uint64_t*)cols.src[c].ptr)[i];
          break;
      }
    }
  }
}
