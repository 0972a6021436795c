// Vector-aggregate experiment (BASELINE north-star names "MFMA for
// vector aggregates"): column-sum of a row-major [N,16] f32 matrix —
// the inner primitive of per-key vector aggregation — implemented two
// ways: plain VALU accumulation and MFMA matrix cores via the
// ones-matrix trick (D = ones[16x4] x X[4x16] makes every row of D the
// 4-row column sum).  At 0.25 FLOP/byte the op is HBM-bound: MFMA's
// 2.5 PFLOP/s cannot beat an ~8 TB/s read stream, which the
// measurement in benchmarks/vecagg_ab.py confirms — documented
// evidence, not a production path (production vector aggregates use
// the rocPRIM reduce-by-key sort-combine).

#include <hip/hip_runtime.h>

#define VA_COLS 16

extern "C" __global__ void k_colsum_valu(const float* __restrict__ x,
                                         int64_t n, float* out) {
  float acc[VA_COLS];
#pragma unroll
  for (int c = 0; c < VA_COLS; ++c) acc[c] = 0.f;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const float* row = x + i * VA_COLS;
#pragma unroll
    for (int c = 0; c < VA_COLS; ++c) acc[c] += row[c];
  }
  __shared__ float lsum[VA_COLS];
  if (threadIdx.x < VA_COLS) lsum[threadIdx.x] = 0.f;
  __syncthreads();
#pragma unroll
  for (int c = 0; c < VA_COLS; ++c) atomicAdd(&lsum[c], acc[c]);
  __syncthreads();
  if (threadIdx.x < VA_COLS) atomicAdd(&out[threadIdx.x],
                                       lsum[threadIdx.x]);
}

typedef float v4f __attribute__((ext_vector_type(4)));

extern "C" __global__ void k_colsum_mfma(const float* __restrict__ x,
                                         int64_t n, float* out) {
  // one wave per 64-thread group; each MFMA consumes a 4-row x 16-col
  // tile: lane L supplies B[k][j] with k=L/16, j=L%16
  const int lane = threadIdx.x & 63;
  const int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x)
                       >> 6;
  const int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int k = lane >> 4, j = lane & 15;
  v4f acc = {0.f, 0.f, 0.f, 0.f};
  const int64_t rows4 = n >> 2;  // full 4-row tiles
  for (int64_t t = wave; t < rows4; t += nwaves) {
    const float b = x[(t * 4 + k) * VA_COLS + j];
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(1.0f, b, acc, 0, 0, 0);
  }
  // every D row holds the column sums; lanes 0..15 carry columns 0..15
  if (lane < 16) atomicAdd(&out[j], acc[0]);
  // tail rows (n % 4) via lanes of the first wave of block 0
  if (blockIdx.x == 0 && threadIdx.x < VA_COLS) {
    float t = 0.f;
    for (int64_t r = rows4 * 4; r < n; ++r)
      t += x[r * VA_COLS + threadIdx.x];
    if (t != 0.f) atomicAdd(&out[threadIdx.x], t);
  }
}
