// bigslice_amd._C — torch extension binding the CDNA4 kernels:
//   hash_columns      (K3 murmur3 row hash)
//   hash_partition /  (K4 fused hash+histogram+scatter partitioner)
//   scatter_by_partition
//   groupby           (K9 hash-aggregate)
//   radix_argsort     (K6 device radix sort)
//
// Single translation unit: the kernel files are included directly.

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "hash_partition.hip"
#include "groupby.hip"
#include "sort.hip"
#include "radix.hip"
#include "runs.hip"
#include "vecagg.hip"
#include "strings.hip"

namespace {

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ",                     \
                hipGetErrorString(_e));                              \
  } while (0)

int32_t dtype_code(const torch::Tensor& t) {
  switch (t.scalar_type()) {
    case torch::kInt8: return DT_I8;
    case torch::kUInt8: return DT_U8;
    case torch::kInt16: return DT_I16;
    case torch::kInt32: return DT_I32;
    case torch::kInt64: return DT_I64;
    case torch::kFloat32: return DT_F32;
    case torch::kFloat64: return DT_F64;
    case torch::kBool: return DT_BOOL;
    case torch::kUInt32: return DT_U32;
    case torch::kUInt64: return DT_U64;
    default:
      TORCH_CHECK(false, "unsupported column dtype ", t.scalar_type());
  }
}

KeyCols make_key_cols(const std::vector<torch::Tensor>& cols) {
  TORCH_CHECK(!cols.empty() && (int)cols.size() <= MAX_KEY_COLS,
              "1..", MAX_KEY_COLS, " key columns supported");
  KeyCols k;
  k.n = (int)cols.size();
  for (size_t i = 0; i < cols.size(); ++i) {
    TORCH_CHECK(cols[i].is_cuda() && cols[i].is_contiguous(),
                "key columns must be contiguous device tensors");
    k.cols[i] = {cols[i].data_ptr(), dtype_code(cols[i])};
  }
  return k;
}

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

constexpr int32_t MAX_LDS_PARTS = 4096;

// Rows per block for the partition kernels: target ~4096 workgroups so
// the grid comfortably fills 256 CUs across 8 XCDs, with a floor that
// keeps per-block LDS histogram work worthwhile.
int64_t rows_per_block(int64_t n) {
  int64_t rpb = (n + 4095) / 4096;
  if (rpb < 1024) rpb = 1024;
  return rpb;
}

// ---------------------------------------------------------------- hash

torch::Tensor hash_columns(std::vector<torch::Tensor> cols, int64_t seed) {
  KeyCols k = make_key_cols(cols);
  int64_t n = cols[0].size(0);
  auto out = torch::empty({n}, cols[0].options().dtype(torch::kUInt32));
  if (n == 0) return out;
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 65535);
  hipLaunchKernelGGL(k_hash_rows, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), k, n, (uint32_t)seed,
                     out.data_ptr<uint32_t>());
  HIP_CHECK(hipGetLastError());
  return out;
}

// ----------------------------------------------------------- partition

std::tuple<std::vector<torch::Tensor>, torch::Tensor> partition_common(
    const std::vector<torch::Tensor>& cols, torch::Tensor pids,
    torch::Tensor block_hist, int64_t n, int64_t nparts, int64_t nblocks,
    int64_t rpb) {
  // Destination offsets: one small kernel computes per-block reserved
  // ranges + partition totals (stays on device).
  auto opts64 = cols[0].options().dtype(torch::kInt64);
  auto block_off = torch::empty({nblocks * nparts}, opts64);
  auto part_counts = torch::empty({nparts}, opts64);
  // 1024 threads = 16 waves; the kernel assigns one wave per partition
  hipLaunchKernelGGL(k_part_offsets, dim3(1), dim3(1024), 0,
                     current_stream(), block_hist.data_ptr<uint32_t>(),
                     nblocks, (int32_t)nparts,
                     block_off.data_ptr<int64_t>(),
                     part_counts.data_ptr<int64_t>());
  HIP_CHECK(hipGetLastError());

  ScatterCols sc;
  sc.n = (int)cols.size();
  TORCH_CHECK(sc.n <= MAX_COLS, "at most ", MAX_COLS, " columns");
  std::vector<torch::Tensor> dst;
  for (int c = 0; c < sc.n; ++c) {
    TORCH_CHECK(cols[c].is_cuda() && cols[c].is_contiguous(),
                "columns must be contiguous device tensors");
    dst.push_back(torch::empty_like(cols[c]));
    sc.src[c] = {cols[c].data_ptr(), dtype_code(cols[c])};
    sc.dst[c] = {dst[c].data_ptr(), dtype_code(cols[c])};
  }
  size_t lds = (size_t)nparts * sizeof(uint32_t);
  hipLaunchKernelGGL(k_part_scatter, dim3((uint32_t)nblocks), dim3(THREADS),
                     lds, current_stream(), sc,
                     pids.data_ptr<int32_t>(), n, (int32_t)nparts,
                     rpb, block_off.data_ptr<int64_t>());
  HIP_CHECK(hipGetLastError());
  return {dst, part_counts.cpu()};
}

std::tuple<std::vector<torch::Tensor>, torch::Tensor> hash_partition(
    std::vector<torch::Tensor> cols, std::vector<torch::Tensor> key_cols,
    int64_t nparts, int64_t seed) {
  TORCH_CHECK(nparts >= 1 && nparts <= MAX_LDS_PARTS,
              "nparts must be in [1, ", MAX_LDS_PARTS, "]");
  KeyCols k = make_key_cols(key_cols);
  int64_t n = cols.at(0).size(0);
  int64_t rpb = rows_per_block(n);
  int64_t nblocks = (n + rpb - 1) / rpb;
  auto pids = torch::empty({n}, cols[0].options().dtype(torch::kInt32));
  auto block_hist = torch::empty({nblocks * nparts},
                                 cols[0].options().dtype(torch::kUInt32));
  size_t lds = (size_t)nparts * sizeof(uint32_t);
  hipLaunchKernelGGL(k_part_hist, dim3((uint32_t)nblocks), dim3(THREADS),
                     lds, current_stream(), k, n, (int32_t)nparts,
                     (uint32_t)seed, rpb,
                     pids.data_ptr<int32_t>(),
                     block_hist.data_ptr<uint32_t>());
  HIP_CHECK(hipGetLastError());
  return partition_common(cols, pids, block_hist, n, nparts, nblocks, rpb);
}

std::tuple<std::vector<torch::Tensor>, torch::Tensor> scatter_by_partition(
    std::vector<torch::Tensor> cols, torch::Tensor pids, int64_t nparts) {
  TORCH_CHECK(nparts >= 1 && nparts <= MAX_LDS_PARTS);
  TORCH_CHECK(pids.scalar_type() == torch::kInt32 && pids.is_cuda());
  pids = pids.contiguous();
  int64_t n = cols.at(0).size(0);
  int64_t rpb = rows_per_block(n);
  int64_t nblocks = (n + rpb - 1) / rpb;
  auto block_hist = torch::empty({nblocks * nparts},
                                 cols[0].options().dtype(torch::kUInt32));
  size_t lds = (size_t)nparts * sizeof(uint32_t);
  hipLaunchKernelGGL(k_pids_hist, dim3((uint32_t)nblocks), dim3(THREADS),
                     lds, current_stream(), pids.data_ptr<int32_t>(), n,
                     (int32_t)nparts, rpb,
                     block_hist.data_ptr<uint32_t>());
  HIP_CHECK(hipGetLastError());
  return partition_common(cols, pids, block_hist, n, nparts, nblocks, rpb);
}

// ------------------------------------------------------------- groupby

torch::Tensor agg_identity(torch::Tensor like, int32_t agg, int64_t size) {
  auto opt = like.options();
  switch (agg) {
    case AGG_SUM: return torch::zeros({size}, opt);
    case AGG_PROD: return torch::ones({size}, opt);
    case AGG_MIN:
      if (like.is_floating_point())
        return torch::full({size}, std::numeric_limits<double>::infinity(),
                           opt);
      return torch::full({size},
                         like.scalar_type() == torch::kInt64
                             ? std::numeric_limits<int64_t>::max()
                             : (int64_t)std::numeric_limits<int32_t>::max(),
                         opt);
    default:  // AGG_MAX
      if (like.is_floating_point())
        return torch::full({size}, -std::numeric_limits<double>::infinity(),
                           opt);
      return torch::full({size},
                         like.scalar_type() == torch::kInt64
                             ? std::numeric_limits<int64_t>::min()
                             : (int64_t)std::numeric_limits<int32_t>::min(),
                         opt);
  }
}

ValCols make_val_cols(const std::vector<torch::Tensor>& vals,
                      const std::vector<torch::Tensor>& tabs,
                      const std::vector<int64_t>& aggs) {
  ValCols vc;
  vc.n = (int)vals.size();
  TORCH_CHECK(vals.size() == tabs.size() && vals.size() == aggs.size());
  TORCH_CHECK((int)vals.size() <= MAX_COLS);
  for (size_t c = 0; c < vals.size(); ++c) {
    TORCH_CHECK(vals[c].is_cuda() && vals[c].is_contiguous());
    TORCH_CHECK(tabs[c].scalar_type() == vals[c].scalar_type());
    vc.src[c] = {vals[c].data_ptr(), dtype_code(vals[c])};
    vc.tab[c] = {tabs[c].data_ptr(), dtype_code(tabs[c])};
    vc.agg[c] = (int32_t)aggs[c];
  }
  return vc;
}

// Streaming insert into a caller-owned table (tkeys: cap+1 slots filled
// with the sentinel; tabs[c]: cap+1 slots at the agg identity; flags:
// int32[2] = {sentinel_seen, overflow}).  Multiple insert calls
// accumulate into the same table; the host checks flags[1] after
// compaction and regrows on overflow.
void groupby_insert(torch::Tensor keys, std::vector<torch::Tensor> vals,
                    std::vector<int64_t> aggs, torch::Tensor tkeys,
                    std::vector<torch::Tensor> tabs, torch::Tensor flags,
                    int64_t max_probes) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  int64_t n = keys.size(0);
  if (n == 0) return;
  int64_t cap = tkeys.size(0) - 1;
  TORCH_CHECK(cap > 0 && (cap & (cap - 1)) == 0, "capacity must be 2^k");
  ValCols vc = make_val_cols(vals, tabs, aggs);
  const uint32_t seed = 0x9acb0442u;  // reference combiner hashSeed
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_groupby_insert, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), keys.data_ptr<int64_t>(), n, vc,
                     tkeys.data_ptr<int64_t>(), cap, seed,
                     flags.data_ptr<int32_t>(),
                     flags.data_ptr<int32_t>() + 1, max_probes);
  HIP_CHECK(hipGetLastError());
}

torch::Tensor slot_pids(torch::Tensor keys, int64_t cap,
                        int64_t nparts) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  int64_t n = keys.size(0);
  auto pids = torch::empty({n}, keys.options().dtype(torch::kInt32));
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_slot_pids, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), keys.data_ptr<int64_t>(), n, cap,
                     (int32_t)nparts, 0x9acb0442u,
                     pids.data_ptr<int32_t>());
  HIP_CHECK(hipGetLastError());
  return pids;
}

// Two-level LDS insert (single int64 SUM value): see groupby.hip.
void groupby_insert_lds(torch::Tensor keys, torch::Tensor vals,
                        torch::Tensor tkeys, torch::Tensor tab,
                        torch::Tensor flags, int64_t max_probes,
                        int64_t force, int64_t target_blocks,
                        torch::Tensor hits) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  TORCH_CHECK(vals.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  vals = vals.contiguous();
  int64_t n = keys.size(0);
  if (n == 0) return;
  int64_t cap = tkeys.size(0) - 1;
  TORCH_CHECK(cap > 0 && (cap & (cap - 1)) == 0);
  const uint32_t seed = 0x9acb0442u;
  // grid-stride geometry (matches the one-level kernel); target_blocks
  // caps the grid so the LDS flush volume stays bounded
  if (target_blocks <= 0) target_blocks = 32768;
  int64_t nblocks = std::min<int64_t>((n + THREADS - 1) / THREADS,
                                      target_blocks);
  int64_t rpb = 0;  // unused by the grid-stride kernel
  hipLaunchKernelGGL(k_groupby_insert_sum_i64_lds, dim3((uint32_t)nblocks),
                     dim3(THREADS), 0, current_stream(),
                     keys.data_ptr<int64_t>(), vals.data_ptr<int64_t>(), n,
                     tkeys.data_ptr<int64_t>(),
                     (long long*)tab.data_ptr<int64_t>(), cap, seed,
                     flags.data_ptr<int32_t>(),
                     flags.data_ptr<int32_t>() + 1, max_probes, rpb,
                     (int32_t)force,
                     hits.numel() ? (uint32_t*)hits.data_ptr() : nullptr);
  HIP_CHECK(hipGetLastError());
}

// Multi-row software-pipelined one-level insert (single int64 SUM
// value): see groupby.hip.
void groupby_insert_mlp(torch::Tensor keys, torch::Tensor vals,
                        torch::Tensor tkeys, torch::Tensor tab,
                        torch::Tensor flags, int64_t max_probes) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  TORCH_CHECK(vals.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  vals = vals.contiguous();
  int64_t n = keys.size(0);
  if (n == 0) return;
  int64_t cap = tkeys.size(0) - 1;
  TORCH_CHECK(cap > 0 && (cap & (cap - 1)) == 0);
  const uint32_t seed = 0x9acb0442u;
  int64_t nblocks = std::min<int64_t>(
      (n + THREADS * 4 - 1) / (THREADS * 4), 32768);
  hipLaunchKernelGGL(k_groupby_insert_sum_i64_mlp, dim3((uint32_t)nblocks),
                     dim3(THREADS), 0, current_stream(),
                     keys.data_ptr<int64_t>(), vals.data_ptr<int64_t>(), n,
                     tkeys.data_ptr<int64_t>(),
                     (long long*)tab.data_ptr<int64_t>(), cap, seed,
                     flags.data_ptr<int32_t>(),
                     flags.data_ptr<int32_t>() + 1, max_probes);
  HIP_CHECK(hipGetLastError());
}

// Packed-slot fast path (single int64 SUM value): see groupby.hip.
void groupby_insert_packed(torch::Tensor keys, torch::Tensor vals,
                           torch::Tensor table, torch::Tensor flags,
                           int64_t max_probes) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  TORCH_CHECK(vals.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  vals = vals.contiguous();
  int64_t n = keys.size(0);
  if (n == 0) return;
  int64_t cap = table.size(0) / 2 - 1;
  TORCH_CHECK(cap > 0 && (cap & (cap - 1)) == 0);
  const uint32_t seed = 0x9acb0442u;
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_groupby_insert_packed_sum_i64, dim3(blocks),
                     dim3(THREADS), 0, current_stream(),
                     keys.data_ptr<int64_t>(), vals.data_ptr<int64_t>(), n,
                     table.data_ptr<int64_t>(), cap, seed,
                     flags.data_ptr<int32_t>(),
                     flags.data_ptr<int32_t>() + 1, max_probes);
  HIP_CHECK(hipGetLastError());
}

// Replicated-insert experiment (see k_groupby_insert_packed_rep):
// `table` holds nrep packed tables back to back.
void groupby_insert_packed_rep(torch::Tensor keys, torch::Tensor vals,
                               torch::Tensor table, int64_t nrep,
                               torch::Tensor flags, int64_t max_probes) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  keys = keys.contiguous();
  vals = vals.contiguous();
  int64_t n = keys.size(0);
  if (n == 0) return;
  int64_t cap = table.size(0) / (2 * nrep) - 1;
  TORCH_CHECK(cap > 0 && (cap & (cap - 1)) == 0);
  const uint32_t seed = 0x9acb0442u;
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_groupby_insert_packed_rep, dim3(blocks),
                     dim3(THREADS), 0, current_stream(),
                     keys.data_ptr<int64_t>(), vals.data_ptr<int64_t>(), n,
                     table.data_ptr<int64_t>(), cap, (int32_t)nrep, seed,
                     flags.data_ptr<int32_t>(),
                     flags.data_ptr<int32_t>() + 1, max_probes);
  HIP_CHECK(hipGetLastError());
}

std::vector<torch::Tensor> groupby_compact_packed(torch::Tensor table,
                                                  torch::Tensor cursor) {
  int64_t cap = table.size(0) / 2 - 1;
  auto out_keys = torch::empty({cap}, table.options());
  auto out_vals = torch::empty({cap}, table.options());
  int64_t spb = (cap + 2047) / 2048;
  if (spb < THREADS) spb = THREADS;
  int blocks = (int)((cap + spb - 1) / spb);
  hipLaunchKernelGGL(k_groupby_compact_packed, dim3(blocks), dim3(THREADS),
                     0, current_stream(), table.data_ptr<int64_t>(), cap,
                     spb, out_keys.data_ptr<int64_t>(),
                     out_vals.data_ptr<int64_t>(),
                     (unsigned long long*)cursor.data_ptr());
  HIP_CHECK(hipGetLastError());
  return {out_keys, out_vals};
}

torch::Tensor alloc_packed_table(int64_t cap, torch::Tensor like) {
  auto table = torch::empty({2 * (cap + 1)}, like.options());
  int64_t nslots = cap + 1;
  int blocks = (int)std::min<int64_t>((nslots + THREADS - 1) / THREADS,
                                      16384);
  hipLaunchKernelGGL(k_fill_packed_slots, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), table.data_ptr<int64_t>(), nslots);
  HIP_CHECK(hipGetLastError());
  return table;
}

// Compact used slots into freshly-allocated output arrays; returns
// [out_keys(cap), out_vals...(cap)]; cursor (u64[1], zeroed by caller)
// receives the used-slot count — the caller narrows after reading it.
std::vector<torch::Tensor> groupby_compact(torch::Tensor tkeys,
                                           std::vector<torch::Tensor> tabs,
                                           torch::Tensor cursor) {
  int64_t cap = tkeys.size(0) - 1;
  CompactCols cc;
  cc.n = (int)tabs.size();
  std::vector<torch::Tensor> out;
  auto out_keys = torch::empty({cap}, tkeys.options());
  out.push_back(out_keys);
  for (size_t c = 0; c < tabs.size(); ++c) {
    out.push_back(torch::empty({cap}, tabs[c].options()));
    cc.tab[c] = {tabs[c].data_ptr(), dtype_code(tabs[c])};
    cc.out[c] = {out[c + 1].data_ptr(), dtype_code(tabs[c])};
  }
  // ~2048 blocks: fills the chip while keeping cursor atomics rare.
  int64_t spb = (cap + 2047) / 2048;
  if (spb < THREADS) spb = THREADS;
  int blocks = (int)((cap + spb - 1) / spb);
  hipLaunchKernelGGL(k_groupby_compact, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), tkeys.data_ptr<int64_t>(), cap, spb,
                     cc, out_keys.data_ptr<int64_t>(),
                     (unsigned long long*)cursor.data_ptr());
  HIP_CHECK(hipGetLastError());
  return out;
}

// ---------------------------------------------------------------- sort


// ---------------------------------------------------------------------
// Hand-written LSD radix sort orchestration (kernels in radix.hip):
// one fused histogram prepass (all digit positions in one read), then
// one decoupled-lookback scatter kernel per non-constant digit.
// Set BIGSLICE_SORT_ROCPRIM=1 to force the rocPRIM path (A/B harness).

// Re-read each call so one process can A/B implementations.
static bool force_rocprim_sort() {
  const char* e = getenv("BIGSLICE_SORT_ROCPRIM");
  return e && e[0] == '1';
}

// Geometry A/B harness: BIGSLICE_RADIX_VARIANT selects (tile, block).
// Re-read each call so one process can A/B variants back-to-back.
static int radix_variant() {
  const char* e = getenv("BIGSLICE_RADIX_VARIANT");
  return e ? atoi(e) : 0;
}

template <typename K, int HAS_VAL, int TILE, int BLOCK, int SPLIT,
          int PERSIST = 0, int NT = 0, int CHUNKED = 0, int WPS = 8>
static void hand_radix_passes(const torch::Tensor& keys,
                              const torch::Tensor& vals,
                              torch::Tensor& keys_out,
                              torch::Tensor& vals_out,
                              const std::vector<int>& passes,
                              int npass, uint64_t bias,
                              hipStream_t stream) {
  const int64_t n = keys.size(0);
  const int P = (int)passes.size();
  const int64_t ntiles = (n + TILE - 1) / TILE;
  auto state = torch::empty({ntiles * RDX_RADIX},
                            keys.options().dtype(torch::kInt64));
  // hist ping-pong + digit_base, all device-resident: after the first
  // pass's histogram, each scatter tallies the NEXT pass's histogram
  // itself, so no further prepass and no host syncs between passes.
  auto i64 = keys.options().dtype(torch::kInt64);
  auto hist_a = torch::zeros({RDX_RADIX}, i64);
  auto hist_b = torch::zeros({RDX_RADIX}, i64);
  auto dbase = torch::empty({RDX_RADIX}, i64);
  {
    const int64_t nb = std::min<int64_t>((n + 255) / 256, 4096);
    hipLaunchKernelGGL((k_radix_hist_one<K>), dim3((int)nb), dim3(256),
                       0, stream, (const K*)keys.data_ptr(), n,
                       passes[0] * 8,
                       (unsigned long long*)hist_a.data_ptr<int64_t>());
    HIP_CHECK(hipGetLastError());
  }
  torch::Tensor keys_tmp, vals_tmp;
  if (P > 1) {
    keys_tmp = torch::empty_like(keys);
    if (HAS_VAL) vals_tmp = torch::empty_like(vals);
  }
  const K* src_k = (const K*)keys.data_ptr();
  const int64_t* src_v =
      HAS_VAL ? (const int64_t*)vals.data_ptr() : nullptr;
  torch::Tensor hist_cur = hist_a, hist_next = hist_b;
  for (int i = 0; i < P; ++i) {
    const bool to_out = ((P - i) % 2) == 1;
    K* dst_k = (K*)(to_out ? keys_out : keys_tmp).data_ptr();
    int64_t* dst_v =
        HAS_VAL ? (int64_t*)(to_out ? vals_out : vals_tmp).data_ptr()
                : nullptr;
    const int xm = (passes[i] == npass - 1) ? 0x80 : 0;
    hipLaunchKernelGGL(
        k_radix_scan_hist, dim3(1), dim3(RDX_RADIX), 0, stream,
        (const unsigned long long*)hist_cur.data_ptr<int64_t>(),
        (unsigned long long*)dbase.data_ptr<int64_t>(), xm);
    HIP_CHECK(hipGetLastError());
    const int next_shift = (i + 1 < P) ? passes[i + 1] * 8 : -1;
    if (next_shift >= 0)
      HIP_CHECK(hipMemsetAsync(hist_next.data_ptr(), 0,
                               RDX_RADIX * 8, stream));
    HIP_CHECK(hipMemsetAsync(state.data_ptr(), 0,
                             (size_t)ntiles * RDX_RADIX * 8, stream));
    if (CHUNKED) {
      hipLaunchKernelGGL(
          (k_radix_scatter_chunked<K, HAS_VAL, TILE, BLOCK, WPS>),
          dim3((int)ntiles), dim3(BLOCK), 0, stream, src_k, dst_k,
          src_v, dst_v, n, passes[i] * 8, bias,
          (const unsigned long long*)dbase.data_ptr<int64_t>(),
          (unsigned long long*)state.data_ptr<int64_t>(), next_shift,
          (unsigned long long*)hist_next.data_ptr<int64_t>());
    } else if (PERSIST) {
      const int grid = (int)std::min<int64_t>(ntiles, 256);
      hipLaunchKernelGGL(
          (k_radix_scatter_persist<K, HAS_VAL, TILE, BLOCK>),
          dim3(grid), dim3(BLOCK), 0, stream, src_k, dst_k, src_v,
          dst_v, n, passes[i] * 8, bias,
          (const unsigned long long*)dbase.data_ptr<int64_t>(),
          (unsigned long long*)state.data_ptr<int64_t>(), next_shift,
          (unsigned long long*)hist_next.data_ptr<int64_t>());
    } else {
      hipLaunchKernelGGL(
          (k_radix_scatter<K, HAS_VAL, TILE, BLOCK, SPLIT, NT>),
          dim3((int)ntiles),
          dim3(BLOCK), 0, stream, src_k, dst_k, src_v, dst_v, n,
          passes[i] * 8, bias,
          (const unsigned long long*)dbase.data_ptr<int64_t>(),
          (unsigned long long*)state.data_ptr<int64_t>(), next_shift,
          (unsigned long long*)hist_next.data_ptr<int64_t>());
    }
    HIP_CHECK(hipGetLastError());
    src_k = dst_k;
    src_v = dst_v;
    std::swap(hist_cur, hist_next);
  }
}

// pass-skip probe: byte position p is constant across all keys iff the
// bitwise AND and OR of the keys agree on that byte — one
// pure-bandwidth reduction replaces a full histogram prepass AND the
// old aminmax end_bit probe.
template <typename K>
static std::vector<int> probe_passes(const torch::Tensor& keys) {
  const int64_t n = keys.size(0);
  auto stream = current_stream();
  const int npass = (int)keys.element_size();
  auto i64 = keys.options().dtype(torch::kInt64);
  auto andor = torch::empty({2}, i64);
  andor[0] = -1;  // all ones
  andor[1] = 0;
  // 512 grid-strided blocks saturate HBM with the 8-deep load burst
  // while keeping the same-address atomic tail at ~1 op/block
  const int64_t nb = std::min<int64_t>((n + 2047) / 2048, 512);
  hipLaunchKernelGGL(
      (k_radix_andor<K>), dim3((int)nb), dim3(256), 0, stream,
      (const K*)keys.data_ptr(), n,
      (unsigned long long*)andor.data_ptr<int64_t>(),
      (unsigned long long*)(andor.data_ptr<int64_t>() + 1));
  HIP_CHECK(hipGetLastError());
  auto andor_h = andor.cpu();  // the one host sync
  const uint64_t band = (uint64_t)andor_h[0].item<int64_t>();
  const uint64_t bor = (uint64_t)andor_h[1].item<int64_t>();
  std::vector<int> passes;
  for (int p = 0; p < npass; ++p) {
    if (((band >> (p * 8)) & 0xFF) != ((bor >> (p * 8)) & 0xFF))
      passes.push_back(p);  // constant digit: pass skipped
  }
  return passes;
}

template <typename K, int HAS_VAL>
static void hand_radix_sort(const torch::Tensor& keys,
                            const torch::Tensor& vals,
                            torch::Tensor& keys_out,
                            torch::Tensor& vals_out,
                            const std::vector<int>& passes) {
  auto stream = current_stream();
  const int width = (int)keys.element_size() * 8;
  const int npass = width / 8;
  const uint64_t bias = 1ull << (width - 1);
  switch (radix_variant()) {
    case 1:
      hand_radix_passes<K, HAS_VAL, 4096, 512, 0>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 2:
      hand_radix_passes<K, HAS_VAL, 8192, 1024, 0>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 5:
      hand_radix_passes<K, HAS_VAL, 4096, 512, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 6:
      hand_radix_passes<K, HAS_VAL, 3584, 512, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 8:
      hand_radix_passes<K, HAS_VAL, 8192, 1024, 0, 0, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 7:
      hand_radix_passes<K, HAS_VAL, 8192, 1024, 0, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 9:
      // chunked reorder (rocPRIM-occupancy experiment, measured
      // SLOWER: see k_radix_scatter_chunked's header comment)
      hand_radix_passes<K, HAS_VAL, 4096, 512, 0, 0, 0, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 10:
      hand_radix_passes<K, HAS_VAL, 8192, 1024, 0, 0, 0, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
      break;
    case 0:
    case 3:
    default:
      // default: split reorder, 8K tiles, two workgroups per CU
      hand_radix_passes<K, HAS_VAL, 8192, 512, 1>(
          keys, vals, keys_out, vals_out, passes, npass, bias, stream);
  }
}

// Production dispatch: the AND/OR probe picks the executed byte set.
// A CONTIGUOUS set runs on the in-tree rocPRIM onesweep with the bit
// range trimmed at both ends (measured fastest on uniform keys); a set
// with interior constant bytes runs on the hand-written scatter, which
// skips arbitrary bytes.  BIGSLICE_SORT_HAND=1 forces the hand path,
// BIGSLICE_SORT_ROCPRIM=1 forces full-control rocPRIM (A/B harness).
static bool force_hand_sort() {
  const char* e = getenv("BIGSLICE_SORT_HAND");
  return e && e[0] == '1';
}

static bool probed_sortable(const torch::Tensor& keys) {
  return keys.scalar_type() == torch::kInt64 ||
         keys.scalar_type() == torch::kInt32;
}

static bool use_hand_path(const std::vector<int>& passes) {
  if (force_rocprim_sort()) return false;
  if (force_hand_sort()) return true;
  const bool contiguous =
      (int)passes.size() == passes.back() - passes.front() + 1;
  return !contiguous;
}

torch::Tensor radix_sort_keys(torch::Tensor keys, bool probe) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous());
  int64_t n = keys.size(0);
  auto keys_out = torch::empty_like(keys);
  int begin_bit = 0, end_bit = (int)keys.element_size() * 8;
  // probe=false: full-width rocPRIM with no byte-constancy probe (the
  // probe's host readback is a stream sync — callers that must stay
  // async, like the cardinality sample, opt out)
  if (probe && n > 0 && probed_sortable(keys)) {
    std::vector<int> passes =
        keys.scalar_type() == torch::kInt64
            ? probe_passes<int64_t>(keys) : probe_passes<int32_t>(keys);
    if (passes.empty()) {
      keys_out.copy_(keys);
      return keys_out;
    }
    if (use_hand_path(passes)) {
      torch::Tensor nil;
      if (keys.scalar_type() == torch::kInt64)
        hand_radix_sort<int64_t, 0>(keys, nil, keys_out, nil, passes);
      else
        hand_radix_sort<int32_t, 0>(keys, nil, keys_out, nil, passes);
      return keys_out;
    }
    begin_bit = passes.front() * 8;
    end_bit = (passes.back() + 1) * 8;
  }
  auto run = [&](auto fn) {
    size_t temp_bytes = 0;
    fn(keys.data_ptr(), keys_out.data_ptr(), n, begin_bit, end_bit,
       nullptr, temp_bytes, current_stream());
    auto temp = torch::empty({(int64_t)temp_bytes},
                             keys.options().dtype(torch::kUInt8));
    fn(keys.data_ptr(), keys_out.data_ptr(), n, begin_bit, end_bit,
       temp.data_ptr(), temp_bytes, current_stream());
  };
  switch (keys.scalar_type()) {
    case torch::kInt64: run(radix_sort_keys_int64_t); break;
    case torch::kInt32: run(radix_sort_keys_int32_t); break;
    case torch::kFloat32: run(radix_sort_keys_float); break;
    case torch::kFloat64: run(radix_sort_keys_double); break;
    default:
      TORCH_CHECK(false, "radix_sort_keys: unsupported dtype");
  }
  return keys_out;
}

// Direct (key, value) sort for 2-column frames: no permutation, no
// gather passes.  The value column is bit-cast to int64 (it does not
// participate in ordering).
std::vector<torch::Tensor> radix_sort_kv(torch::Tensor keys,
                                         torch::Tensor vals) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous());
  TORCH_CHECK(vals.is_cuda() && vals.is_contiguous());
  TORCH_CHECK(vals.element_size() == 8, "8-byte values only");
  int64_t n = keys.size(0);
  auto keys_out = torch::empty_like(keys);
  auto vals_out = torch::empty_like(vals);
  int begin_bit = 0, end_bit = (int)keys.element_size() * 8;
  if (n > 0 && probed_sortable(keys)) {
    std::vector<int> passes =
        keys.scalar_type() == torch::kInt64
            ? probe_passes<int64_t>(keys) : probe_passes<int32_t>(keys);
    if (passes.empty()) {
      keys_out.copy_(keys);
      vals_out.copy_(vals);
      return {keys_out, vals_out};
    }
    if (use_hand_path(passes)) {
      if (keys.scalar_type() == torch::kInt64)
        hand_radix_sort<int64_t, 1>(keys, vals, keys_out, vals_out,
                                    passes);
      else
        hand_radix_sort<int32_t, 1>(keys, vals, keys_out, vals_out,
                                    passes);
      return {keys_out, vals_out};
    }
    begin_bit = passes.front() * 8;
    end_bit = (passes.back() + 1) * 8;
  }
  auto run = [&](auto fn) {
    size_t temp_bytes = 0;
    fn(keys.data_ptr(), keys_out.data_ptr(),
       (const int64_t*)vals.data_ptr(), (int64_t*)vals_out.data_ptr(), n,
       begin_bit, end_bit, nullptr, temp_bytes, current_stream());
    auto temp = torch::empty({(int64_t)temp_bytes},
                             keys.options().dtype(torch::kUInt8));
    fn(keys.data_ptr(), keys_out.data_ptr(),
       (const int64_t*)vals.data_ptr(), (int64_t*)vals_out.data_ptr(), n,
       begin_bit, end_bit, temp.data_ptr(), temp_bytes,
       current_stream());
  };
  switch (keys.scalar_type()) {
    case torch::kInt64: run(radix_sort_pairs_int64_t); break;
    case torch::kInt32: run(radix_sort_pairs_int32_t); break;
    case torch::kFloat32: run(radix_sort_pairs_float); break;
    case torch::kFloat64: run(radix_sort_pairs_double); break;
    default:
      TORCH_CHECK(false, "radix_sort_kv: unsupported key dtype");
  }
  return {keys_out, vals_out};
}

// K16: (unique_keys, aggregates, count) over key-sorted pairs via
// rocPRIM reduce-by-key: deterministic segmented reduction (fixed tree
// order — float sums are run-to-run reproducible, unlike atomics).
// count comes back as a 1-element device tensor; the caller slices
// after one sync.  code: 0=sum 1=min 2=max.
std::vector<torch::Tensor> segment_reduce_sorted(torch::Tensor keys,
                                                 torch::Tensor vals,
                                                 int64_t code) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous());
  TORCH_CHECK(vals.is_cuda() && vals.is_contiguous());
  TORCH_CHECK(keys.scalar_type() == torch::kInt64,
              "segment_reduce_sorted: int64 keys only");
  int64_t n = keys.size(0);
  auto uniq = torch::empty_like(keys);
  auto aggs = torch::empty_like(vals);
  auto count = torch::zeros({1}, keys.options());
  if (n == 0) return {uniq, aggs, count};
  auto run = [&](auto fn) {
    size_t temp_bytes = 0;
    fn(keys.data_ptr<int64_t>(), vals.data_ptr(), n,
       uniq.data_ptr<int64_t>(), aggs.data_ptr(),
       count.data_ptr<int64_t>(), (int)code, nullptr, temp_bytes,
       current_stream());
    auto temp = torch::empty({(int64_t)temp_bytes},
                             keys.options().dtype(torch::kUInt8));
    fn(keys.data_ptr<int64_t>(), vals.data_ptr(), n,
       uniq.data_ptr<int64_t>(), aggs.data_ptr(),
       count.data_ptr<int64_t>(), (int)code, temp.data_ptr(), temp_bytes,
       current_stream());
  };
  switch (vals.scalar_type()) {
    case torch::kInt64: run(reduce_by_key_int64_t); break;
    case torch::kInt32: run(reduce_by_key_int32_t); break;
    case torch::kFloat32: run(reduce_by_key_float); break;
    case torch::kFloat64: run(reduce_by_key_double); break;
    default:
      TORCH_CHECK(false, "segment_reduce_sorted: unsupported val dtype");
  }
  return {uniq, aggs, count};
}

std::vector<torch::Tensor> segment_sum_sorted_i64(torch::Tensor keys,
                                                  torch::Tensor vals) {
  return segment_reduce_sorted(keys, vals, 0);
}

// K16 fast path (runs.hip): int64 segmented reduction over sorted
// values with precomputed run boundaries.  The caller computes runs
// once via runs_sorted and reuses them for every value column (the
// rocPRIM reduce_by_key path redoes the key scan per column).
torch::Tensor segment_reduce_runs(torch::Tensor vals,
                                  torch::Tensor starts, int64_t n,
                                  int64_t code) {
  TORCH_CHECK(vals.is_cuda() && vals.is_contiguous() &&
              vals.scalar_type() == torch::kInt64,
              "segment_reduce_runs: contiguous int64 values required");
  TORCH_CHECK(starts.is_cuda() && starts.is_contiguous());
  TORCH_CHECK(code >= 0 && code <= 2,
              "segment_reduce_runs: sum/min/max only");
  const int64_t m = starts.size(0);
  auto out = torch::empty({m}, vals.options());
  if (m == 0) return out;
  const int64_t nb = std::min<int64_t>((m + 255) / 256, 4096);
  hipLaunchKernelGGL(k_segreduce_i64, dim3((int)nb), dim3(256), 0,
                     current_stream(), vals.data_ptr<int64_t>(),
                     starts.data_ptr<int64_t>(), m, n, (int)code,
                     out.data_ptr<int64_t>());
  HIP_CHECK(hipGetLastError());
  return out;
}

// (count, max_run_length) of a runs_sorted result in ONE device
// tensor, so the host guard pays a single readback.
torch::Tensor runs_guard(torch::Tensor starts, torch::Tensor cnt,
                         int64_t n) {
  TORCH_CHECK(starts.is_cuda() && starts.is_contiguous() &&
              cnt.is_cuda());
  auto out = torch::zeros({2}, starts.options());
  out.index_put_({0}, cnt[0]);
  hipLaunchKernelGGL(k_runs_guard, dim3(512), dim3(256), 0,
                     current_stream(),
                     starts.data_ptr<int64_t>(),
                     cnt.data_ptr<int64_t>(), n,
                     (unsigned long long*)(out.data_ptr<int64_t>() + 1));
  HIP_CHECK(hipGetLastError());
  return out;
}

// K18: (unique_keys, run_starts, count) of a SORTED key array in one
// pass (runs.hip: ballot compaction + single-channel lookback;
// measured 5.4x the rocPRIM reduce-by-key fallback kept behind
// BIGSLICE_RUNS_ROCPRIM=1).  count is a 1-element device tensor; the
// caller slices after one sync.
std::vector<torch::Tensor> runs_sorted(torch::Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous() &&
              keys.scalar_type() == torch::kInt64,
              "runs_sorted: contiguous int64 device keys required");
  const int64_t n = keys.size(0);
  auto uniq = torch::empty_like(keys);
  auto starts = torch::empty_like(keys);
  auto count = torch::zeros({1}, keys.options());
  if (n == 0) return {uniq, starts, count};
  const char* rp = getenv("BIGSLICE_RUNS_ROCPRIM");
  if (rp && rp[0] == '1') {
    size_t temp_bytes = 0;
    runs_sorted_i64(keys.data_ptr<int64_t>(), n,
                    uniq.data_ptr<int64_t>(), starts.data_ptr<int64_t>(),
                    count.data_ptr<int64_t>(), nullptr, temp_bytes,
                    current_stream());
    auto temp = torch::empty({(int64_t)temp_bytes},
                             keys.options().dtype(torch::kUInt8));
    runs_sorted_i64(keys.data_ptr<int64_t>(), n,
                    uniq.data_ptr<int64_t>(), starts.data_ptr<int64_t>(),
                    count.data_ptr<int64_t>(), temp.data_ptr(),
                    temp_bytes, current_stream());
    return {uniq, starts, count};
  }
  auto stream = current_stream();
  // 64K-row tiles amortize per-block fixed costs on large inputs;
  // 8K-row tiles keep small inputs on >=2 blocks/CU
  const int ipt = (n >= (16 << 20)) ? 256 : 32;
  const int64_t tile = (int64_t)RUNS_BLOCK * ipt;
  const int64_t ntiles = (n + tile - 1) / tile;
  auto state = torch::empty({ntiles},
                            keys.options().dtype(torch::kInt64));
  HIP_CHECK(hipMemsetAsync(state.data_ptr(), 0, (size_t)ntiles * 8,
                           stream));
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3((int)ntiles), dim3(RUNS_BLOCK), 0,
                       stream, keys.data_ptr<int64_t>(), n,
                       uniq.data_ptr<int64_t>(),
                       starts.data_ptr<int64_t>(),
                       count.data_ptr<int64_t>(),
                       (unsigned long long*)state.data_ptr<int64_t>());
  };
  if (ipt == 256)
    launch(k_runs_sorted<256>);
  else
    launch(k_runs_sorted<32>);
  HIP_CHECK(hipGetLastError());
  return {uniq, starts, count};
}

// K17: murmur3 over variable-length byte rows (device strings).
torch::Tensor hash_bytes(torch::Tensor bytes, torch::Tensor offsets,
                         int64_t seed) {
  TORCH_CHECK(bytes.is_cuda() && bytes.is_contiguous() &&
              bytes.scalar_type() == torch::kUInt8);
  TORCH_CHECK(offsets.is_cuda() && offsets.is_contiguous() &&
              offsets.scalar_type() == torch::kInt64);
  int64_t n = offsets.size(0) - 1;
  auto out = torch::empty({n}, bytes.options().dtype(torch::kUInt32));
  if (n == 0) return out;
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_hash_bytes, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), bytes.data_ptr<uint8_t>(),
                     offsets.data_ptr<int64_t>(), n, (uint32_t)seed,
                     out.data_ptr<uint32_t>());
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor hash_bytes64(torch::Tensor bytes, torch::Tensor offsets,
                           int64_t seed_hi, int64_t seed_lo) {
  TORCH_CHECK(bytes.is_cuda() && bytes.is_contiguous() &&
              bytes.scalar_type() == torch::kUInt8);
  TORCH_CHECK(offsets.is_cuda() && offsets.is_contiguous() &&
              offsets.scalar_type() == torch::kInt64);
  int64_t n = offsets.size(0) - 1;
  auto out = torch::empty({n}, bytes.options().dtype(torch::kInt64));
  if (n == 0) return out;
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 32768);
  hipLaunchKernelGGL(k_hash_bytes64, dim3(blocks), dim3(THREADS), 0,
                     current_stream(), bytes.data_ptr<uint8_t>(),
                     offsets.data_ptr<int64_t>(), n, (uint32_t)seed_hi,
                     (uint32_t)seed_lo, out.data_ptr<int64_t>());
  HIP_CHECK(hipGetLastError());
  return out;
}

// Vector-aggregate experiment (see vecagg.hip): colsum of [N,16] f32.
torch::Tensor colsum16(torch::Tensor x, bool mfma) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kFloat32 && x.dim() == 2 &&
              x.size(1) == 16);
  int64_t n = x.size(0);
  auto out = torch::zeros({16}, x.options());
  int blocks = (int)std::min<int64_t>((n + THREADS - 1) / THREADS, 8192);
  if (mfma)
    hipLaunchKernelGGL(k_colsum_mfma, dim3(blocks), dim3(THREADS), 0,
                       current_stream(), x.data_ptr<float>(), n,
                       out.data_ptr<float>());
  else
    hipLaunchKernelGGL(k_colsum_valu, dim3(blocks), dim3(THREADS), 0,
                       current_stream(), x.data_ptr<float>(), n,
                       out.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor radix_argsort(torch::Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous());
  int64_t n = keys.size(0);
  auto perm_in = torch::arange(n, keys.options().dtype(torch::kInt64));
  auto perm_out = torch::empty_like(perm_in);
  auto keys_out = torch::empty_like(keys);
  int begin_bit = 0, end_bit = (int)keys.element_size() * 8;
  if (n > 0 && probed_sortable(keys)) {
    std::vector<int> passes =
        keys.scalar_type() == torch::kInt64
            ? probe_passes<int64_t>(keys) : probe_passes<int32_t>(keys);
    if (passes.empty()) return perm_in;
    if (use_hand_path(passes)) {
      if (keys.scalar_type() == torch::kInt64)
        hand_radix_sort<int64_t, 1>(keys, perm_in, keys_out, perm_out,
                                    passes);
      else
        hand_radix_sort<int32_t, 1>(keys, perm_in, keys_out, perm_out,
                                    passes);
      return perm_out;
    }
    begin_bit = passes.front() * 8;
    end_bit = (passes.back() + 1) * 8;
  }
  auto run = [&](auto fn) {
    size_t temp_bytes = 0;
    fn(keys.data_ptr(), keys_out.data_ptr(),
       perm_in.data_ptr<int64_t>(), perm_out.data_ptr<int64_t>(), n,
       begin_bit, end_bit, nullptr, temp_bytes, current_stream());
    auto temp = torch::empty({(int64_t)temp_bytes},
                             keys.options().dtype(torch::kUInt8));
    fn(keys.data_ptr(), keys_out.data_ptr(),
       perm_in.data_ptr<int64_t>(), perm_out.data_ptr<int64_t>(), n,
       begin_bit, end_bit, temp.data_ptr(), temp_bytes,
       current_stream());
  };
  switch (keys.scalar_type()) {
    case torch::kInt64: run(radix_sort_pairs_int64_t); break;
    case torch::kInt32: run(radix_sort_pairs_int32_t); break;
    case torch::kFloat32: run(radix_sort_pairs_float); break;
    case torch::kFloat64: run(radix_sort_pairs_double); break;
    default:
      TORCH_CHECK(false, "radix_argsort: unsupported dtype");
  }
  return perm_out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("hash_columns", &hash_columns, "murmur3 row hash (K3)");
  m.def("hash_partition", &hash_partition,
        "fused hash+histogram+scatter partitioner (K4)");
  m.def("scatter_by_partition", &scatter_by_partition,
        "scatter rows by precomputed partition ids");
  m.def("groupby_insert", &groupby_insert,
        "hash-aggregate insert pass (K9)");
  m.def("groupby_insert_lds", &groupby_insert_lds,
        "two-level LDS+global insert (int64 sum)");
  m.def("groupby_insert_mlp", &groupby_insert_mlp,
        "multi-row software-pipelined insert (int64 sum)");
  m.def("groupby_insert_packed_rep", &groupby_insert_packed_rep,
        "replicated packed insert (fabric-ceiling experiment)");
  m.def("groupby_insert_packed", &groupby_insert_packed,
        "packed-slot insert (int64 sum fast path)");
  m.def("groupby_compact_packed", &groupby_compact_packed,
        "packed-slot compaction");
  m.def("alloc_packed_table", &alloc_packed_table,
        "allocate+init packed table");
  m.def("groupby_compact", &groupby_compact,
        "hash-aggregate table compaction (K9)");
  m.def("agg_identity", &agg_identity, "aggregation identity fill");
  m.def("radix_argsort", &radix_argsort, "device radix argsort (K6)");
  m.def("radix_sort_keys", &radix_sort_keys, "device radix key sort",
        py::arg("keys"), py::arg("probe") = true);
  m.def("segment_sum_sorted", &segment_sum_sorted_i64,
        "reduce-by-key sum over sorted int64 pairs (K16)");
  m.def("runs_sorted", &runs_sorted,
        "run boundaries of sorted keys (K18)");
  m.def("segment_reduce_runs", &segment_reduce_runs,
        "int64 segmented reduce with precomputed runs (K16 fast path)");
  m.def("runs_guard", &runs_guard,
        "(count, max run length) of a runs_sorted result");
  m.def("segment_reduce_sorted", &segment_reduce_sorted,
        "typed reduce-by-key over sorted pairs (K16; deterministic)");
  m.def("hash_bytes", &hash_bytes,
        "murmur3-32 over varlen byte rows (K17, device strings)");
  m.def("hash_bytes64", &hash_bytes64,
        "64-bit two-seed murmur3 dictionary ids (K17)");
  m.def("colsum16", &colsum16,
        "vector-aggregate experiment: [N,16] f32 colsum, VALU vs MFMA");
  m.def("radix_sort_kv", &radix_sort_kv,
        "direct (key, 8-byte value) radix sort");
  m.def("slot_pids", &slot_pids, "table-slot-range partition ids");
}
