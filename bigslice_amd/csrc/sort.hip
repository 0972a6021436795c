// K6: device radix sort (argsort) via rocPRIM headers compiled into this
// extension (no external library dependency).  Replaces the reference's
// comparison sort of spill chunks (sortio/sort.go:52) with an LSD radix
// sort over whole device batches; the k-way merge of sorted runs (K7/K8)
// consumes its output.

#include <hip/hip_runtime.h>
#include <rocprim/device/device_radix_sort.hpp>

#include <stdexcept>

template <typename K>
static void radix_sort_pairs_impl(const K* keys_in, K* keys_out,
                                  const int64_t* vals_in, int64_t* vals_out,
                                  int64_t n, int begin_bit, int end_bit,
                                  void* temp,
                                  size_t& temp_bytes, hipStream_t stream) {
  hipError_t err = rocprim::radix_sort_pairs(
      temp, temp_bytes, keys_in, keys_out, vals_in, vals_out, (size_t)n,
      begin_bit, end_bit, stream);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::radix_sort_pairs: ") +
                             hipGetErrorString(err));
}

template <typename K>
static void radix_sort_keys_impl(const K* keys_in, K* keys_out, int64_t n,
                                 int begin_bit, int end_bit, void* temp,
                                 size_t& temp_bytes, hipStream_t stream) {
  hipError_t err = rocprim::radix_sort_keys(
      temp, temp_bytes, keys_in, keys_out, (size_t)n, begin_bit, end_bit,
      stream);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::radix_sort_keys: ") +
                             hipGetErrorString(err));
}

// Explicit type dispatch used by ext.hip.
#define INSTANTIATE_SORT(K)                                               \
  void radix_sort_pairs_##K(const void* ki, void* ko, const int64_t* vi,  \
                            int64_t* vo, int64_t n, int begin_bit,        \
                            int end_bit,                                  \
                            void* temp, size_t& temp_bytes,               \
                            hipStream_t s) {                              \
    radix_sort_pairs_impl<K>((const K*)ki, (K*)ko, vi, vo, n, begin_bit,  \
                             end_bit, temp, temp_bytes, s);               \
  }

INSTANTIATE_SORT(int64_t)
INSTANTIATE_SORT(int32_t)
INSTANTIATE_SORT(float)
INSTANTIATE_SORT(double)

#define INSTANTIATE_SORT_KEYS(K)                                          \
  void radix_sort_keys_##K(const void* ki, void* ko, int64_t n,           \
                           int begin_bit, int end_bit, void* temp,        \
                           size_t& temp_bytes,                            \
                           hipStream_t s) {                               \
    radix_sort_keys_impl<K>((const K*)ki, (K*)ko, n, begin_bit, end_bit,  \
                            temp, temp_bytes, s);                         \
  }

INSTANTIATE_SORT_KEYS(int64_t)
INSTANTIATE_SORT_KEYS(int32_t)
INSTANTIATE_SORT_KEYS(float)
INSTANTIATE_SORT_KEYS(double)

// K16: segment reduce over key-sorted (or run-grouped) pairs via
// rocPRIM's tuned reduce-by-key (decoupled-lookback single pass).
// Replaces the mask/nonzero/cumsum torch chain in the sort-combine.
#include <rocprim/device/device_reduce_by_key.hpp>

template <typename V, typename Op>
static void reduce_by_key_impl(const int64_t* keys, const V* vals,
                               int64_t n, int64_t* uniq_out, V* aggs_out,
                               int64_t* count_out, Op op, void* temp,
                               size_t& temp_bytes, hipStream_t s) {
  hipError_t err = rocprim::reduce_by_key(
      temp, temp_bytes, keys, vals, (size_t)n, uniq_out, aggs_out,
      count_out, op, rocprim::equal_to<int64_t>(), s);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::reduce_by_key: ") +
                             hipGetErrorString(err));
}

// agg codes match kernels._AGG_CODES: 0=sum 1=min 2=max 3=prod
#define INSTANTIATE_RBK(V)                                                \
  void reduce_by_key_##V(const int64_t* keys, const void* vals,           \
                         int64_t n, int64_t* uniq_out, void* aggs_out,    \
                         int64_t* count_out, int code, void* temp,        \
                         size_t& temp_bytes, hipStream_t s) {             \
    switch (code) {                                                       \
      case 0:                                                             \
        reduce_by_key_impl(keys, (const V*)vals, n, uniq_out,             \
                           (V*)aggs_out, count_out, rocprim::plus<V>(),   \
                           temp, temp_bytes, s);                          \
        break;                                                            \
      case 1:                                                             \
        reduce_by_key_impl(keys, (const V*)vals, n, uniq_out,             \
                           (V*)aggs_out, count_out,                       \
                           rocprim::minimum<V>(), temp, temp_bytes, s);   \
        break;                                                            \
      case 2:                                                             \
        reduce_by_key_impl(keys, (const V*)vals, n, uniq_out,             \
                           (V*)aggs_out, count_out,                       \
                           rocprim::maximum<V>(), temp, temp_bytes, s);   \
        break;                                                            \
      default:                                                            \
        throw std::runtime_error("reduce_by_key: bad agg code");          \
    }                                                                     \
  }

INSTANTIATE_RBK(int64_t)
INSTANTIATE_RBK(int32_t)
INSTANTIATE_RBK(float)
INSTANTIATE_RBK(double)

// K18 (cogroup assembly): run boundaries of a SORTED key array in ONE
// rocPRIM reduce-by-key pass — values are a counting iterator and the
// reduction is min, so each run's aggregate IS its start index.
// Replaces the ne-mask -> nonzero -> gather chain (3 reads of the key
// array + a bool mask round trip + rocPRIM partition) with a single
// 8 B/row read.
#include <rocprim/iterator/counting_iterator.hpp>

void runs_sorted_i64(const int64_t* keys, int64_t n, int64_t* uniq_out,
                     int64_t* starts_out, int64_t* count_out, void* temp,
                     size_t& temp_bytes, hipStream_t s) {
  hipError_t err = rocprim::reduce_by_key(
      temp, temp_bytes, keys,
      rocprim::counting_iterator<int64_t>(0), (size_t)n, uniq_out,
      starts_out, count_out, rocprim::minimum<int64_t>(),
      rocprim::equal_to<int64_t>(), s);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::runs_sorted: ") +
                             hipGetErrorString(err));
}
