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
                                  int64_t n, int end_bit, void* temp,
                                  size_t& temp_bytes, hipStream_t stream) {
  hipError_t err = rocprim::radix_sort_pairs(
      temp, temp_bytes, keys_in, keys_out, vals_in, vals_out, (size_t)n, 0,
      end_bit, stream);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::radix_sort_pairs: ") +
                             hipGetErrorString(err));
}

template <typename K>
static void radix_sort_keys_impl(const K* keys_in, K* keys_out, int64_t n,
                                 int end_bit, void* temp,
                                 size_t& temp_bytes, hipStream_t stream) {
  hipError_t err = rocprim::radix_sort_keys(
      temp, temp_bytes, keys_in, keys_out, (size_t)n, 0, end_bit,
      stream);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::radix_sort_keys: ") +
                             hipGetErrorString(err));
}

// Explicit type dispatch used by ext.hip.
#define INSTANTIATE_SORT(K)                                               \
  void radix_sort_pairs_##K(const void* ki, void* ko, const int64_t* vi,  \
                            int64_t* vo, int64_t n, int end_bit,          \
                            void* temp, size_t& temp_bytes,               \
                            hipStream_t s) {                              \
    radix_sort_pairs_impl<K>((const K*)ki, (K*)ko, vi, vo, n, end_bit,    \
                             temp, temp_bytes, s);                        \
  }

INSTANTIATE_SORT(int64_t)
INSTANTIATE_SORT(int32_t)
INSTANTIATE_SORT(float)
INSTANTIATE_SORT(double)

#define INSTANTIATE_SORT_KEYS(K)                                          \
  void radix_sort_keys_##K(const void* ki, void* ko, int64_t n,           \
                           int end_bit, void* temp, size_t& temp_bytes,   \
                           hipStream_t s) {                               \
    radix_sort_keys_impl<K>((const K*)ki, (K*)ko, n, end_bit, temp,       \
                            temp_bytes, s);                               \
  }

INSTANTIATE_SORT_KEYS(int64_t)
INSTANTIATE_SORT_KEYS(int32_t)
INSTANTIATE_SORT_KEYS(float)
INSTANTIATE_SORT_KEYS(double)

// K16: segment reduce over key-sorted (or run-grouped) pairs via
// rocPRIM's tuned reduce-by-key (decoupled-lookback single pass).
// Replaces the mask/nonzero/cumsum torch chain in the sort-combine.
#include <rocprim/device/device_reduce_by_key.hpp>

void reduce_by_key_sum_i64(const int64_t* keys, const int64_t* vals,
                           int64_t n, int64_t* uniq_out, int64_t* sums_out,
                           int64_t* count_out, void* temp,
                           size_t& temp_bytes, hipStream_t s) {
  hipError_t err = rocprim::reduce_by_key(
      temp, temp_bytes, keys, vals, (size_t)n, uniq_out, sums_out,
      count_out, rocprim::plus<int64_t>(), rocprim::equal_to<int64_t>(),
      s);
  if (err != hipSuccess)
    throw std::runtime_error(std::string("rocprim::reduce_by_key: ") +
                             hipGetErrorString(err));
}
