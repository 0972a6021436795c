set -e
timeout 200 python -m pytest tests/test_gpu_kernels.py::test_hand_radix_sort_edges -q 2>&1 | tail -2
for v in 0 2; do
  echo "== variant $v =="
  BIGSLICE_RADIX_VARIANT=$v timeout 300 python benchmarks/sort_ab.py --impl hand --quick 2>&1 | grep -E "n=|correctness"
done
