set -x
free -g | head -2
df -h /tmp | tail -1
df -h . | tail -1
timeout 240 python -m pytest tests/test_gpu_kernels.py::test_store_high_water_tiering tests/test_gpu_kernels.py::test_external_sort_spill_path_gpu tests/test_sortio.py -m gpu -q 2>&1 | tail -2
BIGSLICE_HOST_BUDGET_BYTES=8000000000 timeout 500 python benchmarks/configs.py --config 5 --rows 2000000000 --steps 1 --warmup 0 2>&1 | tail -2
