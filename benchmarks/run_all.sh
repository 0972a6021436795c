#!/bin/bash
# Run every BASELINE measurement config; one JSON line each.
# Usage: bash benchmarks/run_all.sh [outdir]
set -e
cd "$(dirname "$0")/.."
OUT=${1:-bench_results}
mkdir -p "$OUT"
python benchmarks/configs.py --config 1 --steps 3 --warmup 1 \
    2>/dev/null | tee "$OUT/config1_wordcount_cpu.json"
python bench.py --steps 5 --warmup 2 \
    2>/dev/null | tee "$OUT/config3_groupby_125M.json"
python bench.py --steps 3 --warmup 1 --rows-per-gpu 1000000000 \
    2>/dev/null | tee "$OUT/config3_groupby_1B.json"
python benchmarks/configs.py --config 2 --steps 5 --warmup 2 \
    2>/dev/null | tee "$OUT/config2_mapfilter_100M.json"
python benchmarks/configs.py --config 4 --rows 500000000 --steps 2 \
    --warmup 1 2>/dev/null | tee "$OUT/config4_cogroup_2x500M.json"
python benchmarks/configs.py --config 5 --rows 500000000 --steps 2 \
    --warmup 1 2>/dev/null | tee "$OUT/config5_sort_500M_inHBM.json"
python benchmarks/configs.py --config 5 --rows 2000000000 --steps 1 \
    --warmup 1 2>/dev/null | tee "$OUT/config5_sort_2B_spill.json"
