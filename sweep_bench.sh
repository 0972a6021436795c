for P in 8 4 2; do
  for S in 524288 131072; do
    echo "== parallelism=$P sample=$S"
    BIGSLICE_PARALLELISM=$P BIGSLICE_GB_SAMPLE_ROWS=$S timeout 200 python bench.py --gpus 1 --steps 15 --warmup 4 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('  ms_per_step:', round(d['ms_per_step'],2), ' Grows/s:', round(d['value']/1e9,2))"
  done
done
