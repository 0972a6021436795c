for P in 1 2 3; do
  echo "== parallelism=$P"
  BIGSLICE_PARALLELISM=$P timeout 200 python bench.py --gpus 1 --steps 20 --warmup 5 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('  ms_per_step:', round(d['ms_per_step'],2), ' Grows/s:', round(d['value']/1e9,2))"
done
echo "== 1B rows, parallelism best-of"
for P in 2 8; do
  BIGSLICE_PARALLELISM=$P timeout 300 python bench.py --gpus 1 --steps 5 --warmup 2 --rows-per-gpu 1000000000 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('  P=$P ms_per_step:', round(d['ms_per_step'],2), ' Grows/s:', round(d['value']/1e9,2))"
done
