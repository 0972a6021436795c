cd /tmp && export PYTHONPATH=/root/repo
for nk in 1000 100000 1000000 10000000 100000000; do
  timeout 300 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys $nk --iters 5 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print(d['nkeys'], 'keys:', round(d['ms'],2), 'ms,', round(d['grows_per_sec'],1), 'G/s')"
done
