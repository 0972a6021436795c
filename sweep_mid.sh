cd /tmp && export PYTHONPATH=/root/repo
for nk in 100000 1000000; do
  timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys $nk --iters 3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print(d['nkeys'], 'keys:', round(d['ms'],2), 'ms')"
done
