cd /tmp && export PYTHONPATH=/root/repo
echo p1-heavy; timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 2 2>/dev/null | python3 -c "import json,sys; print('  ->', round(json.load(sys.stdin)['ms'],2))"
for it in 1 2 6 12; do
  timeout 300 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters $it 2>/dev/null | python3 -c "import json,sys; print('  iters=$it mean:', round(json.load(sys.stdin)['ms'],2), 'ms')"
done
