cd /tmp && export PYTHONPATH=/root/repo
R() { timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys $1 --iters 3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print(d['nkeys'],'->',round(d['ms'],2),'ms')"; }
echo run1; R 1000000
echo run2; R 1000000
echo run3-100k; R 100000
echo run4; R 1000000
