cd /tmp && export PYTHONPATH=/root/repo
R() { timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 3 2>/dev/null | python3 -c "import json,sys; print('  1M ->', round(json.load(sys.stdin)['ms'],2), 'ms')"; }
echo run1; R
rocm-smi --showgpuclocks | grep -i "sclk clock level" || true
echo run2; R
rocm-smi --showgpuclocks | grep -i "sclk clock level" || true
echo set-perflevel-high:
rocm-smi --setperflevel high 2>&1 | tail -2
echo run3; R
