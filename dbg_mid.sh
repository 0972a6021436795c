cd /tmp && export PYTHONPATH=/root/repo BIGSLICE_GB_DEBUG=1
timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 3 2>&1 | grep -E "\[gb\]|ms" | head -8
