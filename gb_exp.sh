set -x
cd /tmp && export TMPDIR=/tmp && export PYTHONPATH=/root/repo
R=/root/repo
# A/B: insert vs insert_rep at several cardinalities and rep counts
for nk in 30000 100000 300000 1000000 10000000; do
  timeout 120 python -m bigslice_amd.tools.microprof insert --rows 125000000 --nkeys $nk --iters 5
  for rep in 2 8; do
    timeout 120 python -m bigslice_amd.tools.microprof insert_rep --rows 125000000 --nkeys $nk --nrep $rep --iters 5
  done
done 2>&1 | grep -v Warn
# PMC: SQ picture of the steady-state insert
