set -x
cd /tmp && export TMPDIR=/tmp
cat > /tmp/one_sort.py <<'PY'
import sys, torch
sys.path.insert(0, "/root/repo")
from bigslice_amd import kernels
_C = kernels._C
g = torch.Generator(device="cuda").manual_seed(3)
keys = torch.randint(0, 1 << 62, (125_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
for _ in range(3):
    _C.radix_sort_kv(keys, vals)
torch.cuda.synchronize()
print("done")
PY
BIGSLICE_RADIX_VARIANT=2 timeout 240 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /root/repo/gpurun_out/pmc_sort -o pmc --output-format csv -- python /tmp/one_sort.py 2>&1 | tail -3
BIGSLICE_RADIX_VARIANT=2 timeout 240 rocprofv3 --stats -d /root/repo/gpurun_out/stats_sort -o st --output-format csv -- python /tmp/one_sort.py 2>&1 | tail -3
find /root/repo/gpurun_out -name "*.csv" | head
for f in $(find /root/repo/gpurun_out/stats_sort -name "*stats*.csv"); do echo "== $f"; cat "$f" | head -15; done
for f in $(find /root/repo/gpurun_out/pmc_sort -name "*.csv"); do echo "== $f"; head -3 "$f"; done
