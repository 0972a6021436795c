cd /tmp && export TMPDIR=/tmp
cat > /tmp/one_sort.py <<'PY'
import sys, torch
sys.path.insert(0, "/root/repo")
from bigslice_amd import kernels
_C = kernels._C
g = torch.Generator(device="cuda").manual_seed(3)
keys = torch.randint(0, 1 << 62, (500_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
for _ in range(2):
    _C.radix_sort_kv(keys, vals)
torch.cuda.synchronize()
print("done")
PY
BIGSLICE_RADIX_VARIANT=2 timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/st2 -o st --output-format csv -- python /tmp/one_sort.py 2>&1 | tail -2
find /root/repo/gpurun_out/st2 -name "*stats*" | head -3
