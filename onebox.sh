cd /tmp && export PYTHONPATH=/root/repo
R() { timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 3 2>/dev/null | python3 -c "import json,sys; print('  microprof ->', round(json.load(sys.stdin)['ms'],2), 'ms')"; }
L() { python3 <<'PY'
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from bigslice_amd import kernels
g = torch.Generator(device="cuda"); g.manual_seed(1)
keys = torch.randint(0, 1_000_000, (125_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
table = kernels._C.alloc_packed_table(4*1024*1024, keys)
flags = torch.zeros(2, dtype=torch.int32, device="cuda")
ts = []
for it in range(12):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    kernels._C.groupby_insert_packed(keys, vals, table, flags, 4096)
    torch.cuda.synchronize(); ts.append((time.perf_counter()-t0)*1000)
print("  loop ->", [round(x,1) for x in ts[::3]])
PY
}
echo p1; R
echo p2; R
echo p3-loop; L
echo p4; R
