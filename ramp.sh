cd /tmp && export PYTHONPATH=/root/repo
timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 3 >/dev/null 2>&1; echo heavy proc 1 done
python3 <<'PY'
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from bigslice_amd import kernels
g = torch.Generator(device="cuda"); g.manual_seed(1)
keys = torch.randint(0, 1_000_000, (125_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
cap = 4 * 1024 * 1024
table = kernels._C.alloc_packed_table(cap, keys)
flags = torch.zeros(2, dtype=torch.int32, device="cuda")
t_start = time.perf_counter()
for it in range(120):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    kernels._C.groupby_insert_packed(keys, vals, table, flags, 4096)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) * 1000
    if it % 12 == 0:
        print(f"t={time.perf_counter()-t_start:5.1f}s insert {dt:6.2f} ms", flush=True)
PY
