cd /tmp && export PYTHONPATH=/root/repo
echo p1-heavy; timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 2 2>/dev/null | python3 -c "import json,sys; print('  ->', round(json.load(sys.stdin)['ms'],2))"
echo p2-unpacked-loop
python3 <<'PY'
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from bigslice_amd import kernels
_C = kernels._C
g = torch.Generator(device="cuda"); g.manual_seed(1)
keys = torch.randint(0, 1_000_000, (125_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
cap = 4*1024*1024
tkeys = torch.full((cap+1,), -(1<<63), dtype=torch.int64, device="cuda")
tabs = [torch.zeros(cap+1, dtype=torch.int64, device="cuda")]
flags = torch.zeros(2, dtype=torch.int32, device="cuda")
for it in range(6):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    _C.groupby_insert(keys, [vals], [0], tkeys, tabs, flags, 128)
    torch.cuda.synchronize()
    print(f"  unpacked {(time.perf_counter()-t0)*1000:6.2f} ms", flush=True)
PY
echo p3-packed-microprof
BIGSLICE_GB_PACKED=1 timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 2 2>/dev/null | python3 -c "import json,sys; print('  ->', round(json.load(sys.stdin)['ms'],2))"
echo p4-unpacked-microprof
timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 2 2>/dev/null | python3 -c "import json,sys; print('  ->', round(json.load(sys.stdin)['ms'],2))"
