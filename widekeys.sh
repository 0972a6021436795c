cd /tmp && export TMPDIR=/tmp && export PYTHONPATH=/root/repo
for nk in 1000 100000 1000000 10000000 100000000; do
  timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys $nk --iters 3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print(f'{d[\"nkeys\"]:>11,} keys: {d[\"ms\"]:7.2f} ms  {d[\"grows_per_sec\"]:6.2f} G/s')"
done
# PMC on the partition scatter (config-4 hot kernel): SQ picture
cat > /tmp/part.py <<'PY'
import sys; sys.path.insert(0, "/root/repo")
import torch
from bigslice_amd import kernels
from bigslice_amd.frame import Frame
g = torch.Generator(device="cuda"); g.manual_seed(1)
keys = torch.randint(0, 1 << 20, (500_000_000,), dtype=torch.int64, device="cuda", generator=g)
vals = torch.ones_like(keys)
f = Frame([keys, vals], 1)
for _ in range(3):
    kernels.partition_frame(f, 8, None)
torch.cuda.synchronize(); print("ok")
PY
timeout 240 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /root/repo/gpurun_out/part_sq -o p --output-format csv -- python /tmp/part.py 2>&1 | tail -1
timeout 240 rocprofv3 --pmc TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum -d /root/repo/gpurun_out/part_tcc -o t --output-format csv -- python /tmp/part.py 2>&1 | tail -1
