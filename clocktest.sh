cd /tmp && export PYTHONPATH=/root/repo
R() { timeout 200 python -m bigslice_amd.tools.microprof groupby --rows 125000000 --nkeys 1000000 --iters 3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('   1M ->',round(json.load(open('/dev/stdin')) if 0 else d['ms'],2),'ms')"; }
echo first; R
echo second-with-clock-probe
rocm-smi --showgpuclocks 2>/dev/null | grep -i sclk || true
R
echo third-with-burn
python3 -c "
import torch, time
a = torch.randn(8192, 8192, device='cuda', dtype=torch.float32)
t0 = time.time()
while time.time() - t0 < 2.0:
    a = a @ a * 1e-6  # keep values bounded
torch.cuda.synchronize()
print('burn done')
"
rocm-smi --showgpuclocks 2>/dev/null | grep -i sclk || true
R
