python3 - <<'PY'
import torch, time
x = torch.empty(4 << 30 >> 3, dtype=torch.int64, device="cuda")  # 4 GiB
h = torch.empty_like(x, device="cpu", pin_memory=True)
torch.cuda.synchronize()
for name, fn in (("D2H", lambda: h.copy_(x, non_blocking=True)),
                 ("H2D", lambda: x.copy_(h, non_blocking=True))):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    print(f"{name}: {4 / dt:.1f} GiB/s")
PY
timeout 500 python benchmarks/configs.py --config 5 --rows 2000000000 --steps 2 --warmup 1 2>/dev/null | python3 -c "import json,sys; print('2B spill:', round(json.load(sys.stdin)['ms_per_step']/1000,2), 's')"
