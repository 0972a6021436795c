set -e
J() { timeout 500 python benchmarks/configs.py "$@" 2>/dev/null | tail -1; }
J --config 1 --steps 5 --warmup 2            > gpurun_out/p_c1.json
J --config 2 --steps 8 --warmup 3            > gpurun_out/p_c2.json
J --config 4 --rows 500000000 --steps 4 --warmup 1 > gpurun_out/p_c4.json
J --config 5 --rows 500000000 --steps 4 --warmup 1 > gpurun_out/p_c5_inHBM.json
J --config 5 --rows 2000000000 --steps 2 --warmup 1 > gpurun_out/p_c5_2B.json
timeout 200 python bench.py --gpus 1 --steps 15 --warmup 4 2>/dev/null | tail -1 > gpurun_out/p_c3_125M.json
timeout 300 python bench.py --gpus 1 --steps 5 --warmup 2 --rows-per-gpu 1000000000 2>/dev/null | tail -1 > gpurun_out/p_c3_1B.json
timeout 240 python bench.py --gpus 1 --steps 8 --warmup 2 --trace gpurun_out/flagship_trace.json 2>/dev/null | tail -1 > gpurun_out/p_c3_traced.json
grep -h ms_per_step gpurun_out/p_*.json | python3 -c "
import sys, json
for line in sys.stdin:
    d = json.loads(line)
    print(d['config']['model'][:44], d['config'].get('rows_total'), round(d['ms_per_step'],2), 'ms')"
