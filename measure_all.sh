M() { timeout 400 python benchmarks/configs.py "$@" 2>/dev/null | tail -1; }
echo "== config1 wordcount CPU"; M --config 1 --steps 5 --warmup 2
echo "== config2 100M"; M --config 2 --steps 8 --warmup 3
echo "== config3 via bench 125M"; timeout 200 python bench.py --gpus 1 --steps 15 --warmup 4 2>/dev/null | tail -1
echo "== config3 1B"; timeout 300 python bench.py --gpus 1 --steps 5 --warmup 2 --rows-per-gpu 1000000000 2>/dev/null | tail -1
echo "== config4 2x500M"; M --config 4 --rows 500000000 --steps 3 --warmup 1
echo "== config5 500M inHBM"; M --config 5 --rows 500000000 --steps 3 --warmup 1
