cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/bstats -o b --output-format csv -- python /root/repo/bench.py --gpus 1 --steps 5 --warmup 2 2>&1 | tail -2
