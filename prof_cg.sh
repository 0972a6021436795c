cd /tmp && export TMPDIR=/tmp
timeout 400 python /root/repo/benchmarks/configs.py --config 4 --rows 1000000000 --steps 3 --warmup 1 2>&1 | tail -1
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/cgstats -o c --output-format csv -- python /root/repo/benchmarks/configs.py --config 4 --rows 1000000000 --steps 2 --warmup 1 2>&1 | tail -1
