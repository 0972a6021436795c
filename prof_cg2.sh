cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/cg2 -o c --output-format csv -- python /root/repo/benchmarks/configs.py --config 4 --rows 500000000 --steps 2 --warmup 1 >/dev/null 2>&1
echo done
