set -x
grep -E "MemTotal|MemAvailable" /proc/meminfo
avail_kb=$(grep MemAvailable /proc/meminfo | awk '{print $2}')
budget=$((avail_kb*1024*6/10))
disk_free=$(df --output=avail -B1 /tmp | tail -1)
# demand = rows * 16B * 2 tiers; keep 10B only if it fits budget + 80% of disk
rows=$(python3 -c "
b=$budget; d=int($disk_free*0.8)
cap=int((b+d)*0.95//32)
print(min(10_000_000_000, cap))")
echo "budget=$budget disk_free=$disk_free rows=$rows"
BIGSLICE_HOST_BUDGET_BYTES=$budget timeout 1100 python benchmarks/configs.py --config 5 --rows $rows --steps 1 --warmup 0 2>&1 | tail -2
