"""Per-machine health introspection (reference bigmachine
MemInfo/DiskInfo/LoadInfo polling, exec/slicemachine.go:148-215):
host RSS/free memory, HBM allocation, load average."""

from __future__ import annotations

import os
import resource
from typing import Dict

import torch


def machine_stats() -> Dict[str, float]:
    out: Dict[str, float] = {}
    ru = resource.getrusage(resource.RUSAGE_SELF)
    out["rss_gb"] = ru.ru_maxrss / 1e6
    try:
        load1, load5, load15 = os.getloadavg()
        out["load1"] = load1
    except OSError:
        pass
    try:
        with open("/proc/meminfo") as fp:
            for line in fp:
                if line.startswith("MemAvailable:"):
                    out["mem_available_gb"] = \
                        int(line.split()[1]) / 1e6
                    break
    except OSError:
        pass
    if torch.cuda.is_available():
        out["hbm_allocated_gb"] = torch.cuda.memory_allocated() / 1e9
        out["hbm_reserved_gb"] = torch.cuda.memory_reserved() / 1e9
        free, total = torch.cuda.mem_get_info()
        out["hbm_free_gb"] = free / 1e9
        out["hbm_total_gb"] = total / 1e9
    return out
