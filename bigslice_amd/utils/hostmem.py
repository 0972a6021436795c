"""Process-wide host-DRAM accountant for the spill/store tiers.

Round-1 gap (NOTES item 5 / VERDICT task 3): the Spiller's host budget
and the MemoryStore's host tier were accounted independently, so a
beyond-HBM job (10B-row single-GPU sort: 160 GB of sorted runs + 160 GB
of stored output) could exceed the box's DRAM even though each tier was
individually under its cap.  The reference's spiller instead tiers
unboundedly to disk (sliceio/spiller.go:27-127).

One global accountant now gates every host-tier placement: ``reserve``
either charges the shared budget or answers False, in which case the
caller writes the batch to disk (the unbounded tier).  The budget
defaults to a fraction of the machine's physical DRAM, never more than
BIGSLICE_HOST_BUDGET_BYTES.
"""

from __future__ import annotations

import os
import threading

_lock = threading.Lock()
_used = 0


def _phys_bytes() -> int:
    try:
        return os.sysconf("SC_PHYS_PAGES") * os.sysconf("SC_PAGE_SIZE")
    except (ValueError, OSError):
        return 256 << 30


def budget_bytes() -> int:
    env = os.environ.get("BIGSLICE_HOST_BUDGET_BYTES")
    if env:
        try:
            return int(env)
        except ValueError:
            pass
    frac = float(os.environ.get("BIGSLICE_HOST_FRACTION", "0.5"))
    return int(_phys_bytes() * frac)


def reserve(nbytes: int) -> bool:
    """Charge nbytes against the shared host budget; False = the caller
    must use the disk tier instead."""
    global _used
    with _lock:
        if _used + nbytes > budget_bytes():
            return False
        _used += nbytes
        return True


def release(nbytes: int) -> None:
    global _used
    with _lock:
        _used = max(0, _used - nbytes)


def used_bytes() -> int:
    with _lock:
        return _used
