"""Global configuration knobs for the engine.

The reference makes its global chunk size a flag
(internal/defaultsize/size.go:14-19, default 128 rows).  On MI355X the unit of
work is a device batch: a column-parallel chunk sized to keep all 256 CUs busy
and to amortize kernel-launch overhead against 288 GB of HBM3E.  We therefore
default to multi-million-row device batches and small (64k) host batches.
"""

import os


def _env_int(name: str, default: int) -> int:
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


# Rows per device batch flowing through a task pipeline on GPU.
DEVICE_CHUNK_ROWS = _env_int("BIGSLICE_DEVICE_CHUNK_ROWS", 8 << 20)

# Rows per host (CPU) batch; CPU paths are for plumbing/tests, keep it modest.
HOST_CHUNK_ROWS = _env_int("BIGSLICE_HOST_CHUNK_ROWS", 1 << 16)

# Seed used by the combiner hash table so a previous partitioning step does
# not strip hash entropy (reference: exec/combiner.go:43).
COMBINER_HASH_SEED = 0x9ACB0442

# In-memory target number of keys for a spilling combiner before it spills
# (reference: exec/bigmachine.go:1105 uses chunk*100; ours is device-scaled).
COMBINER_TARGET_KEYS = _env_int("BIGSLICE_COMBINER_TARGET_KEYS", 1 << 26)

# Spill target bytes for external sort runs (reference cogroup.go:126-127
# uses 32 MiB; device-scaled default 4 GiB keeps runs HBM-resident).
SORT_SPILL_TARGET_BYTES = _env_int("BIGSLICE_SORT_SPILL_BYTES", 4 << 30)
# Sorted runs spill in slices of this size so a k-way merge's per-run
# readahead windows stay small (80 runs x 2 GiB windows would not fit
# HBM beside a large resident input).
SORT_SPILL_CHUNK_BYTES = _env_int("BIGSLICE_SORT_SPILL_CHUNK_BYTES",
                                  256 << 20)
# Spill backpressure: cap un-drained device bytes held by in-flight
# D2H copies (the sort produces runs faster than the host link drains
# them; unbounded backlog OOMs large jobs).
SPILL_BACKPRESSURE_BYTES = _env_int("BIGSLICE_SPILL_BACKPRESSURE_BYTES",
                                    32 << 30)

# Initial hash-aggregate table capacity (slots); the table grows x2 on
# probe-chain overflow (reference combiner grow policy, exec/combiner.go:47).
GROUPBY_INITIAL_CAP = _env_int("BIGSLICE_GROUPBY_INITIAL_CAP", 1 << 22)

# Machine-combiners mode: producer tasks of one shuffle phase on a GPU
# share one combiner table (reference exec/session.go:166-176).  Shared
# tables keep the working set LLC-resident; disabled automatically when
# fault injection is active (the mode has no loss recovery).
MACHINE_COMBINERS = os.environ.get("BIGSLICE_MACHINE_COMBINERS",
                                   "1") == "1"

# Maximum consecutive losses of a single task before giving up
# (reference: exec/eval.go:30).
MAX_CONSECUTIVE_LOST = 5

# Default parallelism for the local executor (concurrent shard tasks).
DEFAULT_PARALLELISM = _env_int("BIGSLICE_PARALLELISM", 8)


def chunk_rows(device: str) -> int:
    """Default batch size for a given device kind ('cuda' or 'cpu')."""
    return DEVICE_CHUNK_ROWS if device.startswith("cuda") else HOST_CHUNK_ROWS
