"""Device kernel dispatch.

Loads the in-tree HIP extension (bigslice_amd._C, built by setup.py for
gfx950).  Policy: on a GPU host the extension MUST be present — device ops
raise rather than silently falling back to eager torch, so a missing build
is loud.  On CPU-only hosts everything falls back to torch/numpy reference
implementations (used by tests as numerics oracles).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch

_C = None
_load_error: Optional[BaseException] = None

try:
    from bigslice_amd import _C as _C  # built by setup.py build_ext --inplace
except Exception as e:  # pragma: no cover
    _load_error = e


ALLOW_FALLBACK = os.environ.get("BIGSLICE_ALLOW_EAGER_FALLBACK", "0") == "1"


def have_extension() -> bool:
    return _C is not None


def _require(feature: str):
    if _C is None:
        raise RuntimeError(
            f"bigslice_amd._C extension not available for {feature} on a "
            f"GPU tensor (build with `python setup.py build_ext --inplace`); "
            f"import error: {_load_error!r}")


# -- hashing --------------------------------------------------------------

def hash_columns_device(cols: List[torch.Tensor], seed: int) -> torch.Tensor:
    """murmur3-32 XOR-combined over key columns; returns uint32 tensor."""
    if _C is None and ALLOW_FALLBACK:
        from .. import hashing
        host = [c.cpu() for c in cols]
        return hashing.hash_columns(host, seed).to(cols[0].device)
    _require("hash_columns")
    return _C.hash_columns(cols, seed)


# -- partition (K4) -------------------------------------------------------

def partition_supported(frame) -> bool:
    if _C is None:
        return False
    return all(isinstance(c, torch.Tensor) for c in frame.columns)


def partition_frame(frame, num_partitions: int, partitioner):
    """Fused hash+histogram+scatter: returns per-partition sub-frames
    (views over one reordered buffer)."""
    from ..frame import Frame
    if partitioner is not None:
        p = partitioner(frame, num_partitions).to(torch.int32)
        reordered_cols, counts = _C.scatter_by_partition(
            list(frame.columns), p, num_partitions)
    else:
        key_cols = list(frame.columns[: frame.prefix])
        reordered_cols, counts = _C.hash_partition(
            list(frame.columns), key_cols, num_partitions, 0)
    sorted_f = Frame(reordered_cols, frame.prefix)
    out = []
    off = 0
    for c in counts.tolist():
        out.append(sorted_f.slice(off, off + c) if c else None)
        off += c
    return out


# -- group-by aggregate (K9) ----------------------------------------------

_AGG_CODES = {"sum": 0, "min": 1, "max": 2, "prod": 3}

_GB_KEY_DTYPES = (torch.int64, torch.int32, torch.uint32, torch.uint64)
_GB_VAL_DTYPES = (torch.int64, torch.int32, torch.float32, torch.float64)


def groupby_supported(keys: torch.Tensor, vals: List[torch.Tensor],
                      aggs: List[str]) -> bool:
    if _C is None:
        return False
    if keys.dtype not in _GB_KEY_DTYPES:
        return False
    if not all(v.dtype in _GB_VAL_DTYPES for v in vals):
        return False
    return all(a in _AGG_CODES for a in aggs)


def groupby(keys: torch.Tensor, vals: List[torch.Tensor],
            aggs: List[str]):
    """Hash-aggregate: returns (unique_keys, [combined values])."""
    codes = [_AGG_CODES[a] for a in aggs]
    out = _C.groupby(keys, list(vals), codes)
    return out[0], list(out[1:])


# -- sort (K6) -------------------------------------------------------------

def sort_pairs_supported(keys: torch.Tensor) -> bool:
    return _C is not None and keys.dtype in (torch.int64, torch.int32,
                                             torch.float32, torch.float64)


def radix_argsort(keys: torch.Tensor) -> torch.Tensor:
    """Device radix sort; returns the sorting permutation (int64)."""
    if _C is None:
        return torch.argsort(keys, stable=True)
    return _C.radix_argsort(keys)
