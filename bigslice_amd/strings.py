"""Device string support: packed varlen byte columns + dictionary ids.

The reference hashes strings on the host per row
(frame/ops_builtin.go:143-150); here a whole batch of strings packs
into ONE (bytes, offsets) pair and a HIP kernel (K17,
csrc/strings.hip) murmur3-hashes every row in parallel, bit-identical
to hashing.murmur3_bytes.  64-bit two-seed ids serve as dictionary
codes so string group-bys run device-resident (see
recipes.gpu_wordcount)."""

from __future__ import annotations

from typing import Sequence, Tuple

import numpy as np
import torch

from . import hashing


def pack_strings(strs: Sequence[str]) -> Tuple[np.ndarray, np.ndarray]:
    """(bytes u8, offsets i64) arrays for a batch of strings (utf-8)."""
    enc = [s.encode("utf-8") for s in strs]
    offsets = np.zeros(len(enc) + 1, dtype=np.int64)
    np.cumsum(np.fromiter((len(b) for b in enc), np.int64, len(enc)),
              out=offsets[1:])
    # copy: frombuffer views are read-only, which torch rejects
    data = np.frombuffer(b"".join(enc), dtype=np.uint8).copy()
    return data, offsets


def to_device(data: np.ndarray, offsets: np.ndarray, device):
    b = torch.from_numpy(np.ascontiguousarray(data)).to(device,
                                                        non_blocking=True)
    o = torch.from_numpy(offsets).to(device, non_blocking=True)
    return b, o


def hash_strings(strs: Sequence[str], device, seed: int = 0) -> torch.Tensor:
    """murmur3-32 of every string, computed on device (uint32->int64)."""
    from . import kernels
    if not str(device).startswith("cuda") or kernels._C is None:
        return torch.tensor([hashing.murmur3_bytes(s.encode("utf-8"), seed)
                             for s in strs], dtype=torch.int64)
    b, o = to_device(*pack_strings(strs), device)
    return kernels._C.hash_bytes(b, o, seed).to(torch.int64)


_ID_SEED_HI = 0x9ACB0442
_ID_SEED_LO = 0x85EBCA6B


def string_ids(strs: Sequence[str], device) -> torch.Tensor:
    """64-bit dictionary ids (two-seed murmur3) on device; host
    fallback is bit-identical."""
    from . import kernels
    if str(device).startswith("cuda") and kernels._C is not None:
        b, o = to_device(*pack_strings(strs), device)
        return kernels._C.hash_bytes64(b, o, _ID_SEED_HI, _ID_SEED_LO)
    mm = hashing.murmur3_bytes
    vals = []
    for s in strs:
        e = s.encode("utf-8")
        v = (mm(e, _ID_SEED_HI) << 32) | mm(e, _ID_SEED_LO)
        vals.append(v - (1 << 64) if v >= (1 << 63) else v)
    return torch.tensor(vals, dtype=torch.int64)
