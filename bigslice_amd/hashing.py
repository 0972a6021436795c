"""Murmur3-32 hashing, bit-identical to the reference.

The reference hashes key columns with seeded murmur3-32 over little-endian
bytes (frame/ops_builtin.go:140-164: scalars of width<=4 via 4 LE bytes,
8-byte scalars via 8 LE bytes; strings over their bytes; bool -> seed+{0,1};
frame/frame.go:395-401 XOR-combines across prefix columns).  Keeping the hash
bit-identical keeps partition assignment reproducible against the reference.

Host path: numpy-vectorized murmur3 (uint32 wraparound arithmetic).
Device path: the same algorithm as a HIP kernel (csrc/hash_partition.hip),
dispatched through bigslice_amd.kernels.
"""

from __future__ import annotations

from typing import Sequence

import numpy as np
import torch


_C1 = np.uint32(0xCC9E2D51)
_C2 = np.uint32(0x1B873593)


def _rotl32(x: np.ndarray, r: int) -> np.ndarray:
    return (x << np.uint32(r)) | (x >> np.uint32(32 - r))


def _mix_block(h: np.ndarray, k: np.ndarray) -> np.ndarray:
    k = k * _C1
    k = _rotl32(k, 15)
    k = k * _C2
    h = h ^ k
    h = _rotl32(h, 13)
    return h * np.uint32(5) + np.uint32(0xE6546B64)


def _fmix(h: np.ndarray, total_len: int) -> np.ndarray:
    h = h ^ np.uint32(total_len)
    h = h ^ (h >> np.uint32(16))
    h = h * np.uint32(0x85EBCA6B)
    h = h ^ (h >> np.uint32(13))
    h = h * np.uint32(0xC2B2AE35)
    h = h ^ (h >> np.uint32(16))
    return h


def murmur3_u32(vals: np.ndarray, seed: int) -> np.ndarray:
    """murmur3_x86_32 of each 4-byte LE value (reference hash32)."""
    h = np.full(vals.shape, np.uint32(seed), dtype=np.uint32)
    h = _mix_block(h, vals.astype(np.uint32, copy=False))
    return _fmix(h, 4)


def murmur3_u64(vals: np.ndarray, seed: int) -> np.ndarray:
    """murmur3_x86_32 of each 8-byte LE value (reference hash64)."""
    v = vals.astype(np.uint64, copy=False)
    lo = (v & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    hi = (v >> np.uint64(32)).astype(np.uint32)
    h = np.full(v.shape, np.uint32(seed), dtype=np.uint32)
    h = _mix_block(h, lo)
    h = _mix_block(h, hi)
    return _fmix(h, 8)


_M32 = 0xFFFFFFFF


def murmur3_bytes(data: bytes, seed: int) -> int:
    """Scalar murmur3_x86_32 over arbitrary bytes (strings).  Pure-int
    arithmetic: ~50x faster per call than numpy scalar ops (this runs
    per word in host string pipelines)."""
    h = seed & _M32
    n = len(data)
    nblocks4 = n & ~3
    for off in range(0, nblocks4, 4):
        k = int.from_bytes(data[off:off + 4], "little")
        k = (k * 0xCC9E2D51) & _M32
        k = ((k << 15) | (k >> 17)) & _M32
        k = (k * 0x1B873593) & _M32
        h ^= k
        h = ((h << 13) | (h >> 19)) & _M32
        h = (h * 5 + 0xE6546B64) & _M32
    tail = data[nblocks4:]
    if tail:
        k1 = 0
        if len(tail) >= 3:
            k1 ^= tail[2] << 16
        if len(tail) >= 2:
            k1 ^= tail[1] << 8
        k1 ^= tail[0]
        k1 = (k1 * 0xCC9E2D51) & _M32
        k1 = ((k1 << 15) | (k1 >> 17)) & _M32
        k1 = (k1 * 0x1B873593) & _M32
        h ^= k1
    h ^= n
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & _M32
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & _M32
    h ^= h >> 16
    return h


# User-defined per-type ops (reference frame.RegisterOps,
# frame/ops.go:31-96): custom hash/order for object column values.
_custom_ops = {}


def register_ops(py_type, hash_fn=None, less_key=None):
    """Register custom ops for a Python value type appearing in object
    columns: hash_fn(value, seed) -> uint32 int; less_key(value) -> a
    sortable key.  Mirrors frame.RegisterOps (the reference's only
    in-repo use is a key type hashed by string length,
    reshuffle_test.go:86-94)."""
    _custom_ops[py_type] = (hash_fn, less_key)


def custom_less_key(value):
    ops = _custom_ops.get(type(value))
    if ops is not None and ops[1] is not None:
        return ops[1](value)
    return value


def _mm3_u32_int(x: int, seed: int) -> int:
    """Pure-int reference hash32 (4 LE bytes)."""
    return murmur3_bytes((x & _M32).to_bytes(4, "little"), seed)


def _mm3_u64_int(x: int, seed: int) -> int:
    """Pure-int reference hash64 (8 LE bytes)."""
    return murmur3_bytes((x & 0xFFFFFFFFFFFFFFFF).to_bytes(8, "little"),
                         seed)


def _hash_host_column(col, seed: int) -> np.ndarray:
    """Hash one host column -> uint32 numpy array."""
    with np.errstate(over="ignore"):
        if isinstance(col, torch.Tensor):
            dt = col.dtype
            if dt == torch.bool:
                return np.uint32(seed) + col.numpy().astype(np.uint32)
            a = col.numpy()
            if dt in (torch.int64, torch.uint64, torch.float64):
                if dt == torch.float64:
                    a = a.view(np.uint64)
                return murmur3_u64(a.astype(np.int64).view(np.uint64)
                                   if a.dtype != np.uint64 else a, seed)
            if dt == torch.float32:
                return murmur3_u32(a.view(np.uint32), seed)
            # narrower ints: Go converts via uint32(v), sign-extending
            # signed types (two's complement reinterpretation).
            return murmur3_u32(a.astype(np.int64).astype(np.uint32)
                               if a.dtype.kind == "i"
                               else a.astype(np.uint32), seed)
        # object column (strings / arbitrary python values)
        out = np.empty(len(col), dtype=np.uint32)
        for i, v in enumerate(col):
            ops = _custom_ops.get(type(v))
            if ops is not None and ops[0] is not None:
                out[i] = np.uint32(ops[0](v, seed) & 0xFFFFFFFF)
            elif isinstance(v, str):
                out[i] = murmur3_bytes(v.encode("utf-8"), seed)
            elif isinstance(v, bytes):
                out[i] = murmur3_bytes(v, seed)
            elif isinstance(v, bool):
                out[i] = (seed + int(v)) & _M32
            elif isinstance(v, int):
                out[i] = _mm3_u64_int(v, seed)
            elif isinstance(v, float):
                import struct as _struct
                bits = _struct.unpack("<Q", _struct.pack("<d", v))[0]
                out[i] = _mm3_u64_int(bits, seed)
            elif isinstance(v, tuple):
                # hash tuples by hashing the concatenated member hashes
                h = np.uint32(seed)
                for m in v:
                    hv = _hash_host_column([m], seed)[0]
                    h = _mix_block(h, np.uint32(hv))
                out[i] = _fmix(h, 4 * len(v))
            else:
                raise TypeError(f"unhashable object column value {type(v)}")
        return out


def hash_columns(cols: Sequence, seed: int = 0) -> torch.Tensor:
    """XOR-combined 32-bit hash of key columns (reference
    frame.HashWithSeed).  Device tensors dispatch to the HIP kernel."""
    if not cols:
        raise ValueError("hash of zero key columns")
    from .frame import BytesColumn
    tensors = [c for c in cols if not isinstance(c, BytesColumn)]
    bytes_cols = [c for c in cols if isinstance(c, BytesColumn)]
    first = cols[0]
    on_device = (isinstance(first, torch.Tensor) and first.is_cuda) or (
        isinstance(first, BytesColumn) and first.data.is_cuda)
    if on_device:
        from . import kernels
        h = None
        if tensors:
            h = kernels.hash_columns_device(tensors, seed)
        for b in bytes_cols:
            hb = b.hash32(seed)
            h = hb if h is None else (h ^ hb)
        return h
    h = None
    for c in cols:
        if isinstance(c, BytesColumn):
            hc = c.hash32(seed).numpy().astype(np.uint32)
        else:
            hc = _hash_host_column(c, seed)
        h = hc if h is None else (h ^ hc)
    return torch.from_numpy(h.astype(np.int64))  # non-negative values
