"""Frame wire codec with per-batch integrity checksums.

Role-parity with the reference's gob codec (sliceio/codec.go:68-235): each
batch is encoded with a small header and a CRC32-IEEE checksum of the payload,
and decode verifies the checksum (corruption detection parity with
sliceio/codec.go:111,226-234).  Unlike gob, the wire layout here IS the device
layout: numeric columns are raw little-endian column buffers that can be
hipMemcpy'd straight into HBM with no per-row decode; object columns are
pickled (object columns are only ever decoded from files this process
wrote: spill, cache and checkpoint stores — the same trust model as
the reference's gob).

Batch layout:
    magic  u32 = 0xB16S11CE (0xB16511CE)
    nrows  u64
    schema_key_len u16, schema_key bytes
    ncols  u16
    per column: kind u8 (0=tensor,1=object), payload_len u64, payload
    crc32  u32  (of everything from nrows through last payload)
"""

from __future__ import annotations

import io
import pickle
import struct
import zlib
from typing import Optional

import numpy as np
import torch

from ..frame import BytesColumn, Frame
from ..schema import Schema

MAGIC = 0xB16511CE


class CorruptionError(IOError):
    pass


def _tensor_bytes(t: torch.Tensor) -> bytes:
    a = t.detach().cpu().contiguous()
    if a.dtype in (torch.bfloat16, torch.float16):
        # numpy has no bf16: ship the raw 16-bit pattern
        return a.view(torch.int16).numpy().tobytes()
    return a.numpy().tobytes()


def _tensor_from_bytes(buf: bytes, dtype: torch.dtype) -> torch.Tensor:
    if dtype == torch.bfloat16 or dtype == torch.float16:
        t = torch.frombuffer(bytearray(buf), dtype=torch.int16)
        return t.view(dtype)
    np_dt = torch.empty(0, dtype=dtype).numpy().dtype
    return torch.from_numpy(np.frombuffer(bytearray(buf), dtype=np_dt))


def encode_frame(frame: Frame, out: io.RawIOBase) -> int:
    """Encode one frame; returns bytes written."""
    body = io.BytesIO()
    body.write(struct.pack("<Q", len(frame)))
    key = frame.schema.key().encode()
    body.write(struct.pack("<H", len(key)))
    body.write(key)
    body.write(struct.pack("<H", frame.num_columns))
    for c in frame.columns:
        if isinstance(c, torch.Tensor):
            payload = _tensor_bytes(c)
            body.write(struct.pack("<BQ", 0, len(payload)))
        elif isinstance(c, BytesColumn):
            # kind 2: varlen byte rows = offsets (i64) then data (u8)
            cc = c.to("cpu").compacted()
            off = _tensor_bytes(cc.offsets)
            dat = _tensor_bytes(cc.data)
            payload = struct.pack("<Q", len(off)) + off + dat
            body.write(struct.pack("<BQ", 2, len(payload)))
        else:
            payload = pickle.dumps(c, protocol=pickle.HIGHEST_PROTOCOL)
            body.write(struct.pack("<BQ", 1, len(payload)))
        body.write(payload)
    raw = body.getvalue()
    crc = zlib.crc32(raw) & 0xFFFFFFFF
    out.write(struct.pack("<I", MAGIC))
    out.write(raw)
    out.write(struct.pack("<I", crc))
    return 8 + len(raw)


def _read_exact(inp: io.RawIOBase, n: int) -> bytes:
    buf = inp.read(n)
    if buf is None or len(buf) != n:
        raise CorruptionError(f"short read: wanted {n}, got "
                              f"{0 if buf is None else len(buf)}")
    return buf


def decode_frame(inp: io.RawIOBase, device: str = "cpu") -> Optional[Frame]:
    """Decode one frame; returns None at clean EOF.  Verifies CRC32."""
    head = inp.read(4)
    if head is None or len(head) == 0:
        return None
    if len(head) != 4:
        raise CorruptionError("truncated magic")
    (magic,) = struct.unpack("<I", head)
    if magic != MAGIC:
        raise CorruptionError(f"bad magic {magic:#x}")
    # Read body incrementally while accumulating CRC.
    crc = 0
    def take(n):
        nonlocal crc
        b = _read_exact(inp, n)
        crc = zlib.crc32(b, crc)
        return b
    (nrows,) = struct.unpack("<Q", take(8))
    (klen,) = struct.unpack("<H", take(2))
    schema = Schema.from_key(take(klen).decode())
    (ncols,) = struct.unpack("<H", take(2))
    cols = []
    for i in range(ncols):
        kind, plen = struct.unpack("<BQ", take(9))
        payload = take(plen)
        if kind == 0:
            t = _tensor_from_bytes(payload, schema.dtypes[i])
            if device != "cpu":
                t = t.to(device, non_blocking=True)
            cols.append(t)
        elif kind == 2:
            (olen,) = struct.unpack("<Q", payload[:8])
            offs = _tensor_from_bytes(payload[8:8 + olen], torch.int64)
            data = _tensor_from_bytes(payload[8 + olen:], torch.uint8)
            col = BytesColumn(data, offs)
            if device != "cpu":
                col = col.to(device, non_blocking=True)
            cols.append(col)
        else:
            cols.append(pickle.loads(payload))
    (want_crc,) = struct.unpack("<I", _read_exact(inp, 4))
    if (crc & 0xFFFFFFFF) != want_crc:
        raise CorruptionError(
            f"checksum mismatch: {crc & 0xFFFFFFFF:#x} != {want_crc:#x}")
    f = Frame(cols, schema.prefix)
    if len(f) != nrows:
        raise CorruptionError(f"row count mismatch {len(f)} != {nrows}")
    return f


def encode_stream(reader, out: io.RawIOBase) -> int:
    """Encode every frame of a reader; returns row count (the count is
    also the store's footer; see runtime.store)."""
    n = 0
    for f in reader:
        encode_frame(f, out)
        n += len(f)
    return n


class DecodingReader:
    """sliceio.Reader over an encoded byte stream."""

    def __init__(self, inp: io.RawIOBase, device: str = "cpu"):
        self.inp = inp
        self.device = device

    def read(self) -> Optional[Frame]:
        return decode_frame(self.inp, self.device)

    def close(self) -> None:
        try:
            self.inp.close()
        except Exception:
            pass

    def __iter__(self):
        while True:
            f = self.read()
            if f is None:
                return
            yield f
