"""Column-tuple schemas and type checking.

Role-parity with the reference's slicetype package (slicetype/slicetype.go:17-27:
a Type is an ordered tuple of column types with a key ``prefix``) and
typecheck/ (panic-with-location type errors).  Columns here are torch dtypes
(device-resident numeric data in HBM) or the sentinel ``OBJECT`` for host-side
Python objects (strings, tuples, ...), which flow on the CPU fallback path.
"""

from __future__ import annotations

from typing import Iterable, Sequence, Tuple

import torch

# Sentinel dtype for host-object columns (strings and arbitrary Python values).
OBJECT = "object"

# Sentinel dtype for DEVICE-RESIDENT variable-length byte rows
# (frame.BytesColumn: one uint8 data tensor + int64 offsets).  Unlike
# OBJECT, BYTES columns hash, partition, sort (by 64-bit dictionary
# id), exchange and store on the device path — the first-class string
# type of the reference (frame/ops_builtin.go:143-164), MI355X-native.
BYTES = "bytes"

_TORCH_DTYPES = {
    torch.int8, torch.uint8, torch.int16, torch.int32, torch.int64,
    torch.float16, torch.bfloat16, torch.float32, torch.float64,
    torch.bool, torch.uint32, torch.uint64,
}

# Canonical short names for dtypes (used in task names and wire headers).
_DTYPE_NAMES = {
    torch.int8: "i8", torch.uint8: "u8", torch.int16: "i16",
    torch.int32: "i32", torch.int64: "i64", torch.uint32: "u32",
    torch.uint64: "u64", torch.float16: "f16", torch.bfloat16: "bf16",
    torch.float32: "f32", torch.float64: "f64", torch.bool: "b1",
    OBJECT: "obj", BYTES: "byt",
}
_NAME_DTYPES = {v: k for k, v in _DTYPE_NAMES.items()}


class TypeError_(TypeError):
    """Type error raised by slice construction, with op location context."""


def dtype_name(dt) -> str:
    return _DTYPE_NAMES[dt]


def dtype_from_name(name: str):
    return _NAME_DTYPES[name]


def is_object(dt) -> bool:
    return dt == OBJECT


def is_bytes(dt) -> bool:
    return dt == BYTES


def check_dtype(dt):
    if dt != OBJECT and dt != BYTES and dt not in _TORCH_DTYPES:
        raise TypeError_(f"unsupported column dtype {dt!r}")
    return dt


class Schema:
    """An ordered tuple of column dtypes with a key prefix.

    ``prefix`` is the number of leading key columns (reference
    slicetype.Type.Prefix; default 1 when there are columns, like the
    reference's slices).
    """

    __slots__ = ("dtypes", "prefix")

    def __init__(self, dtypes: Iterable, prefix: int = None):
        self.dtypes: Tuple = tuple(check_dtype(d) for d in dtypes)
        if prefix is None:
            prefix = 1 if self.dtypes else 0
        if not (0 <= prefix <= len(self.dtypes)):
            raise TypeError_(
                f"invalid prefix {prefix} for {len(self.dtypes)} columns")
        self.prefix = prefix

    @property
    def num_columns(self) -> int:
        return len(self.dtypes)

    def with_prefix(self, prefix: int) -> "Schema":
        return Schema(self.dtypes, prefix)

    def __eq__(self, other):
        return (isinstance(other, Schema) and self.dtypes == other.dtypes
                and self.prefix == other.prefix)

    def __hash__(self):
        return hash((self.dtypes, self.prefix))

    def __repr__(self):
        names = ",".join(dtype_name(d) for d in self.dtypes)
        return f"Schema({names}; prefix={self.prefix})"

    def key(self) -> str:
        """Stable string form (for wire headers / task names)."""
        return ",".join(dtype_name(d) for d in self.dtypes) + f";{self.prefix}"

    @staticmethod
    def from_key(key: str) -> "Schema":
        cols, prefix = key.rsplit(";", 1)
        dts = [dtype_from_name(n) for n in cols.split(",")] if cols else []
        return Schema(dts, int(prefix))

    def concat(self, other: "Schema") -> "Schema":
        """Concatenate column tuples (reference slicetype.Concat)."""
        return Schema(self.dtypes + other.dtypes, self.prefix)


def schemas_compatible(a: Schema, b: Schema) -> bool:
    """Column-type equality ignoring prefix (reference typecheck.Equal)."""
    return a.dtypes == b.dtypes


def infer_dtype(value):
    """Infer a column dtype from a Python scalar value."""
    if isinstance(value, bool):
        return torch.bool
    if isinstance(value, int):
        return torch.int64
    if isinstance(value, float):
        return torch.float64
    return OBJECT


def common_key_schema(schemas: Sequence[Schema]) -> Tuple:
    """Check that all schemas share identical key-prefix column types
    (required by Cogroup; reference cogroup.go:88-110).  Returns the key
    dtype tuple."""
    if not schemas:
        raise TypeError_("no slices")
    keys = schemas[0].dtypes[: schemas[0].prefix]
    for s in schemas[1:]:
        if s.dtypes[: s.prefix] != keys:
            raise TypeError_(
                f"mismatched key prefixes: {keys} vs {s.dtypes[: s.prefix]}")
    return keys
