"""Elementwise/pipelined operators: Map, Filter, Flatmap, Head, Scan,
WriterFunc, Prefixed.

Role-parity: slice.go:566-637 (Map), :657-725 (Filter), :745-841 (Flatmap),
:966-994 (Head), :1005-1032 (Scan), :443-548 (WriterFunc), :1044-1071
(Prefixed).

The reference applies reflected UDFs row-at-a-time over 128-row chunks; that
has no efficient GPU analog.  The MI355X-native UDF convention is
*vectorized*: the function receives whole device columns (torch tensors in
HBM) and returns columns — torch ops dispatch HIP kernels, so user code runs
device-native.  ``rowwise=True`` selects a host fallback for object columns
(strings etc.), matching the reference's semantics exactly.
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch

from ..frame import Frame
from ..schema import OBJECT, Schema, is_object
from ..sliceio import IterReader, Reader
from .slice_base import Dep, Name, Slice, TaskContext


def schema_of(*pytypes, prefix: int = None) -> Schema:
    """Build a Schema from Python types or torch dtypes:
    str->OBJECT (host path), bytes->BYTES (device varlen byte rows),
    int->int64, float->float64, bool->bool."""
    from ..schema import BYTES
    dts = []
    for t in pytypes:
        if t is bytes or t == BYTES:
            dts.append(BYTES)
        elif t is str or t is object or t == OBJECT:
            dts.append(OBJECT)
        elif t is int:
            dts.append(torch.int64)
        elif t is float:
            dts.append(torch.float64)
        elif t is bool:
            dts.append(torch.bool)
        else:
            dts.append(t)  # assume torch dtype; Schema validates
    return Schema(dts, prefix)


def _warn_host_rowwise(op: str) -> None:
    """One-time (per call site) notice that a rowwise UDF leaves the
    device path on a GPU session: rows round-trip through Python."""
    import warnings
    warnings.warn(
        f"{op}(rowwise=True) runs its UDF as a host row loop on a GPU "
        "session (rows leave the device); prefer a vectorized UDF over "
        "whole columns", RuntimeWarning, stacklevel=3)


def _normalize_out(res, device_hint: str) -> List:
    """Normalize a UDF result into a column list."""
    if isinstance(res, Frame):
        return list(res.columns)
    if isinstance(res, torch.Tensor):
        return [res]
    if isinstance(res, (tuple, list)):
        out = []
        for c in res:
            if isinstance(c, torch.Tensor) or isinstance(c, list):
                out.append(c)
            elif isinstance(c, (int, float, bool)):
                raise TypeError(
                    "vectorized UDF returned a scalar; return columns "
                    "(tensors/lists) or use rowwise=True")
            else:
                out.append(list(c))
        return out
    raise TypeError(f"UDF returned {type(res)}")


def _infer_schema_vectorized(fn, in_schema: Schema, prefix: int) -> Optional[Schema]:
    """Try inferring output schema by applying fn to an empty frame."""
    try:
        empty = Frame.empty(in_schema)
        res = fn(*empty.columns)
        cols = _normalize_out(res, "cpu")
        dts = [c.dtype if isinstance(c, torch.Tensor) else OBJECT
               for c in cols]
        return Schema(dts, min(prefix, len(dts)) or None)
    except Exception:
        return None


class _PipelinedSlice(Slice):
    """Base for ops with a single non-shuffle dep."""

    def __init__(self, dep_slice: Slice, schema: Schema, op: str,
                 combiner=None):
        super().__init__(schema, dep_slice.num_shards,
                         deps=[Dep(dep_slice)], name=Name(op),
                         combiner=combiner)


class Map(_PipelinedSlice):
    def __init__(self, dep: Slice, fn: Callable, out_schema=None,
                 rowwise: bool = False, prefix: int = None):
        self.fn = fn
        self.rowwise = rowwise
        in_schema = dep.schema
        if out_schema is not None:
            schema = out_schema if isinstance(out_schema, Schema) \
                else schema_of(*out_schema, prefix=prefix)
        elif rowwise:
            raise TypeError("rowwise Map requires out_schema")
        else:
            schema = _infer_schema_vectorized(fn, in_schema,
                                              prefix or in_schema.prefix)
            if schema is None:
                raise TypeError(
                    "could not infer Map output schema; pass out_schema")
        if prefix is not None:
            schema = schema.with_prefix(prefix)
        super().__init__(dep, schema, "map")

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        fn, rowwise, schema = self.fn, self.rowwise, self.schema
        if rowwise and ctx.device != "cpu":
            _warn_host_rowwise("Map")

        def gen():
            for f in src:
                if rowwise:
                    out_rows = [fn(*row) for row in zip(*f.column_lists())] \
                        if f.num_columns > 1 else [fn(v) for v in
                                                   f.column_lists()[0]]
                    cols = _rows_to_columns(out_rows, schema, ctx.device)
                else:
                    cols = _normalize_out(fn(*f.columns), ctx.device)
                    cols = _coerce_columns(cols, schema, ctx.device)
                yield Frame(cols, schema.prefix)
        return IterReader(gen())


class Filter(_PipelinedSlice):
    def __init__(self, dep: Slice, fn: Callable, rowwise: bool = False):
        self.fn = fn
        self.rowwise = rowwise
        super().__init__(dep, dep.schema, "filter")

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        fn, rowwise = self.fn, self.rowwise
        if rowwise and ctx.device != "cpu":
            _warn_host_rowwise("Filter")

        def gen():
            for f in src:
                if rowwise:
                    lists = f.column_lists()
                    keep = torch.tensor(
                        [bool(fn(*row)) for row in zip(*lists)]
                        if f.num_columns > 1 else
                        [bool(fn(v)) for v in lists[0]], dtype=torch.bool)
                else:
                    keep = fn(*f.columns)
                    if not isinstance(keep, torch.Tensor):
                        keep = torch.tensor(list(keep), dtype=torch.bool)
                out = f.mask(keep)
                if len(out):
                    yield out
        return IterReader(gen())


class Flatmap(_PipelinedSlice):
    """Vector-output UDF.  Vectorized form: fn(*cols) -> columns of any
    (uniform) length.  Rowwise form: fn(*row) -> iterable of rows.
    fn_factory, when given, builds a fresh fn per shard reader (for
    UDFs carrying per-shard state, the reference's ReaderFunc-state
    analog for flatmaps)."""

    def __init__(self, dep: Slice, fn: Callable, out_schema=None,
                 rowwise: bool = False, prefix: int = None,
                 fn_factory: Callable = None):
        self.fn = fn
        self.fn_factory = fn_factory
        self.rowwise = rowwise
        if out_schema is not None:
            schema = out_schema if isinstance(out_schema, Schema) \
                else schema_of(*out_schema, prefix=prefix)
        elif rowwise:
            raise TypeError("rowwise Flatmap requires out_schema")
        else:
            schema = _infer_schema_vectorized(fn, dep.schema,
                                              prefix or dep.schema.prefix)
            if schema is None:
                raise TypeError(
                    "could not infer Flatmap output schema; pass out_schema")
        if prefix is not None:
            schema = schema.with_prefix(prefix)
        super().__init__(dep, schema, "flatmap")

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        fn, rowwise, schema = self.fn, self.rowwise, self.schema
        if self.fn_factory is not None:
            fn = self.fn_factory()
        if rowwise and ctx.device != "cpu":
            _warn_host_rowwise("Flatmap")

        def gen():
            for f in src:
                if rowwise:
                    out_rows: list = []
                    lists = f.column_lists()
                    it = zip(*lists) if f.num_columns > 1 else \
                        ((v,) for v in lists[0])
                    for row in it:
                        for orow in fn(*row):
                            out_rows.append(orow)
                    if out_rows:
                        cols = _rows_to_columns(out_rows, schema, ctx.device)
                        yield Frame(cols, schema.prefix)
                else:
                    cols = _normalize_out(fn(*f.columns), ctx.device)
                    cols = _coerce_columns(cols, schema, ctx.device)
                    out = Frame(cols, schema.prefix)
                    if len(out):
                        yield out
        return IterReader(gen())


class Head(_PipelinedSlice):
    """First n rows per shard (slice.go:966-994)."""

    def __init__(self, dep: Slice, n: int):
        self.n = n
        super().__init__(dep, dep.schema, "head")

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        n = self.n

        def gen():
            left = n
            for f in src:
                if left <= 0:
                    return
                if len(f) > left:
                    f = f.slice(0, left)
                left -= len(f)
                yield f
        return IterReader(gen())


class Scan(Slice):
    """Terminal consumer: fn(shard, row_iterator) (slice.go:1005-1032).
    Produces zero columns; its tasks are driven, not read."""

    def __init__(self, dep: Slice, fn: Callable):
        self.fn = fn
        super().__init__(Schema([]), dep.num_shards, deps=[Dep(dep)],
                         name=Name("scan"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        fn = self.fn

        def gen():
            from ..sliceio import Scanner
            fn(shard, Scanner(src).rows())
            return
            yield  # pragma: no cover
        return IterReader(gen())


class WriterFunc(Slice):
    """Side-effecting pass-through (slice.go:443-548): write_fn(shard,
    frame) is invoked on every batch, and the rows flow on."""

    def __init__(self, dep: Slice, write_fn: Callable):
        self.write_fn = write_fn
        super().__init__(dep.schema, dep.num_shards, deps=[Dep(dep)],
                         name=Name("writer"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        write_fn = self.write_fn
        schema = self.schema

        def gen():
            for f in src:
                write_fn(shard, f)
                yield f
            # the reference invokes the writer on the EOF read too
            # (slice.go:516-539), letting writers flush
            write_fn(shard, Frame.empty(schema))
        return IterReader(gen())


class Prefixed(Slice):
    """Widen the key prefix (slice.go:1044-1071)."""

    def __init__(self, dep: Slice, prefix: int):
        if not (0 < prefix <= dep.schema.num_columns):
            raise TypeError(f"invalid prefix {prefix}")
        self.unwrap_target = dep
        super().__init__(dep.schema.with_prefix(prefix), dep.num_shards,
                         deps=[Dep(dep)], name=Name("prefixed"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        prefix = self.schema.prefix

        def gen():
            for f in src:
                yield f.with_prefix(prefix)
        return IterReader(gen())


def _rows_to_columns(rows: List, schema: Schema, device: str) -> List:
    """Transpose rows into typed columns."""
    ncol = schema.num_columns
    if ncol == 1 and rows and not isinstance(rows[0], (tuple, list)):
        rows = [(r,) for r in rows]
    cols: List = []
    for i, dt in enumerate(schema.dtypes):
        vals = [r[i] for r in rows]
        if is_object(dt):
            cols.append(vals)
        else:
            t = torch.tensor(vals, dtype=dt)
            if device != "cpu":
                t = t.to(device, non_blocking=True)
            cols.append(t)
    return cols


def _coerce_columns(cols: List, schema: Schema, device: str) -> List:
    if len(cols) != schema.num_columns:
        raise TypeError(
            f"UDF returned {len(cols)} columns, schema has "
            f"{schema.num_columns}")
    out = []
    for c, dt in zip(cols, schema.dtypes):
        if is_object(dt):
            out.append(c if isinstance(c, list) else c.cpu().tolist())
        else:
            if not isinstance(c, torch.Tensor):
                c = torch.tensor(c, dtype=dt)
            elif c.dtype != dt:
                c = c.to(dt)
            if device != "cpu" and not c.is_cuda:
                c = c.to(device, non_blocking=True)
            out.append(c)
    return out
