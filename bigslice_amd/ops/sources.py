"""Source slices: Const, ReaderFunc, ScanReader.

Role-parity: slice.go:212-290 (Const: splits materialized columns evenly
across shards), slice.go:321-402 (ReaderFunc: vectorized user source with
per-shard state), scan.go:22-69 (ScanReader: line-sharded text source).
"""

from __future__ import annotations

from typing import Callable, Iterable, List

import torch

from ..frame import Frame
from ..schema import OBJECT, Schema, infer_dtype
from ..sliceio import FrameReader, IterReader, Reader, EmptyReader
from .slice_base import Name, Slice, TaskContext


def _columns_to_frame(cols, prefix=None) -> Frame:
    if isinstance(cols, Frame):
        return cols
    if isinstance(cols, (tuple, list)):
        from ..frame import BytesColumn, SegmentedColumn
        conv = []
        for c in cols:
            if isinstance(c, (torch.Tensor, BytesColumn,
                              SegmentedColumn)):
                conv.append(c)
            elif isinstance(c, (list, tuple)):
                if c and isinstance(c[0], (str, bytes, tuple)) or not c:
                    conv.append(list(c))
                else:
                    dt = infer_dtype(c[0])
                    conv.append(list(c) if dt == OBJECT
                                else torch.tensor(c, dtype=dt))
            else:
                raise TypeError(f"bad column type {type(c)}")
        return Frame(conv, prefix)
    raise TypeError(f"reader function yielded {type(cols)}; "
                    "expected Frame or tuple of columns")


class Const(Slice):
    """A constant slice: provided columns split evenly over nshard
    (slice.go:212-290 constShard)."""

    def __init__(self, num_shards: int, *cols, prefix: int = None):
        frame = _columns_to_frame(cols if len(cols) != 1 or
                                  not isinstance(cols[0], Frame) else cols[0],
                                  prefix)
        self.frame = frame
        if num_shards < 1:
            raise ValueError("Const: num_shards must be >= 1")
        super().__init__(frame.schema, num_shards, name=Name("const"))

    def reader(self, shard: int, dep_readers, ctx: TaskContext) -> Reader:
        n = len(self.frame)
        per = (n + self.num_shards - 1) // self.num_shards
        start = min(shard * per, n)
        stop = min(start + per, n)
        if start >= stop:
            return EmptyReader()
        sub = self.frame.slice(start, stop)
        if ctx.device != "cpu" and not sub.has_objects:
            sub = sub.to(ctx.device)
        return FrameReader(sub, ctx.chunk)


class ReaderFunc(Slice):
    """Vectorized user source: ``fn(shard, ctx) -> iterable of
    Frame | tuple-of-columns`` (slice.go:321-402).  The per-shard generator
    state replaces the reference's reflected state pointer."""

    def __init__(self, num_shards: int, fn: Callable, schema: Schema,
                 prefix: int = None):
        if prefix is not None:
            schema = schema.with_prefix(prefix)
        self.fn = fn
        super().__init__(schema, num_shards, name=Name("reader"))

    def reader(self, shard: int, dep_readers, ctx: TaskContext) -> Reader:
        prefix = self.schema.prefix

        def gen():
            it = self.fn(shard, ctx)
            empties = 0
            for item in it:
                f = _columns_to_frame(item, prefix)
                if len(f) == 0:
                    # warn on busy-looping sources (slice.go:382-385)
                    empties += 1
                    if empties == 8:
                        import logging
                        logging.getLogger("bigslice_amd").warning(
                            "reader func returned 8 consecutive empty "
                            "batches (shard %d)", shard)
                    continue
                empties = 0
                if ctx.device != "cpu" and not f.has_objects:
                    f = f.to(ctx.device, non_blocking=True)
                yield f
        return IterReader(gen())


class ScanReader(Slice):
    """Line-sharded text source producing one string column
    (scan.go:22-69): each shard re-reads the stream, keeping every
    nshard'th line starting at its own index."""

    def __init__(self, num_shards: int, open_fn: Callable[[], Iterable[str]]):
        self.open_fn = open_fn
        super().__init__(Schema([OBJECT]), num_shards, name=Name("scanreader"))

    def reader(self, shard: int, dep_readers, ctx: TaskContext) -> Reader:
        nshard = self.num_shards
        from .. import config
        # object columns are host-resident regardless of ctx device:
        # batch at host granularity
        chunk = min(ctx.chunk, config.HOST_CHUNK_ROWS)

        def gen():
            buf: List[str] = []
            for i, line in enumerate(self.open_fn()):
                if i % nshard != shard:
                    continue
                buf.append(line.rstrip("\n"))
                if len(buf) >= chunk:
                    yield Frame([list(buf)])
                    buf.clear()
            if buf:
                yield Frame([list(buf)])
        return IterReader(gen())
