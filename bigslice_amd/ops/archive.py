"""Tar-archive source (reference archive/tarslice/tarslice.go:29-84):
shards stripe over archive entries; each shard re-opens the archive and
keeps every nshard'th entry."""

from __future__ import annotations

import tarfile
from typing import Callable, List

from ..schema import OBJECT, Schema
from ..sliceio import IterReader, Reader
from ..frame import Frame
from .slice_base import Name, Slice, TaskContext


class TarReader(Slice):
    """Slice<name: str, data: bytes> over tar entries.  open_fn() must
    return a fresh binary file object for the archive."""

    def __init__(self, num_shards: int, open_fn: Callable):
        self.open_fn = open_fn
        super().__init__(Schema([OBJECT, OBJECT], prefix=1), num_shards,
                         name=Name("tarreader"))

    def reader(self, shard: int, dep_readers, ctx: TaskContext) -> Reader:
        nshard = self.num_shards
        from .. import config
        chunk = min(ctx.chunk, config.HOST_CHUNK_ROWS)

        def gen():
            names: List[str] = []
            datas: List[bytes] = []
            with tarfile.open(fileobj=self.open_fn(), mode="r|*") as tf:
                for i, entry in enumerate(tf):
                    if not entry.isfile() or i % nshard != shard:
                        continue
                    names.append(entry.name)
                    datas.append(tf.extractfile(entry).read())
                    if len(names) >= chunk:
                        yield Frame([list(names), list(datas)], prefix=1)
                        names.clear()
                        datas.clear()
            if names:
                yield Frame([list(names), list(datas)], prefix=1)
        return IterReader(gen())
