"""Cache / CachePartial / ReadCache: per-shard file checkpoints.

Role-parity: cache.go:45-100 + internal/slicecache — shard `i` of `n` is
checkpointed to ``{prefix}-{i:04d}-of-{n:04d}`` files (the reference's
task-checkpoint naming, kept verbatim per BASELINE.json); presence is
stat-checked up front, cached shards compile to cache-reading tasks with
their deps cut off (exec/compile.go:359-368), and partial caches recompute
only missing shards.  Cache-or-not decisions are frozen into the CompileEnv
so every worker process compiles the identical graph.
"""

from __future__ import annotations

import os
from typing import List

from ..sliceio import IterReader, Reader, codec
from .slice_base import Dep, Name, Slice, TaskContext


class ShardCache:
    """File-backed shard cache (internal/slicecache/slicecache.go)."""

    def __init__(self, prefix: str, num_shards: int):
        self.prefix = prefix
        self.num_shards = num_shards

    def path(self, shard: int) -> str:
        return f"{self.prefix}-{shard:04d}-of-{self.num_shards:04d}"

    def present(self) -> List[bool]:
        return [os.path.exists(self.path(s))
                for s in range(self.num_shards)]

    def write_through(self, shard: int, reader: Reader) -> Reader:
        """Tee a stream into the cache file; commit (rename) at EOF so
        partial writes are never observed."""
        path = self.path(shard)
        tmp = path + ".tmp"
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)

        def gen():
            with open(tmp, "wb") as fp:
                for f in reader:
                    codec.encode_frame(f.to("cpu"), fp)
                    yield f
            os.replace(tmp, path)
        return IterReader(gen())

    def read(self, shard: int, device: str) -> Reader:
        fp = open(self.path(shard), "rb")
        return codec.DecodingReader(fp, device)


class Cache(Slice):
    """Fully cache a slice; all shards must be cached or all recomputed
    (cache.go:45-68).  ``partial=True`` gives CachePartial semantics:
    only missing shards are recomputed (cache.go:70-89)."""

    def __init__(self, dep: Slice, prefix: str, partial: bool = False):
        self.cache = ShardCache(prefix, dep.num_shards)
        self.partial = partial
        self.unwrap_target = dep
        super().__init__(dep.schema, dep.num_shards, deps=[Dep(dep)],
                         name=Name("cache"))

    def cache_decisions(self) -> List[bool]:
        """Frozen at compile time into the CompileEnv."""
        present = self.cache.present()
        if self.partial:
            return present
        return [all(present)] * self.num_shards

    def cached_reader(self, shard: int, ctx: TaskContext) -> Reader:
        return self.cache.read(shard, ctx.device)

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        # Uncached shard: compute through the dep and write through.
        return self.cache.write_through(shard, dep_readers[0])


def ReadCache(schema, num_shards: int, prefix: str) -> Slice:
    """Read a previously written cache without a compute fallback
    (cache.go:91-100)."""
    cache = ShardCache(prefix, num_shards)

    class _ReadCacheSlice(Slice):
        def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
            return cache.read(shard, ctx.device)

    return _ReadCacheSlice(schema, num_shards, name=Name("readcache"))
