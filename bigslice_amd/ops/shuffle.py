"""Shuffle operators: Reshuffle, Repartition, Reshard.

Role-parity: reshuffle.go:37-88, reshard.go:24-45.  A shuffle dep is a phase
boundary: the producer task partitions each output batch by key hash (fused
hash+scatter HIP kernel, K4) and the exchange runs as an RCCL all-to-all over
xGMI in the distributed executor (or through in-memory partition buffers in
the local executor).  The reader side of these slices is the identity over
the shuffled stream.
"""

from __future__ import annotations

from typing import Callable, Optional

from ..sliceio import Reader
from .slice_base import Dep, Name, Slice, TaskContext


class Reshuffle(Slice):
    """Repartition rows by prefix hash; same schema, same shard count
    (reshuffle.go:37-52)."""

    def __init__(self, dep: Slice, partitioner: Optional[Callable] = None):
        super().__init__(dep.schema, dep.num_shards,
                         deps=[Dep(dep, shuffle=True, partitioner=partitioner)],
                         name=Name("reshuffle"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        return dep_readers[0]


class Repartition(Slice):
    """Reshuffle with a custom partition function (reshuffle.go:52-76).
    ``partition_fn(frame, nshard) -> int32 tensor of shard ids``."""

    def __init__(self, dep: Slice, partition_fn: Callable):
        super().__init__(dep.schema, dep.num_shards,
                         deps=[Dep(dep, shuffle=True,
                                   partitioner=partition_fn)],
                         name=Name("repartition"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        return dep_readers[0]


class Reshard(Slice):
    """Shuffle to a new shard count; no-op when equal (reshard.go:24-45)."""

    def __new__(cls, dep: Slice, num_shards: int):
        if dep.num_shards == num_shards:
            return dep
        return super().__new__(cls)

    def __init__(self, dep: Slice, num_shards: int):
        if self is dep:
            return
        super().__init__(dep.schema, num_shards,
                         deps=[Dep(dep, shuffle=True)],
                         name=Name("reshard"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        return dep_readers[0]
