"""The Slice interface: typed, sharded, lazily-evaluated datasets.

Role-parity with the reference root package (slice.go:78-105 ``Slice``,
:40-49 ``Dep``, :65 ``Partitioner``, :109-200 ``Pragma``, :1114-1173
``Name``).  A Slice describes columns (Schema), sharding, dependencies and a
reader factory; the compiler (runtime.compile) fuses chains of non-shuffle
deps into tasks running on HIP streams.
"""

from __future__ import annotations

import inspect
import threading
from typing import Callable, List, Optional, Sequence

from ..schema import Schema


class Name:
    """Operation identity captured at construction site (slice.go:1114)."""

    __slots__ = ("op", "file", "line", "index")
    _counter = 0
    _lock = threading.Lock()

    def __init__(self, op: str):
        self.op = op
        frame = inspect.currentframe()
        # walk out of bigslice_amd to the user call site
        f = frame
        while f is not None and "bigslice_amd" in (f.f_code.co_filename or ""):
            f = f.f_back
        if f is not None:
            self.file = f.f_code.co_filename
            self.line = f.f_lineno
        else:
            self.file, self.line = "<unknown>", 0
        with Name._lock:
            Name._counter += 1
            self.index = Name._counter

    def __repr__(self):
        return f"{self.op}@{self.file}:{self.line}"


class Dep:
    """A dependency on another slice (slice.go:40-49).

    shuffle: re-partition producer output by key hash (a phase boundary;
    on MI355X this is the RCCL all-to-all over xGMI).
    expand: hand each producer-shard stream to the consumer unmerged
    (needed for merge-sorting streams; reduce.go:70).
    """

    __slots__ = ("slice", "shuffle", "partitioner", "expand")

    def __init__(self, slice_: "Slice", shuffle: bool = False,
                 partitioner: Optional[Callable] = None,
                 expand: bool = False):
        self.slice = slice_
        self.shuffle = shuffle
        self.partitioner = partitioner
        self.expand = expand


class Pragma:
    """Execution directives (slice.go:109-200)."""

    __slots__ = ("procs", "exclusive", "materialize")

    def __init__(self, procs: int = 1, exclusive: bool = False,
                 materialize: bool = False):
        self.procs = procs
        self.exclusive = exclusive
        self.materialize = materialize


DEFAULT_PRAGMA = Pragma()


def exclusive(slice_: "Slice") -> "Slice":
    """Mark a slice's tasks Exclusive (slice.go Exclusive pragma)."""
    slice_.pragma = Pragma(procs=slice_.pragma.procs, exclusive=True,
                           materialize=slice_.pragma.materialize)
    return slice_


def procs(slice_: "Slice", n: int) -> "Slice":
    """Reserve n procs per task (slice.go Procs pragma)."""
    slice_.pragma = Pragma(procs=n, exclusive=slice_.pragma.exclusive,
                           materialize=slice_.pragma.materialize)
    return slice_


def materialize(slice_: "Slice") -> "Slice":
    """Break pipelining below this slice (ExperimentalMaterialize)."""
    slice_.pragma = Pragma(procs=slice_.pragma.procs,
                           exclusive=slice_.pragma.exclusive,
                           materialize=True)
    return slice_


class TaskContext:
    """Per-task execution context handed to readers: target device,
    batch size, metrics scope and (on GPU) the HIP stream."""

    def __init__(self, device: str = "cpu", chunk: int = None,
                 scope=None, shard_info=None):
        from .. import config
        self.device = device
        self.chunk = chunk or config.chunk_rows(device)
        self.scope = scope  # metrics scope or None
        self.shard_info = shard_info


class Slice:
    """Base class for all slice operators."""

    def __init__(self, schema: Schema, num_shards: int,
                 deps: Sequence[Dep] = (), name: Optional[Name] = None,
                 pragma: Pragma = DEFAULT_PRAGMA,
                 combiner: Optional[object] = None):
        self._schema = schema
        self._num_shards = num_shards
        self._deps = list(deps)
        self.name = name or Name(type(self).__name__.lower())
        self.pragma = pragma
        self._combiner = combiner

    # -- Slice interface (slice.go:78-105) --------------------------------

    @property
    def schema(self) -> Schema:
        return self._schema

    @property
    def num_shards(self) -> int:
        return self._num_shards

    @property
    def num_deps(self) -> int:
        return len(self._deps)

    def dep(self, i: int) -> Dep:
        return self._deps[i]

    @property
    def deps(self) -> List[Dep]:
        return self._deps

    @property
    def combiner(self):
        """A Combiner spec (ops.reduce.Aggregation) or None.  Non-None
        triggers combine-on-partition machinery (reference Reduce)."""
        return self._combiner

    def reader(self, shard: int, dep_readers: List, ctx: TaskContext):
        """Return a sliceio.Reader for the given shard, composing the
        dep readers.  dep_readers[i] corresponds to deps[i]; for expand
        deps it is a list of per-producer-shard readers."""
        raise NotImplementedError

    # convenience chaining API ------------------------------------------------

    def map(self, fn, **kw):
        from .elementwise import Map
        return Map(self, fn, **kw)

    def filter(self, fn, **kw):
        from .elementwise import Filter
        return Filter(self, fn, **kw)

    def flatmap(self, fn, **kw):
        from .elementwise import Flatmap
        return Flatmap(self, fn, **kw)

    def reduce(self, fn, **kw):
        from .reduce import Reduce
        return Reduce(self, fn, **kw)

    def fold(self, fn, **kw):
        from .reduce import Fold
        return Fold(self, fn, **kw)

    def reshuffle(self, **kw):
        from .shuffle import Reshuffle
        return Reshuffle(self, **kw)

    def reshard(self, num_shards: int, **kw):
        from .shuffle import Reshard
        return Reshard(self, num_shards, **kw)

    def head(self, n: int, **kw):
        from .elementwise import Head
        return Head(self, n, **kw)

    def prefixed(self, prefix: int):
        from .elementwise import Prefixed
        return Prefixed(self, prefix)

    def __repr__(self):
        return (f"{type(self).__name__}({self.schema}, "
                f"shards={self.num_shards})")


def unwrap(slice_: Slice) -> Slice:
    """Peel adapter slices (reference bigslice.Unwrap; slice.go:1066).
    Slices that merely decorate another slice override ``unwrap_target``."""
    while True:
        target = getattr(slice_, "unwrap_target", None)
        if target is None:
            return slice_
        slice_ = target
