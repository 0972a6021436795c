"""Compiler: slice DAG -> task DAG.

Role-parity: exec/compile.go — pipeline fusion of non-shuffle/non-materialize
dep chains (:29-48), memoization per (slice, num_partitions) (:50-56),
*Result task reuse with pass-through _shuffle insertion (:226-261), shuffle
deps compiling the producer with the consumer's partition count + combiner
(:317-333), per-(task,shard) cache short-circuit (:359-368), and the frozen
CompileEnv shipped to every worker so all processes compile identical graphs
(:125-184).

A task is one shard of a fused pipeline; on GPU each task is a stream of
device batches flowing through the fused readers on a HIP stream.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

from ..ops.cache import Cache
from ..ops.slice_base import Slice, TaskContext, unwrap
from .task import Task, TaskDep


class CompileEnv:
    """Compile decisions frozen by the driver and shared by all workers
    (exec/compile.go:125-184): cache hits must not be locally derived."""

    def __init__(self):
        self.cached: Dict[Tuple[str, int], bool] = {}
        self.frozen = False
        self.sealed = False

    def freeze(self):
        self.frozen = True

    def seal(self, decisions: Dict[Tuple[str, int], bool]):
        """Install driver-broadcast decisions and forbid local probing.
        In SPMD mode only rank 0 stats the filesystem; other ranks
        compile against the sealed env so a racing cache write cannot
        produce divergent graphs (exec/compile.go:125-184)."""
        self.cached = dict(decisions)
        self.sealed = True

    def cache_decision(self, key: str, shard: int,
                       compute: Callable[[], List[bool]]) -> bool:
        k = (key, shard)
        if k not in self.cached:
            if self.sealed:
                raise RuntimeError(
                    f"cache decision for {k} missing from the "
                    "driver-broadcast CompileEnv: this rank compiled a "
                    "different graph than rank 0 (non-deterministic "
                    "Func?)")
            if self.frozen:
                return False
            decisions = compute()
            for s, d in enumerate(decisions):
                self.cached[(key, s)] = d
        return self.cached[k]


def pipeline_slices(slice_: Slice) -> List[Slice]:
    """The fusable chain starting at slice_ (exec/compile.go:29-48):
    slices[0] is the top; each depends on the next.  Stops at shuffle
    deps, multi-deps, materialize pragmas and *Result slices."""
    from .session import Result
    out: List[Slice] = []
    while True:
        if isinstance(unwrap(slice_), Result):
            return out
        out.append(slice_)
        if slice_.num_deps != 1:
            return out
        dep = slice_.dep(0)
        if dep.shuffle:
            return out
        if dep.slice.pragma.materialize:
            return out
        slice_ = dep.slice


class _TaskNamer:
    """Unique task-name assignment (exec/compile.go:271-279)."""

    def __init__(self, inv_index: int):
        self.inv = inv_index
        self.used: Dict[str, int] = {}

    def name(self, ops: List[str], nshard: int) -> str:
        base = f"inv{self.inv}_" + "_".join(ops)
        n = self.used.get(base, 0)
        self.used[base] = n + 1
        if n:
            base = f"{base}{n}"
        return f"{base}@{nshard}"


class Compiler:
    def __init__(self, inv_index: int, env: Optional[CompileEnv] = None):
        self.inv_index = inv_index
        self.env = env or CompileEnv()
        self.namer = _TaskNamer(inv_index)
        self.memo: Dict[Tuple[int, int], List[Task]] = {}

    def compile(self, slice_: Slice) -> List[Task]:
        """Compile the root slice; returns its tasks (one per shard)."""
        return self._compile(slice_, num_partitions=1, partitioner=None,
                             combiner=None, shuffle_out=False)

    def _compile(self, slice_: Slice, num_partitions: int,
                 partitioner, combiner, shuffle_out: bool = False
                 ) -> List[Task]:
        from .session import Result
        target = unwrap(slice_)
        if isinstance(target, Result):
            return self._reuse_result(target, num_partitions, partitioner,
                                      combiner, shuffle_out)
        # Memoization key: a producer compiled for a shuffle carries the
        # consumer's partitioner AND combiner, so consumers with
        # different combine specs (e.g. Reduce sum vs max over one
        # slice) must compile separate producer tasks.
        comb_key = combiner.key() if combiner is not None else None
        key = (id(slice_), num_partitions, shuffle_out,
               id(partitioner) if partitioner is not None else None,
               comb_key)
        if key in self.memo:
            return self.memo[key]

        chain = pipeline_slices(slice_)
        if not chain:  # slice unwraps straight to a Result
            return self._reuse_result(target, num_partitions, partitioner,
                                      combiner, shuffle_out)
        bottom = chain[-1]
        nshard = chain[0].num_shards

        # Compile dependencies of the bottom slice.
        dep_task_lists: List[List[Task]] = []
        dep_infos = []
        for i in range(bottom.num_deps):
            dep = bottom.dep(i)
            if dep.shuffle:
                dtasks = self._compile(
                    dep.slice, num_partitions=bottom.num_shards,
                    partitioner=dep.partitioner,
                    combiner=bottom.combiner, shuffle_out=True)
            else:
                dtasks = self._compile(dep.slice, 1, None, None,
                                       shuffle_out=False)
                if dep.slice.num_shards != bottom.num_shards:
                    raise ValueError(
                        f"non-shuffle dep shard mismatch: "
                        f"{dep.slice.num_shards} != {bottom.num_shards}")
            dep_task_lists.append(dtasks)
            dep_infos.append(dep)

        ops = [s.name.op for s in reversed(chain)]
        base = self.namer.name(ops, nshard)

        # Per-shard cache short-circuit: find shallowest cached Cache.
        tasks: List[Task] = []
        group: List[Task] = []
        for shard in range(nshard):
            cut_at = None  # index into chain of cached Cache slice
            for ci, s in enumerate(chain):
                if isinstance(s, Cache):
                    if self.env.cache_decision(
                            f"{s.name.op}:{s.cache.prefix}", shard,
                            s.cache_decisions):
                        cut_at = ci
                        break
            if cut_at is not None:
                eff_chain = chain[:cut_at + 1]
                task_deps: List[TaskDep] = []
            else:
                eff_chain = chain
                task_deps = []
                for dep, dtasks in zip(dep_infos, dep_task_lists):
                    if dep.shuffle:
                        task_deps.append(TaskDep(
                            dtasks, partition=shard, expand=dep.expand,
                            combiner=bottom.combiner))
                    else:
                        task_deps.append(TaskDep([dtasks[shard]],
                                                 partition=0))
            do = _make_do(eff_chain, shard,
                          cached=cut_at is not None)
            t = Task(
                name=f"{base}:{shard}",
                invocation_index=self.inv_index,
                do=do, deps=task_deps,
                num_partitions=num_partitions,
                partitioner=partitioner,
                combiner=combiner,
                shuffle_out=shuffle_out,
                group=group,
                num_out_columns=chain[0].schema.num_columns,
                pragma=chain[0].pragma,
                schema=chain[0].schema,
                shard=shard, num_shards=nshard)
            group.append(t)
            tasks.append(t)
        self.memo[key] = tasks
        return tasks

    def _reuse_result(self, result, num_partitions: int, partitioner,
                      combiner, shuffle_out: bool = False) -> List[Task]:
        """Reuse a prior invocation's tasks (exec/compile.go:226-261),
        inserting pass-through _shuffle tasks when the consumer needs a
        different partitioning."""
        prev = result.tasks
        needs_shuffle = (shuffle_out and any(
            t.num_partitions != num_partitions or
            t.combiner is not combiner or not t.shuffle_out
            for t in prev))
        if not needs_shuffle:
            return prev
        # the pass-through tasks carry the consumer's partitioner AND
        # combiner: consumers with different combine specs must not
        # share them (same rule as the main memo key)
        comb_key = combiner.key() if combiner is not None else None
        key = (id(result), num_partitions,
               id(partitioner) if partitioner is not None else None,
               comb_key)
        if key in self.memo:
            return self.memo[key]
        nshard = len(prev)
        base = self.namer.name(["_shuffle"], nshard)
        group: List[Task] = []
        tasks: List[Task] = []
        for shard in range(nshard):
            def do(dep_readers, ctx, _shard=shard):
                return dep_readers[0]
            t = Task(
                name=f"{base}:{shard}",
                invocation_index=self.inv_index,
                do=do,
                deps=[TaskDep([prev[shard]], partition=0)],
                num_partitions=num_partitions,
                partitioner=partitioner,
                combiner=combiner,
                shuffle_out=True,
                group=group,
                num_out_columns=prev[shard].num_out_columns,
                schema=prev[shard].schema,
                shard=shard, num_shards=nshard)
            group.append(t)
            tasks.append(t)
        self.memo[key] = tasks
        return tasks


def _make_do(chain: List[Slice], shard: int, cached: bool) -> Callable:
    """Compose the fused reader pipeline for one shard
    (exec/compile.go:338-385)."""

    def do(dep_readers: List, ctx: TaskContext):
        if cached:
            cache_slice: Cache = chain[-1]
            reader = cache_slice.cached_reader(shard, ctx)
            upper = chain[:-1]
        else:
            bottom = chain[-1]
            reader = bottom.reader(shard, dep_readers, ctx)
            upper = chain[:-1]
        for s in reversed(upper):
            reader = s.reader(shard, [reader], ctx)
        return reader

    return do
