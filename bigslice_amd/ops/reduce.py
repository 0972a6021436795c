"""Keyed reduction operators: Reduce and Fold.

Role-parity: reduce.go:42-78 (Reduce: shuffle dep with a declared combiner;
consumer merges pre-combined per-producer streams) and slice.go:870-955
(Fold: per-shard accumulator over a shuffle dep).

MI355X-native redesign: instead of the reference's sort+merge-combine
(sortio.Reduce over pre-sorted streams), both sides combine through the hash
aggregator (ops.aggregate): producers pre-combine each partition before the
all-to-all (the reference's combiner machinery, K9-K11), and the consumer
hash-aggregates the received batches — no sort needed for a reduce.
"""

from __future__ import annotations

from typing import Callable, Sequence, Union

from ..sliceio import IterReader, Reader
from .aggregate import Aggregation, make_aggregator
from .slice_base import Dep, Name, Slice, TaskContext


class Reduce(Slice):
    """Combine values per key (reduce.go:42-78).  ``fn`` is one combine
    op (or a sequence, one per value column): 'sum'|'min'|'max'|'prod',
    operator.add/mul/min/max, or an arbitrary ``f(v, v) -> v`` (host
    path)."""

    def __init__(self, dep: Slice, fn: Union[str, Callable, Sequence]):
        schema = dep.schema
        if schema.prefix < 1:
            raise TypeError("Reduce requires a key prefix")
        nvals = schema.num_columns - schema.prefix
        if isinstance(fn, (list, tuple)):
            aggs = list(fn)
        else:
            aggs = [fn] * nvals
        if len(aggs) != nvals:
            raise TypeError(
                f"Reduce: {len(aggs)} aggregations for {nvals} value cols")
        agg = Aggregation(aggs)
        super().__init__(
            schema, dep.num_shards,
            deps=[Dep(dep, shuffle=True, expand=False)],
            name=Name("reduce"), combiner=agg)

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        schema, agg = self.schema, self.combiner

        def gen():
            aggregator = make_aggregator(schema, agg, ctx.device)
            for f in src:
                aggregator.add(f)
            yield from aggregator.result_frames(ctx.chunk)
        return IterReader(gen())


class Fold(Slice):
    """Per-shard keyed fold: fn(acc, *values) -> acc (slice.go:870-955).
    Arbitrary accumulators run on the host dict path; use Reduce for
    device-native builtin combines."""

    def __init__(self, dep: Slice, fn: Callable, out_schema=None):
        schema = dep.schema
        if schema.prefix < 1:
            raise TypeError("Fold requires a key prefix")
        self.fn = fn
        from .elementwise import schema_of
        if out_schema is None:
            # acc defaults to the value column tuple type
            self.out_schema_ = schema
        else:
            from ..schema import Schema
            key_dts = schema.dtypes[: schema.prefix]
            out = out_schema if isinstance(out_schema, Schema) else \
                schema_of(*out_schema)
            self.out_schema_ = Schema(tuple(key_dts) + tuple(out.dtypes),
                                      schema.prefix)
        super().__init__(self.out_schema_, dep.num_shards,
                         deps=[Dep(dep, shuffle=True)], name=Name("fold"))

    def reader(self, shard, dep_readers, ctx: TaskContext) -> Reader:
        src = dep_readers[0]
        fn = self.fn
        schema = self.schema
        nkey = schema.prefix
        if ctx.device != "cpu":
            import warnings
            warnings.warn(
                "Fold runs its accumulator as a host row loop (rows "
                "round-trip off the GPU); use Reduce with a builtin "
                "combine ('sum'/'min'/'max'/'prod') for the device "
                "path", RuntimeWarning, stacklevel=2)

        def gen():
            state = {}
            for f in src:
                lists = f.column_lists()
                keys = list(zip(*lists[:nkey])) if nkey > 1 else lists[0]
                vals = lists[nkey:]
                for i, k in enumerate(keys):
                    row = tuple(v[i] for v in vals)
                    acc = state.get(k)
                    state[k] = fn(acc, *row)
            if not state:
                return
            from ..frame import Frame
            items = list(state.items())
            for off in range(0, len(items), ctx.chunk):
                part = items[off:off + ctx.chunk]
                if nkey > 1:
                    key_cols = [list(c) for c in zip(*[k for k, _ in part])]
                else:
                    key_cols = [[k for k, _ in part]]
                accs = [v for _, v in part]
                if accs and isinstance(accs[0], tuple):
                    val_cols = [list(c) for c in zip(*accs)]
                else:
                    val_cols = [accs]
                yield Frame.from_lists(key_cols + val_cols, schema=schema)
        return IterReader(gen())
