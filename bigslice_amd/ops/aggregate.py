"""Hash aggregation: the engine's group-by-combine core.

Role-parity with the reference's combiner machinery (exec/combiner.go: the
open-addressing combiningFrame K9, two-level combine K10, spilling combiner
K11; accum.go accumulators K12).  The MI355X-native design replaces the
row-wise open-addressing table with whole-batch device aggregation:

* single numeric key column + numeric values: a HIP hash-aggregate kernel
  (csrc/groupby.hip) using an HBM open-addressing table with LDS
  pre-aggregation per workgroup; torch scatter_reduce is the portable
  fallback (and the CPU path).
* arbitrary keys/values/fns: a host dict accumulator (the reference's
  accum.go path).

An Aggregation names the per-value-column combine op; builtins run on
device, arbitrary callables run on host.
"""

from __future__ import annotations

import operator
from typing import Callable, List, Optional, Sequence, Union

import torch

from .. import config
from ..frame import Frame
from ..schema import Schema, is_object

BUILTIN_AGGS = ("sum", "min", "max", "prod")

_CALLABLE_MAP = {
    operator.add: "sum",
    operator.mul: "prod",
    min: "min",
    max: "max",
}

_SCATTER_OP = {"sum": "sum", "min": "amin", "max": "amax", "prod": "prod"}


def resolve_agg(fn: Union[str, Callable]) -> Union[str, Callable]:
    """Map a user combine fn to a builtin name when possible."""
    if isinstance(fn, str):
        if fn not in BUILTIN_AGGS:
            raise ValueError(f"unknown aggregation {fn!r}")
        return fn
    return _CALLABLE_MAP.get(fn, fn)


class Aggregation:
    """Per-value-column combine spec for a keyed reduce."""

    def __init__(self, aggs: Sequence[Union[str, Callable]]):
        self.aggs = [resolve_agg(a) for a in aggs]

    @property
    def all_builtin(self) -> bool:
        return all(isinstance(a, str) for a in self.aggs)

    def __repr__(self):
        return f"Aggregation({self.aggs})"


def make_aggregator(schema: Schema, agg: Aggregation, device: str,
                    seed: int = config.COMBINER_HASH_SEED):
    """Choose the right aggregator implementation for a schema."""
    nkey = schema.prefix
    key_dts = schema.dtypes[:nkey]
    val_dts = schema.dtypes[nkey:]
    numeric = (not any(is_object(d) for d in schema.dtypes))
    if agg.all_builtin and numeric and nkey == 1:
        return TensorAggregator(schema, agg, device)
    return DictAggregator(schema, agg)


class TensorAggregator:
    """Device/CPU aggregation for a single numeric key column.

    Keeps a running (keys, values...) state; each added batch is combined
    via segment-reduce over sorted-unique keys (torch fallback), or the
    HIP hash-aggregate kernel on GPU.  State is re-compacted whenever it
    doubles, bounding memory at O(distinct keys).
    """

    def __init__(self, schema: Schema, agg: Aggregation, device: str):
        self.schema = schema
        self.agg = agg
        self.device = device
        self.keys: Optional[torch.Tensor] = None
        self.vals: List[torch.Tensor] = []
        self._pending: List[Frame] = []
        self._pending_rows = 0
        # Device fast path: stream batches straight into the HIP
        # GroupTable (no host sync until finish).
        self._table = None
        if device.startswith("cuda"):
            from .. import kernels
            val_dts = schema.dtypes[schema.prefix:]
            if kernels.have_extension() and all(
                    dt in kernels._GB_VAL_DTYPES for dt in val_dts) and \
                    schema.dtypes[0] in kernels._GB_KEY_DTYPES and \
                    all(isinstance(a, str) for a in agg.aggs):
                self._table = kernels.GroupTable(val_dts, agg.aggs, device)

    def add(self, frame: Frame) -> None:
        if len(frame) == 0:
            return
        if frame.device != self.device:
            frame = frame.to(self.device)
        if self._table is not None:
            self._table.insert(frame.columns[0].contiguous(),
                               [c.contiguous() for c in frame.columns[1:]])
            return
        self._pending.append(frame)
        self._pending_rows += len(frame)
        if self._pending_rows >= config.COMBINER_TARGET_KEYS:
            self._flush()

    def _flush(self) -> None:
        if not self._pending:
            return
        f = Frame.concat(self._pending)
        self._pending.clear()
        self._pending_rows = 0
        keys = f.columns[0]
        vals = list(f.columns[1:])
        if self.keys is not None:
            keys = torch.cat([self.keys, keys])
            vals = [torch.cat([sv, v]) for sv, v in zip(self.vals, vals)]
        self.keys, self.vals = _combine_once(keys, vals, self.agg)

    def result_frames(self, chunk: int):
        if self._table is not None:
            keys, vals = self._table.finish()
            if keys.shape[0]:
                self.keys, self.vals = keys, vals
        else:
            self._flush()
        if self.keys is None:
            return
        n = self.keys.shape[0]
        for off in range(0, n, chunk):
            stop = min(off + chunk, n)
            yield Frame([self.keys[off:stop]] +
                        [v[off:stop] for v in self.vals],
                        self.schema.prefix)

    def num_keys(self) -> int:
        if self._table is not None:
            return self._table.rows and self._table.finish()[0].shape[0]
        self._flush()
        return 0 if self.keys is None else self.keys.shape[0]


def _combine_once(keys: torch.Tensor, vals: List[torch.Tensor],
                  agg: Aggregation):
    """One compaction: group rows by key, combining values.

    GPU path uses the HIP hash-aggregate kernel when available for the
    supported dtype combos; otherwise sort-based segment reduce.
    """
    if keys.is_cuda:
        from .. import kernels
        if kernels.groupby_supported(keys, vals, agg.aggs):
            return kernels.groupby(keys, vals, agg.aggs)
    uk, inv = torch.unique(keys, return_inverse=True)
    out_vals = []
    for v, a in zip(vals, agg.aggs):
        init = _scatter_init(a, v.dtype, uk.shape[0], v.device)
        out = init.scatter_reduce_(0, inv, v, reduce=_SCATTER_OP[a],
                                   include_self=False)
        out_vals.append(out)
    return uk, out_vals


def _scatter_init(aggname: str, dtype, n: int, device):
    return torch.empty(n, dtype=dtype, device=device)


class DictAggregator:
    """Host dict aggregation for object keys and/or arbitrary combine fns
    (reference accum.go:28-186)."""

    def __init__(self, schema: Schema, agg: Aggregation):
        self.schema = schema
        self.agg = agg
        self.state = {}

    def add(self, frame: Frame) -> None:
        nkey = self.schema.prefix
        lists = frame.column_lists()
        keys = list(zip(*lists[:nkey])) if nkey > 1 else lists[0]
        vals = lists[nkey:]
        state = self.state
        aggs = self.agg.aggs
        for i, k in enumerate(keys):
            row = tuple(v[i] for v in vals)
            cur = state.get(k)
            if cur is None:
                state[k] = list(row)
            else:
                for j, a in enumerate(aggs):
                    if a == "sum":
                        cur[j] += row[j]
                    elif a == "min":
                        cur[j] = min(cur[j], row[j])
                    elif a == "max":
                        cur[j] = max(cur[j], row[j])
                    elif a == "prod":
                        cur[j] *= row[j]
                    else:
                        cur[j] = a(cur[j], row[j])

    def result_frames(self, chunk: int):
        if not self.state:
            return
        items = list(self.state.items())
        nkey = self.schema.prefix
        for off in range(0, len(items), chunk):
            part = items[off:off + chunk]
            if nkey > 1:
                key_cols = list(zip(*[k for k, _ in part]))
            else:
                key_cols = [[k for k, _ in part]]
            val_cols = list(zip(*[v for _, v in part])) if part and \
                part[0][1] else []
            cols = [list(c) for c in key_cols] + [list(c) for c in val_cols]
            yield Frame.from_lists(cols, schema=self.schema)

    def num_keys(self) -> int:
        return len(self.state)
