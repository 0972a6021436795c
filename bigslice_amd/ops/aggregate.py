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

    def key(self):
        """Structural identity for compiler memoization: equal specs
        share producer tasks, different specs must not."""
        return tuple(a if isinstance(a, str) else id(a)
                     for a in self.aggs)

    @property
    def name(self) -> str:
        """Stable human label (golden graphs, traces)."""
        return "+".join(
            a if isinstance(a, str) else getattr(a, "__name__", "fn")
            for a in self.aggs)

    def __repr__(self):
        return f"Aggregation({self.aggs})"


def make_aggregator(schema: Schema, agg: Aggregation, device: str,
                    seed: int = config.COMBINER_HASH_SEED):
    """Choose the right aggregator implementation for a schema:
    tensor path for any all-numeric schema (single- or multi-column
    keys), device dictionary path for a single BYTES key, host dict
    path for object keys / arbitrary combine fns."""
    from ..schema import is_bytes
    numeric = (not any(is_object(d) or is_bytes(d)
                       for d in schema.dtypes))
    if agg.all_builtin and numeric and schema.prefix >= 1:
        return TensorAggregator(schema, agg, device)
    if (agg.all_builtin and schema.prefix == 1
            and is_bytes(schema.dtypes[0])
            and not any(is_object(d) or is_bytes(d)
                        for d in schema.dtypes[1:])):
        return BytesKeyAggregator(schema, agg, device)
    return DictAggregator(schema, agg)


class BytesKeyAggregator:
    """Keyed reduce over a device varlen BYTES key: rows group by the
    key's 64-bit dictionary id through the numeric TensorAggregator
    (HIP GroupTable on GPU), while a per-aggregator dictionary maps
    each id back to one exemplar byte row for the output.  This is the
    device-resident string group-by of the reference's string ops
    (frame/ops_builtin.go:143-164); id collisions are 2^-64 per pair,
    the accepted dictionary-encoding tradeoff (see frame.BytesColumn).
    """

    def __init__(self, schema: Schema, agg: Aggregation, device: str):
        from ..schema import Schema as S
        self.schema = schema
        inner_schema = S((torch.int64,) + tuple(schema.dtypes[1:]), 1)
        self.inner = TensorAggregator(inner_schema, agg, device)
        self.device = device
        self.dict_ids: Optional[torch.Tensor] = None  # sorted
        self.dict_keys = None  # BytesColumn exemplars, id-sorted

    def add(self, frame: Frame) -> None:
        if len(frame) == 0:
            return
        key_col = frame.columns[0]
        ids = key_col.ids64()
        uniq = torch.unique(ids)
        if self.dict_ids is not None:
            new = uniq[~torch.isin(uniq, self.dict_ids)]
        else:
            new = uniq
        if new.numel():
            # exemplar row per new id: first occurrence in this frame
            sidx = torch.argsort(ids, stable=True)
            pos = torch.searchsorted(ids[sidx], new)
            exemplars = key_col.select(sidx[pos])
            if self.dict_ids is None:
                merged_ids, merged_keys = new, exemplars
            else:
                from ..frame import BytesColumn
                cat_ids = torch.cat([self.dict_ids, new])
                order = torch.argsort(cat_ids)
                merged_ids = cat_ids[order]
                both = Frame.concat([
                    Frame([self.dict_keys], 1),
                    Frame([exemplars], 1)]).columns[0]
                merged_keys = both.select(order)
            self.dict_ids = merged_ids
            self.dict_keys = merged_keys
        self.inner.add(Frame([ids] + list(frame.columns[1:]), 1))

    def result_frames(self, chunk: int):
        for f in self.inner.result_frames(chunk):
            pos = torch.searchsorted(
                self.dict_ids, f.columns[0].to(self.dict_ids.device))
            keys = self.dict_keys.select(pos)
            yield Frame([keys] + list(f.columns[1:]), 1,
                        combined_id=f.combined_id)


class TensorAggregator:
    """Device/CPU aggregation for numeric key column(s).

    Single int64 keys on GPU stream through the HIP GroupTable;
    everything else combines via sort-based segment reduce (multi-column
    keys use a lexicographic sort + boundary detection).  State is
    re-compacted on flush, bounding memory at O(distinct keys).
    """

    def __init__(self, schema: Schema, agg: Aggregation, device: str):
        self.schema = schema
        self.nkey = schema.prefix
        self.agg = agg
        self.device = device
        self.keys: Optional[List[torch.Tensor]] = None
        self.vals: List[torch.Tensor] = []
        self._pending: List[Frame] = []
        self._pending_rows = 0
        # Pre-combined passthrough: while every added frame carries the
        # SAME combiner instance id, its keys are globally unique and
        # re-aggregation is the identity (the machine-combiners
        # single-stream case, e.g. one GPU's shared combine feeding its
        # own consumer).  A frame from a different (or no) instance
        # demotes everything to normal aggregation.
        self._pt_frames: Optional[List[Frame]] = []
        self._pt_id = None
        # Device fast path: stream batches straight into the HIP
        # GroupTable (no host sync until finish).
        self._table = None
        if device.startswith("cuda") and self.nkey == 1:
            from .. import kernels
            val_dts = schema.dtypes[schema.prefix:]
            shape_ok = (all(dt in kernels._GB_VAL_DTYPES
                            for dt in val_dts)
                        and schema.dtypes[0] in kernels._GB_KEY_DTYPES
                        and all(isinstance(a, str) for a in agg.aggs))
            if shape_ok and not kernels.have_extension() and \
                    not kernels.ALLOW_FALLBACK:
                kernels._require("groupby")  # no silent eager fallback
            if shape_ok and kernels.have_extension():
                self._table = kernels.GroupTable(val_dts, agg.aggs, device)

    # small device batches are concatenated before insert: one launch
    # instead of many (consumer-side shuffle buckets are ~100k rows)
    _SMALL_ROWS = 1 << 20

    def add(self, frame: Frame) -> None:
        if len(frame) == 0:
            return
        if frame.device != self.device:
            frame = frame.to(self.device)
        if self._pt_frames is not None:
            cid = frame.combined_id
            if cid is not None and (self._pt_id is None
                                    or cid == self._pt_id):
                self._pt_id = cid
                self._pt_frames.append(frame)
                return
            # demote: replay buffered frames through normal aggregation
            buffered, self._pt_frames = self._pt_frames, None
            for f in buffered:
                self._add_normal(f)
        self._add_normal(frame)

    def _add_normal(self, frame: Frame) -> None:
        if self._table is not None:
            if len(frame) < self._SMALL_ROWS:
                self._pending.append(frame)
                self._pending_rows += len(frame)
                if self._pending_rows >= 4 * self._SMALL_ROWS:
                    self._flush_table_pending()
                return
            self._table.insert(frame.columns[0].contiguous(),
                               [c.contiguous() for c in frame.columns[1:]])
            return
        self._pending.append(frame)
        self._pending_rows += len(frame)
        if self._pending_rows >= config.COMBINER_TARGET_KEYS:
            self._flush()

    def _flush_table_pending(self) -> None:
        if not self._pending:
            return
        f = Frame.concat(self._pending)
        self._pending.clear()
        self._pending_rows = 0
        self._table.insert(f.columns[0].contiguous(),
                           [c.contiguous() for c in f.columns[1:]])

    def _flush(self) -> None:
        if not self._pending:
            return
        f = Frame.concat(self._pending)
        self._pending.clear()
        self._pending_rows = 0
        keys = [c for c in f.columns[: self.nkey]]
        vals = list(f.columns[self.nkey:])
        if self.keys is not None:
            keys = [torch.cat([sk, k]) for sk, k in zip(self.keys, keys)]
            vals = [torch.cat([sv, v]) for sv, v in zip(self.vals, vals)]
        self.keys, self.vals = _combine_once(keys, vals, self.agg)

    def result_frames(self, chunk: int):
        if self._pt_frames is not None:
            # single pre-combined stream: identity (keys already unique)
            for f in self._pt_frames:
                yield f
            return
        if self._table is not None:
            self._flush_table_pending()
            keys, vals = self._table.finish()
            if keys.shape[0]:
                self.keys, self.vals = [keys], vals
        else:
            self._flush()
        if self.keys is None:
            return
        import uuid
        cid = uuid.uuid4().int & ((1 << 63) - 1)
        n = self.keys[0].shape[0]
        for off in range(0, n, chunk):
            stop = min(off + chunk, n)
            yield Frame([k[off:stop] for k in self.keys] +
                        [v[off:stop] for v in self.vals],
                        self.schema.prefix, combined_id=cid)


def _combine_once(keys: List[torch.Tensor], vals: List[torch.Tensor],
                  agg: Aggregation):
    """One compaction: group rows by key column(s), combining values.

    Single-key GPU batches use the HIP hash-aggregate kernel when
    supported; otherwise a sort-based segment reduce (multi-column keys:
    lexicographic stable sort + boundary detection), which runs on both
    CPU and GPU.
    """
    if len(keys) == 1:
        k = keys[0]
        if k.is_cuda:
            from .. import kernels
            if kernels.groupby_supported(k, vals, agg.aggs):
                uk, ov = kernels.groupby(k, vals, agg.aggs)
                return [uk], ov
        uk, inv = torch.unique(k, return_inverse=True)
        nseg = uk.shape[0]
        uks = [uk]
    else:
        n = keys[0].shape[0]
        perm = torch.arange(n, device=keys[0].device)
        for kc in reversed(keys):
            _, o = torch.sort(kc[perm], stable=True)
            perm = perm[o]
        skeys = [kc[perm] for kc in keys]
        boundary = torch.zeros(n, dtype=torch.bool,
                               device=keys[0].device)
        boundary[0] = True
        for kc in skeys:
            boundary[1:] |= kc[1:] != kc[:-1]
        inv_sorted = boundary.cumsum(0) - 1
        nseg = int(inv_sorted[-1].item()) + 1 if n else 0
        uks = [kc[boundary] for kc in skeys]
        # map back to original row order for scatter
        inv = torch.empty(n, dtype=torch.int64, device=keys[0].device)
        inv[perm] = inv_sorted
        vals = list(vals)
    out_vals = []
    for v, a in zip(vals, agg.aggs):
        init = _scatter_init(a, v.dtype, nseg, v.device)
        out = init.scatter_reduce_(0, inv, v, reduce=_SCATTER_OP[a],
                                   include_self=False)
        out_vals.append(out)
    return uks, out_vals


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
        if len(vals) == 1 and aggs == ["sum"]:
            vcol = vals[0]
            if isinstance(vcol, list) and vcol.count(1) == len(vcol):
                # counting shape (all-ones values): C-speed Counter
                from collections import Counter
                for k, n in Counter(keys).items():
                    cur = state.get(k)
                    if cur is None:
                        state[k] = [n]
                    else:
                        cur[0] += n
            else:
                for k, v in zip(keys, vcol):
                    cur = state.get(k)
                    if cur is None:
                        state[k] = [v]
                    else:
                        cur[0] += v
            return
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
