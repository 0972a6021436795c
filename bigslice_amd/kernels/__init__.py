"""Device kernel dispatch.

Loads the in-tree HIP extension (bigslice_amd._C, built by setup.py for
gfx950).  Policy: on a GPU host the extension MUST be present — device ops
raise rather than silently falling back to eager torch, so a missing build
is loud.  On CPU-only hosts everything falls back to torch/numpy reference
implementations (used by tests as numerics oracles).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch

_C = None
_load_error: Optional[BaseException] = None

try:
    from bigslice_amd import _C as _C  # built by setup.py build_ext --inplace
except Exception as e:  # pragma: no cover
    _load_error = e


ALLOW_FALLBACK = os.environ.get("BIGSLICE_ALLOW_EAGER_FALLBACK", "0") == "1"


def have_extension() -> bool:
    return _C is not None


def _require(feature: str):
    if _C is None:
        raise RuntimeError(
            f"bigslice_amd._C extension not available for {feature} on a "
            f"GPU tensor (build with `python setup.py build_ext --inplace`); "
            f"import error: {_load_error!r}")


# -- hashing --------------------------------------------------------------

def hash_columns_device(cols: List[torch.Tensor], seed: int) -> torch.Tensor:
    """murmur3-32 XOR-combined over key columns; returns uint32 tensor."""
    if _C is None and ALLOW_FALLBACK:
        from .. import hashing
        host = [c.cpu() for c in cols]
        return hashing.hash_columns(host, seed).to(cols[0].device)
    _require("hash_columns")
    return _C.hash_columns(cols, seed)


# -- partition (K4) -------------------------------------------------------

def partition_supported(frame) -> bool:
    if _C is None:
        if not ALLOW_FALLBACK:
            # no silent eager fallback on a GPU box (the HIP path must
            # be the one that runs)
            _require("partition_frame")
        return False
    return all(isinstance(c, torch.Tensor) for c in frame.columns)


def partition_frame(frame, num_partitions: int, partitioner):
    """Fused hash+histogram+scatter: returns per-partition sub-frames
    (views over one reordered buffer)."""
    from ..frame import Frame
    if partitioner is not None:
        p = partitioner(frame, num_partitions).to(torch.int32)
        reordered_cols, counts = _C.scatter_by_partition(
            list(frame.columns), p, num_partitions)
    else:
        key_cols = list(frame.columns[: frame.prefix])
        reordered_cols, counts = _C.hash_partition(
            list(frame.columns), key_cols, num_partitions, 0)
    # partitioning preserves row identity: a unique-keyed (combined)
    # input stays unique per partition
    sorted_f = Frame(reordered_cols, frame.prefix,
                     combined_id=frame.combined_id)
    out = []
    off = 0
    for c in counts.tolist():
        out.append(sorted_f.slice(off, off + c) if c else None)
        off += c
    return out


# -- group-by aggregate (K9) ----------------------------------------------

_AGG_CODES = {"sum": 0, "min": 1, "max": 2, "prod": 3}

_GB_KEY_DTYPES = (torch.int64,)
_GB_VAL_DTYPES = (torch.int64, torch.int32, torch.float32, torch.float64)

GB_SENTINEL = -(1 << 63)
MAX_PROBES = 128


def groupby_supported(keys: torch.Tensor, vals: List[torch.Tensor],
                      aggs: List[str]) -> bool:
    if _C is None:
        if keys.is_cuda and not ALLOW_FALLBACK:
            _require("groupby")
        return False
    if keys.dtype not in _GB_KEY_DTYPES:
        return False
    if not all(v.dtype in _GB_VAL_DTYPES for v in vals):
        return False
    return all(a in _AGG_CODES for a in aggs)


def _next_pow2(n: int) -> int:
    c = 1024
    while c < n:
        c <<= 1
    return c


class GroupTable:
    """Streaming device hash-aggregate (K9): an open-addressing table in
    HBM that multiple batches insert into; finish() compacts used slots.
    On probe-chain overflow the table doubles and re-inserts all retained
    batches (the reference combiner's grow-x2, exec/combiner.go:47)."""

    def __init__(self, val_dtypes, aggs: List[str], device,
                 cap_hint: int = None):
        self.aggs = aggs
        self.codes = [_AGG_CODES[a] for a in aggs]
        self.device = device
        self.val_dtypes = list(val_dtypes)
        self.cap = None
        self.cap_hint = cap_hint
        self.batches: List = []  # retained for overflow re-insert
        self._deferred: List = []  # awaiting the combine-mode decision
        self.rows = 0

    @property
    def _sum_i64(self) -> bool:
        return (len(self.val_dtypes) == 1
                and self.val_dtypes[0] == torch.int64
                and self.aggs == ["sum"])

    @property
    def _packed(self) -> bool:
        """Packed key+sum slots (A/B'd neutral-to-slightly-worse than
        split arrays; off by default, kept for experiments)."""
        import os
        return (self._sum_i64
                and os.environ.get("BIGSLICE_GB_PACKED", "0") == "1")

    @property
    def _lds(self) -> bool:
        """Two-level LDS pre-aggregation (K10): absorbs hot keys on-CU.
        Default on for the int64-sum shape."""
        import os
        return (self._sum_i64 and not self._packed
                and os.environ.get("BIGSLICE_GB_LDS", "1") == "1")

    def _alloc(self, cap: int):
        self.cap = cap
        dev = self.device
        if self._packed:
            like = torch.empty(0, dtype=torch.int64, device=dev)
            self.table = _C.alloc_packed_table(cap, like)
            self.flags = torch.zeros(2, dtype=torch.int32, device=dev)
            return
        self.tkeys = torch.full((cap + 1,), GB_SENTINEL,
                                dtype=torch.int64, device=dev)
        self.tabs = [
            _C.agg_identity(torch.empty(0, dtype=dt, device=dev), code,
                            cap + 1)
            for dt, code in zip(self.val_dtypes, self.codes)]
        self.flags = torch.zeros(2, dtype=torch.int32, device=dev)

    @property
    def _sort_combinable(self) -> bool:
        """Shapes the sort+reduce-by-key combine handles: sum/min/max
        over numeric columns.  The segmented reduction is DETERMINISTIC
        (fixed tree order), so float sums through this path are
        run-to-run reproducible — unlike the hash table's atomic adds.
        Hash vs sort is decided from the sampled cardinality;
        BIGSLICE_GB_COMBINE forces {hash,sort}."""
        return (_C is not None
                and str(self.device).startswith("cuda")
                and all(a in ("sum", "min", "max") for a in self.aggs)
                and all(dt in (torch.int64, torch.int32, torch.float32,
                               torch.float64)
                        for dt in self.val_dtypes)
                and os.environ.get("BIGSLICE_GB_COMBINE", "auto")
                != "hash")

    # Cardinality sample: sorted 512k-row prefix of the first large
    # batch.  Launched WITHOUT a host sync (the device scalar is read
    # at the next insert or at finish, when the count is long done), so
    # the streaming hot path pays only ~0.1 ms of overlapped GPU time.
    _CARD_SAMPLE = int(os.environ.get("BIGSLICE_GB_SAMPLE_ROWS",
                                      str(1 << 19)))

    def _start_sample(self, keys: torch.Tensor) -> None:
        forced = os.environ.get("BIGSLICE_GB_MODE")
        if forced in ("lds", "global"):
            self._sample_forced = forced
            return
        if os.environ.get("BIGSLICE_GB_COMBINE") == "sort":
            self._sample_forced = "sort"
            return
        self._sample_forced = None
        prefix = min(keys.shape[0], self._CARD_SAMPLE)
        # probe=False: the probe's count readback would sync the
        # stream; the sample must stay fire-and-forget
        sk = _C.radix_sort_keys(keys[:prefix].contiguous(),
                                probe=False)
        self._sample_n = prefix
        self._sample_batch_n = keys.shape[0]
        self._sample_ne = (sk[1:] != sk[:-1]).sum()  # device scalar

    def _read_sample(self) -> str:
        """Combine strategy from the sample.  Three regimes (measured
        at 125M rows, profiles/sortcombine_crossover_r2.txt):
        * few keys (d <= s/32 ~ 16k): LDS-tier hash, 2.7 ms at 1k;
        * contended hash (rows/key >~ 500): the memory-side atomic
          unit serializes same-slot adds — insert 10.3 ms vs
          sort+segment-reduce 3.9 ms at 64k keys x 1900 rows/key;
        * wide keys (sample ~saturated, K >~ 2M): sort; a right-sized
          insert is 6.2 ms at 10M keys but the sort path stays ~5.1.
        In between (e.g. the north-star consumer shape, ~125 rows/key)
        the streaming insert wins IN CONTEXT (it overlaps with reads;
        the sort path bunches all work at finish): stay "global"."""
        if getattr(self, "_sample_forced", None):
            return self._sample_forced
        distinct = 1 + int(self._sample_ne.item())
        s = self._sample_n
        dbg = os.environ.get("BIGSLICE_GB_DEBUG")
        # Invert d = K(1-e^(-s/K)) to estimate the key-space size and
        # presize the table: the 10M-key hash insert measured 6.2 ms
        # properly sized vs 22 ms through the overflow-regrow grind.
        if distinct < s:
            import math
            K = float(distinct)
            for _ in range(20):
                K = distinct / (1.0 - math.exp(-s / K))
            if self.cap_hint is None:
                self.cap_hint = _next_pow2(
                    min(max(int(4 * K), 1024), 1 << 30))
        if distinct * 32 <= s:
            mode = "lds"
        elif distinct * 100 > s * 88:
            mode = "sort"  # sample ~saturated: K at least ~2M
        else:
            K = max(K, 1.0)
            rows_per_key = self._sample_batch_n / K
            mode = "sort" if rows_per_key > 500 else "global"
        if dbg:
            import sys
            print(f"[gb] sample={s} distinct={distinct} "
                  f"cap_hint={self.cap_hint} mode={mode}",
                  file=sys.stderr, flush=True)
        return mode

    def insert(self, keys: torch.Tensor, vals: List[torch.Tensor]):
        n = keys.shape[0]
        if n == 0:
            return
        self.rows += n
        if not self._sort_combinable:
            self._insert_now(keys, vals, None)
            return
        mode = getattr(self, "_mode", None)
        if mode is None:
            if n >= self._SMALL_BATCH:
                # First large batch: kick the async cardinality sample
                # and defer; the decision is read at the NEXT insert
                # (or at finish), when the 512k-row sample compute is
                # long finished — the streaming path never stalls on
                # it.  (Provisionally hash-inserting this batch while
                # the sample flies was MEASURED: no gain at 1M keys —
                # the probe-free sample already streams — and 2x worse
                # at 10M keys, where the un-presized provisional table
                # hits the overflow-regrow grind.  Deferral stays.)
                self._start_sample(keys)
                self._mode = "pending"
            self._deferred.append((keys, vals))
            return
        if mode == "pending":
            mode = self._mode = self._read_sample()
        if mode == "sort":
            self._deferred.append((keys, vals))
            return
        if self._deferred:  # flush pre-decision small batches
            pending, self._deferred = self._deferred, []
            for k, v in pending:
                self._insert_now(k, v, mode)
        self._insert_now(keys, vals, mode)

    def _insert_now(self, keys: torch.Tensor, vals: List[torch.Tensor],
                    mode: Optional[str]):
        """Allocate on demand and run the insert kernel for the given
        decided mode (None = legacy per-batch adaptive dispatch)."""
        if self.cap is None:
            from .. import config
            hint = self.cap_hint or min(
                2 * keys.shape[0],
                getattr(config, "GROUPBY_INITIAL_CAP", 1 << 23))
            self._alloc(_next_pow2(hint))
        self.batches.append((keys, vals))
        if self._packed:
            _C.groupby_insert_packed(keys, vals[0], self.table,
                                     self.flags, MAX_PROBES)
        elif self._lds and self._sum_i64 and mode == "lds":
            blocks = int(os.environ.get("BIGSLICE_GB_LDS_BLOCKS",
                                        "4096"))
            dummy = torch.zeros(1, dtype=torch.int32, device=self.device)
            _C.groupby_insert_lds(keys, vals[0], self.tkeys,
                                  self.tabs[0], self.flags, MAX_PROBES,
                                  1, blocks, dummy)
        elif self._lds and mode is None:
            self._insert_adaptive(keys, vals)
        elif (self._sum_i64 and mode == "global"
              and os.environ.get("BIGSLICE_GB_GLOBAL_MLP", "0") == "1"):
            _C.groupby_insert_mlp(keys, vals[0], self.tkeys,
                                  self.tabs[0], self.flags, MAX_PROBES)
        else:
            _C.groupby_insert(keys, list(vals), self.codes, self.tkeys,
                              self.tabs, self.flags, MAX_PROBES)

    # Host-side two-level mode decision: sample a prefix of the first
    # LARGE batch through the LDS kernel (which reports how many rows
    # its LDS tier absorbed) and commit to LDS mode on high duplicate
    # rates (hot keys), else the plain grid-stride kernel.  Small
    # batches (e.g. pre-combined shuffle buckets on the consumer side)
    # skip the sampling sync entirely and go straight to the global
    # kernel — their cost is negligible either way.
    _SAMPLE_ROWS = 1 << 22
    _SMALL_BATCH = 1 << 21

    def _insert_adaptive(self, keys, vals):
        mode = getattr(self, "_mode", None)
        if mode is None:
            forced = os.environ.get("BIGSLICE_GB_MODE", "auto")
            n = keys.shape[0]
            if forced in ("lds", "global"):
                mode = self._mode = forced
            elif n < self._SMALL_BATCH:
                _C.groupby_insert(keys, list(vals), self.codes,
                                  self.tkeys, self.tabs, self.flags,
                                  MAX_PROBES)
                return
            else:
                prefix = min(n, self._SAMPLE_ROWS)
                hits = torch.zeros(1, dtype=torch.int32,
                                   device=self.device)
                # few enough blocks that every thread runs the full
                # 16-iteration sample phase
                blocks = max(prefix // (256 * 16), 64)
                _C.groupby_insert_lds(keys[:prefix], vals[0][:prefix],
                                      self.tkeys, self.tabs[0],
                                      self.flags, MAX_PROBES, -1,
                                      blocks, hits)
                sampled = min(prefix, blocks * 256 * 16)
                frac_ok = int(hits.item()) * 8 >= sampled  # one sync
                mode = self._mode = "lds" if frac_ok else "global"
                keys = keys[prefix:]
                vals = [vals[0][prefix:]]
                if keys.shape[0] == 0:
                    return
        if mode == "lds":
            empty = torch.empty(0, dtype=torch.int32, device=self.device)
            # hot-key regime: fewer blocks bound the per-launch LDS
            # flush volume (blocks x slots contended adds)
            blocks = int(os.environ.get("BIGSLICE_GB_LDS_BLOCKS",
                                        "4096"))
            _C.groupby_insert_lds(keys, vals[0], self.tkeys,
                                  self.tabs[0], self.flags, MAX_PROBES,
                                  1, blocks, empty)
        else:
            _C.groupby_insert(keys, list(vals), self.codes, self.tkeys,
                              self.tabs, self.flags, MAX_PROBES)

    def _finish_deferred(self, mode: Optional[str]):
        """Combine the deferred batches.  mode "sort": sort+segment-
        reduce them; when a provisional table exists (the first large
        batch was hash-inserted before the decision), the combined
        uniques merge into it and compaction follows.  mode None (all
        batches were small): decide from a sample now.  Returns
        (keys, vals), or None when the result is in the table."""
        from ..utils import stats
        batches, self._deferred = self._deferred, []
        keys = (batches[0][0] if len(batches) == 1
                else torch.cat([k for k, _ in batches]))
        if mode is None:
            self._start_sample(keys)
            mode = self._mode = self._read_sample()
            if mode != "sort":
                for k, v in batches:
                    self._insert_now(k, v, mode)
                return None
        ncols = len(self.val_dtypes)
        if ncols == 1 and self.val_dtypes[0].itemsize == 8:
            vals = (batches[0][1][0] if len(batches) == 1
                    else torch.cat([v[0] for _, v in batches]))
            ks, vs = _C.radix_sort_kv(keys.contiguous(),
                                      vals.contiguous())
            vcols = [vs]
        else:
            perm = _C.radix_argsort(keys.contiguous())
            ks = keys[perm]
            vcols = []
            for c in range(ncols):
                col = (batches[0][1][c] if len(batches) == 1
                       else torch.cat([v[c] for _, v in batches]))
                vcols.append(col[perm])
        # K16: segmented reduction per value column.  Fast path for
        # int64 columns: boundaries come from ONE k_runs_sorted pass
        # (reused across columns) and k_segreduce walks each run with
        # one thread — exact for ints in any order.  Guards: skewed
        # runs (max length > 4096, one thread would serialize) and
        # float sums (need reduce_by_key's fixed reduction tree for
        # run-to-run determinism) take the rocPRIM path.
        uk = None
        outs = []
        runs = None
        if all(dt == torch.int64 for dt in self.val_dtypes):
            nrows = ks.shape[0]
            uq, starts, cnt = _C.runs_sorted(ks.contiguous())
            guard = _C.runs_guard(starts, cnt, nrows).cpu()  # one sync
            m, maxlen = int(guard[0]), int(guard[1])
            if m == 0 or maxlen <= 4096:
                uk = uq[:m]
                runs = starts[:m]
        if runs is not None:
            for c, v in enumerate(vcols):
                outs.append(_C.segment_reduce_runs(
                    v.contiguous(), runs, nrows, self.codes[c]))
        else:
            uk = None
            for c, v in enumerate(vcols):
                uq, aggs, cnt = _C.segment_reduce_sorted(
                    ks, v.contiguous(), self.codes[c])
                if uk is None:
                    m = int(cnt.item())
                    uk = uq[:m]
                outs.append(aggs[:m])
        if self.cap is not None:
            # merge with the provisionally hash-inserted first batch
            self._insert_now(uk, outs, "global")
            return None
        stats.DEFAULT.add("combiner/keys", int(uk.shape[0]))
        return uk, outs

    def finish(self):
        """Returns (keys, [vals]) of the aggregated groups."""
        from ..utils import stats
        stats.DEFAULT.add("combiner/records", self.rows)
        mode = getattr(self, "_mode", None)
        if mode == "pending":
            mode = self._mode = self._read_sample()
        if self._deferred:
            if mode in ("sort", None):
                done = self._finish_deferred(mode)
                if done is not None:
                    return done
            else:
                pending, self._deferred = self._deferred, []
                for k, v in pending:
                    self._insert_now(k, v, mode)
        if self.cap is None:
            empty = torch.empty(0, dtype=torch.int64, device=self.device)
            return empty, [torch.empty(0, dtype=dt, device=self.device)
                           for dt in self.val_dtypes]
        while True:
            cursor = torch.zeros(1, dtype=torch.int64, device=self.device)
            if self._packed:
                outs = _C.groupby_compact_packed(self.table, cursor)
            else:
                outs = _C.groupby_compact(self.tkeys, self.tabs, cursor)
            host = torch.cat([cursor,
                              self.flags.to(torch.int64)]).cpu()
            nkeys, sentinel_seen, overflow = (int(host[0]), int(host[1]),
                                              int(host[2]))
            if overflow:
                cap = self.cap
                while cap < 4 * self.rows and cap < (1 << 30):
                    cap <<= 1
                if cap == self.cap:
                    cap <<= 1
                if cap > (1 << 31):
                    raise RuntimeError("groupby table overflow")
                batches = self.batches
                self._alloc(cap)
                self.batches = []
                m = getattr(self, "_mode", None)
                if m not in ("lds", "global"):
                    m = "global" if self._sort_combinable else None
                for keys, vals in batches:
                    self._insert_now(keys, vals, m)
                continue
            keys = outs[0][:nkeys]
            vals = [o[:nkeys] for o in outs[1:]]
            if sentinel_seen:
                keys = torch.cat([keys, torch.full(
                    (1,), GB_SENTINEL, dtype=torch.int64,
                    device=self.device)])
                if self._packed:
                    c = self.cap
                    vals = [torch.cat([vals[0],
                                       self.table[2 * c + 1:2 * c + 2]])]
                else:
                    vals = [torch.cat([v, t[self.cap:self.cap + 1]])
                            for v, t in zip(vals, self.tabs)]
            self.batches = []
            from ..utils import stats
            stats.DEFAULT.add("combiner/keys", int(keys.shape[0]))
            return keys, vals


def groupby(keys: torch.Tensor, vals: List[torch.Tensor],
            aggs: List[str]):
    """One-shot hash-aggregate: returns (unique_keys, [combined vals])."""
    t = GroupTable([v.dtype for v in vals], aggs, keys.device)
    t.insert(keys, list(vals))
    return t.finish()


# -- sort (K6) -------------------------------------------------------------

def sort_pairs_supported(keys: torch.Tensor) -> bool:
    return _C is not None and keys.dtype in (torch.int64, torch.int32,
                                             torch.float32, torch.float64)


def radix_argsort(keys: torch.Tensor) -> torch.Tensor:
    """Device radix sort; returns the sorting permutation (int64)."""
    if _C is None:
        if keys.is_cuda and not ALLOW_FALLBACK:
            _require("radix_argsort")
        return torch.argsort(keys, stable=True)
    return _C.radix_argsort(keys)


def radix_sort_kv(keys: torch.Tensor, val: torch.Tensor):
    """Direct (key, value) radix sort for one 8-byte value column."""
    if _C is None:
        if keys.is_cuda and not ALLOW_FALLBACK:
            _require("radix_sort_kv")
        perm = torch.argsort(keys, stable=True)
        return keys[perm], val[perm]
    v64 = val.view(torch.int64) if val.dtype != torch.int64 else val
    sk, sv = _C.radix_sort_kv(keys, v64)
    return sk, (sv.view(val.dtype) if val.dtype != torch.int64 else sv)


def radix_sort_keys(keys: torch.Tensor) -> torch.Tensor:
    """Device radix key-only sort (no permutation)."""
    if _C is None:
        if keys.is_cuda and not ALLOW_FALLBACK:
            _require("radix_sort_keys")
        return torch.sort(keys).values
    return _C.radix_sort_keys(keys)
