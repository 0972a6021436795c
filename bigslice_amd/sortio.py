"""External sort and merge machinery.

Role-parity: sortio/sort.go (SortReader: read -> sort in-core chunk ->
spill -> k-way merge; NewMergeReader heap merge) and sortio/reader.go
(Reduce: merge + adjacent-equal-key combine).

MI355X redesign: in-core runs are whole device batches sorted with the
radix-sort kernel (K6); spilled runs tier to pinned host DRAM through the
Spiller; the k-way merge is batch-granular on device: take a window from
each run, cut at the minimum window-max key, and merge the cut rows with
one device sort (no per-row heap).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from . import config
from .frame import Frame
from .sliceio import Reader, IterReader
from .sliceio.spiller import Spiller


def sort_frame(frame: Frame) -> Frame:
    """Sort one frame by its key prefix (device radix sort for single
    numeric keys, lexicographic stable sort otherwise)."""
    if len(frame) <= 1:
        return frame
    first = frame.columns[0]
    if (frame.prefix == 1 and isinstance(first, torch.Tensor)
            and first.is_cuda):
        from . import kernels
        from .frame import Frame as _F
        if kernels.sort_pairs_supported(first):
            if (frame.num_columns == 2
                    and isinstance(frame.columns[1], torch.Tensor)
                    and frame.columns[1].element_size() == 8):
                # direct (key, value) sort: no permutation/gather passes
                sk, sv = kernels.radix_sort_kv(
                    first.contiguous(), frame.columns[1].contiguous())
                return _F([sk, sv], 1, combined_id=frame.combined_id)
            perm = kernels.radix_argsort(first.contiguous())
            return frame.select(perm)
    return frame.sort_by_prefix()


class SortReader(Reader):
    """Drains a reader, producing key-sorted output: in-core runs of up
    to `run_bytes` are sorted and spilled; read() merges the runs
    (sortio/sort.go:31-76)."""

    def __init__(self, source: Reader, run_bytes: int = None,
                 device: str = "cpu", chunk: int = None,
                 host_budget: int = 64 << 30):
        self.source = source
        self.run_bytes = run_bytes or config.SORT_SPILL_TARGET_BYTES
        self.device = device
        self.chunk = chunk or config.chunk_rows(device)
        self.host_budget = host_budget
        self._merged: Optional[Reader] = None

    def _build(self) -> Reader:
        spiller = Spiller(host_budget_bytes=self.host_budget)
        pending: List[Frame] = []
        pending_bytes = 0
        runs: List[List[Frame]] = []
        first_run: List[Optional[Frame]] = [None]

        def spill_run(run: Frame):
            # spill in slices: merge readahead then holds one small
            # window per run, not the whole run (config
            # SORT_SPILL_CHUNK_BYTES)
            start = spiller.num_batches()
            rows = len(run)
            per_row = max(1, run.nbytes() // max(rows, 1))
            step = max(1, int(config.SORT_SPILL_CHUNK_BYTES // per_row))
            for off in range(0, rows, step):
                spiller.spill(run.slice(off, min(off + step, rows)))
            runs.append(list(range(start, spiller.num_batches())))

        def flush_run():
            nonlocal pending, pending_bytes
            if not pending:
                return
            run = sort_frame(Frame.concat(pending))
            pending = []
            pending_bytes = 0
            if first_run[0] is None and not runs:
                # Defer spilling the first run: if the whole input is a
                # single run it never leaves HBM (the host round trip is
                # pure overhead when the data fits).
                first_run[0] = run
                return
            if first_run[0] is not None:
                spill_run(first_run[0])
                first_run[0] = None
            spill_run(run)

        for f in self.source:
            pending.append(f)
            pending_bytes += f.nbytes()
            if pending_bytes >= self.run_bytes:
                flush_run()
        flush_run()
        if first_run[0] is not None:
            from .sliceio import FrameReader
            return FrameReader(first_run[0], self.chunk)
        if not runs:
            return IterReader(iter(()))
        readers = [_SpillRunReader(spiller, idxs, self.device)
                   for idxs in runs]
        if len(readers) == 1:
            return readers[0]
        return MergeReader(readers, chunk=self.chunk)

    def read(self) -> Optional[Frame]:
        if self._merged is None:
            self._merged = self._build()
        return self._merged.read()


class _SpillRunReader(Reader):
    def __init__(self, spiller: Spiller, batch_idxs: List[int],
                 device: str):
        from .sliceio.spiller import SpillReader
        self._inner = SpillReader(
            [spiller.batches[i] for i in batch_idxs], device)

    def read(self) -> Optional[Frame]:
        return self._inner.read()


class _RunCursor:
    """Buffered cursor over one sorted run."""

    def __init__(self, reader: Reader):
        self.reader = reader
        self.buf: Optional[Frame] = None
        self.eof = False

    def fill(self, want_rows: int) -> None:
        while not self.eof and (self.buf is None or len(self.buf) <
                                want_rows):
            f = self.reader.read()
            if f is None:
                self.eof = True
                return
            self.buf = f if self.buf is None else Frame.concat(
                [self.buf, f])

    def max_buffered_key(self):
        c = self.buf.columns[0]
        return c[-1]  # tensor scalar for tensor columns, value for lists

    def _take_n(self, n: int) -> Optional[Frame]:
        if n == 0:
            return None
        out = self.buf.slice(0, n)
        self.buf = self.buf.slice(n, len(self.buf)) \
            if n < len(self.buf) else None
        return out

    def take_upto(self, cutoff) -> Optional[Frame]:
        """Rows with key <= cutoff (buffer is sorted)."""
        c = self.buf.columns[0]
        if isinstance(c, torch.Tensor):
            n = int(torch.searchsorted(c, cutoff, right=True).item()) \
                if isinstance(cutoff, torch.Tensor) else \
                int(torch.searchsorted(
                    c, torch.tensor(cutoff, dtype=c.dtype,
                                    device=c.device), right=True).item())
        else:
            import bisect
            n = bisect.bisect_right(c, cutoff)
        return self._take_n(n)

    def take_below(self, cutoff) -> Optional[Frame]:
        """Rows with key < cutoff (buffer is sorted)."""
        c = self.buf.columns[0]
        if isinstance(c, torch.Tensor):
            n = int(torch.searchsorted(c, cutoff, right=False).item()) \
                if isinstance(cutoff, torch.Tensor) else \
                int(torch.searchsorted(
                    c, torch.tensor(cutoff, dtype=c.dtype,
                                    device=c.device),
                    right=False).item())
        else:
            import bisect
            n = bisect.bisect_left(c, cutoff)
        return self._take_n(n)


class MergeReader(Reader):
    """Batch-granular k-way merge of sorted runs (sortio/sort.go:161-222
    redesigned): each read() pulls a window per run, cuts at the minimum
    window-max key (guaranteeing global order), and sorts the union of
    the cut rows on device."""

    def __init__(self, readers: List[Reader], chunk: int = None):
        self.cursors = [_RunCursor(r) for r in readers]
        self.chunk = chunk or 1 << 20

    def read(self) -> Optional[Frame]:
        per_run = max(self.chunk // max(len(self.cursors), 1), 1024)
        live = []
        for c in self.cursors:
            c.fill(per_run)
            if c.buf is not None and len(c.buf) > 0:
                live.append(c)
        if not live:
            return None
        if len(live) == 1:
            out = live[0].buf
            live[0].buf = None
            return out
        # cutoff = min over NON-EOF runs of the last buffered first-key.
        # Rows with first-key STRICTLY BELOW the cutoff are globally
        # complete (every unbuffered row of run i has key >= run i's
        # buffered max >= cutoff); rows AT the cutoff may continue in a
        # later batch of a cutoff run (same first-key, smaller secondary
        # columns), so they stay buffered for the next window.  Only
        # when no run holds anything below the cutoff — the cutoff
        # run's buffer is one giant equal-key block — does that run
        # grow, bounded by the block, not the dataset.  (An earlier
        # version grew the minimum run on EVERY window: the minimum is
        # at its own cutoff by definition, so merges of many runs
        # buffered entire datasets and a 10B-row external sort OOMed.)
        while True:
            non_eof = [c for c in live if not c.eof]
            if not non_eof:
                parts = [c.buf for c in live]
                for c in live:
                    c.buf = None
                return sort_frame(Frame.concat(parts))
            maxes = [c.max_buffered_key() for c in non_eof]
            if isinstance(maxes[0], torch.Tensor):
                cutoff = torch.stack(list(maxes)).min()
            else:
                cutoff = min(maxes)
            parts = []
            for c in live:
                p = c.take_upto(cutoff) if c.eof \
                    else c.take_below(cutoff)
                if p is not None:
                    parts.append(p)
            if parts:
                return sort_frame(Frame.concat(parts))
            for c in non_eof:
                m = c.max_buffered_key()
                eq = bool((m == cutoff).item()) \
                    if isinstance(m, torch.Tensor) else m == cutoff
                if eq:
                    c.fill(len(c.buf) * 2)


def reduce_reader(readers: List[Reader], schema, agg,
                  device: str = "cpu", chunk: int = None) -> Reader:
    """Merge sorted per-producer streams, combining equal keys
    (sortio/reader.go:36-130).  Used by sorted-reduce consumers; the
    hash-aggregate path (ops.aggregate) is the default for Reduce."""
    merged = MergeReader(readers, chunk=chunk)

    def gen():
        from .ops.aggregate import make_aggregator
        carry: Optional[Frame] = None
        for f in merged:
            if carry is not None:
                f = Frame.concat([carry, f])
            # combine adjacent equal keys; hold back the last key group
            # (it may continue in the next batch)
            agg_ = make_aggregator(schema, agg, device)
            agg_.add(f)
            combined = Frame.concat(
                list(agg_.result_frames(1 << 62)))
            combined = sort_frame(combined)
            if len(combined) > 1:
                carry = combined.slice(len(combined) - 1, len(combined))
                yield combined.slice(0, len(combined) - 1)
            else:
                carry = combined
        if carry is not None and len(carry):
            yield carry
    return IterReader(gen())
