"""Windowed, overlapped, placement-aware shuffle exchange.

Role-parity: the reference's data plane streams shuffle data while
producers still run (consumer-pull Worker.Read, exec/bigmachine.go:822-908)
and its machine manager schedules tasks onto the least-loaded machines
(exec/slicemachine.go:444-603, :748-788).  The MI355X equivalents built
here:

* **Windowed all-to-allv with compute/comm overlap**: a phase's exchange
  is cut into windows.  As producer tasks stream output, full windows are
  launched as async RCCL all-to-allvs (transfers ride the collective
  stream over xGMI) while the host keeps launching producer compute on
  the compute stream; received windows are consumed (stored or streamed
  into consumer-side combiners) while later windows are in flight.  Step
  time approaches max(compute, comm) instead of their sum.

* **Count-driven partition placement (LPT)**: the first window's global
  per-partition row counts (from the fixed-width count plane) drive a
  deterministic longest-processing-time assignment of partitions to
  ranks, replacing static ``p % world``.  A hot partition no longer
  shares a rank with another hot partition — the load-aware scheduling
  role of the reference's machineQ/priority heaps, adapted to SPMD where
  every rank computes the identical assignment from identical counts.

Window protocol (collective, deadlock-free by construction): rounds are
lock-step.  In round k every rank contributes a (possibly empty)
per-partition count vector plus "more" and "error" flags via ONE
fixed-width all_gather — this is the whole metadata plane (no object
gathers; exactly one host readback per round).  Round k+1 happens iff
any rank's round-k flag said more; every rank evaluates that from the
same gathered tensor, so all ranks run the same number of rounds, and a
rank that fails mid-phase finishes the protocol with empty windows and
an error flag instead of abandoning its peers inside a collective.
Payload all-to-allvs launch async in round order (uniform collective
launch order on every rank).
"""

from __future__ import annotations

import os
from typing import Callable, Dict, List, Optional

import torch
import torch.distributed as dist

from ..frame import BytesColumn, Frame
from ..schema import Schema, is_bytes
from .comm import Comm


def _env_int(name: str, default: int) -> int:
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


# A window closes when its buffered payload reaches this size: large
# enough that each of the 7 xGMI peer messages amortizes link latency,
# small enough that MAX_IN_FLIGHT windows are cheap beside the data.
WINDOW_BYTES = _env_int("BIGSLICE_EXCHANGE_WINDOW_BYTES", 256 << 20)

# Producer-side partial-combine flush: in combine mode the shared
# aggregator flushes a partial result once this many rows have been
# inserted since the last flush, so high-cardinality combines still
# stream windows (the reference's spilling-combiner targetSize, K11).
PARTIAL_COMBINE_ROWS = _env_int("BIGSLICE_EXCHANGE_PARTIAL_ROWS",
                                256 << 20)

# Max unconsumed windows before the oldest is drained (bounds buffered
# send/recv memory to ~this many windows per direction).
MAX_IN_FLIGHT = _env_int("BIGSLICE_EXCHANGE_IN_FLIGHT", 2)


def lpt_assign(sizes: List[int], world: int) -> List[int]:
    """Deterministic longest-processing-time bin packing: partitions in
    size order (desc, index tie-break) go to the least-loaded rank
    (load tie-break by rank index).  Every rank computes the identical
    assignment from the identical sizes."""
    order = sorted(range(len(sizes)), key=lambda p: (-sizes[p], p))
    load = [0] * world
    nassigned = [0] * world
    assign = [0] * len(sizes)
    for p in order:
        # tie-break equal loads by partition count so zero/equal-size
        # partitions round-robin (consumer tasks still spread)
        r = min(range(world), key=lambda x: (load[x], nassigned[x], x))
        assign[p] = r
        load[r] += sizes[p]
        nassigned[r] += 1
    return assign


class _Window:
    __slots__ = ("count_work", "gathered", "counts", "bcounts",
                 "more_any", "err_any", "send_buf", "payload_works",
                 "recv_cols", "my_meta", "launched", "consumed")

    def __init__(self):
        self.count_work = None
        self.gathered = None      # list of [nparts+2] tensors (transport)
        self.counts = None        # [world, nparts] host tensor after resolve
        self.bcounts = None       # [world, nbcols, nparts] byte counts
        self.more_any = None
        self.err_any = False
        self.send_buf: Optional[Dict[int, List[Frame]]] = None
        self.payload_works = []
        self.recv_cols = []
        self.my_meta = None       # [(src, p, rows)] in recv order
        self.launched = False
        self.consumed = False


class PhaseExchange:
    """One phase's windowed exchange on one rank.

    ``add(p, frame)`` buffers producer output for partition ``p`` and
    flushes windows at the byte threshold; ``finish(err)`` flushes the
    final window and runs drain rounds until every rank is done.
    ``consume(p, frame)`` receives every bucket destined to this rank,
    called while later windows are still in flight.  ``resolve(sizes)``
    maps partitions to ranks: called first with ``None`` (return an
    existing placement for this exchange domain or None), then — if
    that returned None — with the first window's global sizes to
    compute and record one.
    """

    def __init__(self, comm: Comm, schema: Schema, nparts: int,
                 resolve: Callable[[Optional[List[int]]],
                                   Optional[List[int]]],
                 consume: Callable[[int, Frame], None],
                 window_bytes: int = None, tracer=None, pid: int = 0):
        self.comm = comm
        self.schema = schema
        self.nparts = nparts
        # BYTES columns exchange as (lengths, data) tensor pairs; their
        # per-partition byte counts ride extra lanes of the count plane
        self.bytes_cols = [i for i, dt in enumerate(schema.dtypes)
                           if is_bytes(dt)]
        self.resolve = resolve
        self.consume_cb = consume
        self.window_bytes = window_bytes or WINDOW_BYTES
        self.device = comm.device
        # the count plane rides the collective transport: device
        # tensors under nccl, host tensors under gloo
        self._meta_device = (self.device
                             if comm.backend == "nccl" else "cpu")
        self.placement: Optional[List[int]] = None
        self.buf: Dict[int, List[Frame]] = {}
        self.buf_bytes = 0
        self.windows: List[_Window] = []
        self.err_any = False
        self.tracer = tracer
        self.pid = pid

    # -- producer side ----------------------------------------------------

    def add(self, p: int, frame: Frame) -> None:
        if len(frame) == 0:
            return
        self.buf.setdefault(p, []).append(frame)
        self.buf_bytes += frame.nbytes()
        if self.buf_bytes >= self.window_bytes:
            self._push_window(more=True)

    def finish(self, err: bool = False) -> None:
        """Flush the final window, run drain rounds until every rank is
        done, and consume everything received.  After return,
        ``err_any`` says whether any rank flagged an error."""
        if err:
            # promised counts are only gathered per window, so dropping
            # a failed producer's buffered output is safe: nothing was
            # announced for it yet
            self.buf = {}
            self.buf_bytes = 0
        self._push_window(more=False, err=err)
        while True:
            last = self.windows[-1]
            self._resolve_counts(last)
            if not last.more_any:
                break
            self._push_window(more=False, err=err)
        for w in self.windows:
            self._launch_payload(w)
        for w in self.windows:
            self._consume(w)
        self.err_any = any(w.err_any for w in self.windows)

    # -- window machinery -------------------------------------------------

    def _push_window(self, more: bool, err: bool = False) -> None:
        k = len(self.windows)
        # uniform collective launch order on every rank: payload k-1
        # launches before counts k
        if k > 0:
            self._launch_payload(self.windows[k - 1])
        w = _Window()
        nb = len(self.bytes_cols)
        counts = torch.zeros(self.nparts * (1 + nb) + 2,
                             dtype=torch.int64)
        for p, frames in self.buf.items():
            counts[p] = sum(len(f) for f in frames)
            for j, c in enumerate(self.bytes_cols):
                counts[(1 + j) * self.nparts + p] = sum(
                    int(f.columns[c].offsets[-1] -
                        f.columns[c].offsets[0])
                    for f in frames)
        counts[-2] = 1 if err else 0
        counts[-1] = 1 if more else 0
        counts = counts.to(self._meta_device)
        gathered = [torch.empty_like(counts)
                    for _ in range(self.comm.world)]
        w.count_work = dist.all_gather(gathered, counts, async_op=True)
        if self.tracer:
            self.tracer.emit(f"window{k}:counts-launch", "i",
                             pid=self.pid, tid=90)
        w.gathered = gathered
        w.send_buf = self.buf
        self.buf = {}
        self.buf_bytes = 0
        self.windows.append(w)
        pend = [x for x in self.windows if not x.consumed]
        if len(pend) > MAX_IN_FLIGHT + 1:
            self._launch_payload(pend[0])
            self._consume(pend[0])

    def _resolve_counts(self, w: _Window) -> None:
        if w.counts is not None:
            return
        w.count_work.wait()
        host = torch.stack([g.cpu() for g in w.gathered])
        w.gathered = None
        w.err_any = bool(host[:, -2].any())
        w.more_any = bool(host[:, -1].any())
        w.counts = host[:, :self.nparts]
        # per-bytes-col byte counts: [world, nbcols, nparts]
        w.bcounts = host[:, self.nparts:-2].view(
            self.comm.world, len(self.bytes_cols), self.nparts) \
            if self.bytes_cols else None

    def _ensure_placement(self, w: _Window) -> None:
        if self.placement is not None:
            return
        self.placement = self.resolve(None)
        if self.placement is None:
            # first window's global per-partition sizes drive the LPT
            # assignment (a sample of the distribution when more
            # windows follow; exact when this is the only one)
            self.placement = self.resolve(w.counts.sum(dim=0).tolist())

    def _launch_payload(self, w: _Window) -> None:
        if w.launched:
            return
        w.launched = True
        self._resolve_counts(w)
        self._ensure_placement(w)
        world, me = self.comm.world, self.comm.rank
        place = self.placement
        order = sorted(w.send_buf.keys(), key=lambda p: (place[p], p))
        in_splits = [0] * world
        for p in order:
            in_splits[place[p]] += sum(len(f) for f in w.send_buf[p])
        my_meta, out_splits = [], [0] * world
        nb = len(self.bytes_cols)
        bout_splits = [[0] * world for _ in range(nb)]
        for src in range(world):
            for p in range(self.nparts):
                if place[p] != me:
                    continue
                r = int(w.counts[src, p])
                bts = [int(w.bcounts[src, j, p]) for j in range(nb)]
                if r:
                    my_meta.append((src, p, r, bts))
                    out_splits[src] += r
                    for j in range(nb):
                        bout_splits[j][src] += bts[j]
        w.my_meta = my_meta
        total_out = sum(out_splits)
        if int(w.counts.sum()) == 0:
            # globally empty window (every rank sees the same counts):
            # skip the payload collectives entirely
            w.recv_cols = []
            w.send_buf = None
            return
        recv_cols, works = [], []
        dev = torch.device(self.device)

        def exchange(parts, dt, outs, ins):
            send = (torch.cat(parts) if parts else
                    torch.empty(0, dtype=dt, device=self.device))
            if send.device != dev:
                send = send.to(self.device)
            recv = torch.empty(sum(outs), dtype=dt, device=self.device)
            works.append(dist.all_to_all_single(
                recv, send, output_split_sizes=outs,
                input_split_sizes=ins, async_op=True))
            return recv

        for c, dt in enumerate(self.schema.dtypes):
            if is_bytes(dt):
                j = self.bytes_cols.index(c)
                # lengths ride the row splits; data rides byte splits
                lens = exchange(
                    [f.columns[c].lengths()
                     for p in order for f in w.send_buf[p]],
                    torch.int64, out_splits, in_splits)
                bins = [0] * world
                for p in order:
                    bins[place[p]] += sum(
                        int(f.columns[c].offsets[-1] -
                            f.columns[c].offsets[0])
                        for f in w.send_buf[p])
                data = exchange(
                    [f.columns[c].compacted().data
                     for p in order for f in w.send_buf[p]],
                    torch.uint8, bout_splits[j], bins)
                recv_cols.append((lens, data))
            else:
                recv_cols.append(exchange(
                    [f.columns[c].contiguous()
                     for p in order for f in w.send_buf[p]],
                    dt, out_splits, in_splits))
        w.recv_cols = recv_cols
        w.payload_works = works
        if self.tracer:
            self.tracer.emit(
                f"window{self.windows.index(w)}:payload-launch", "i",
                pid=self.pid, tid=90)
        w.send_buf = None  # the work object keeps the send tensors alive

    def _consume(self, w: _Window) -> None:
        if w.consumed:
            return
        w.consumed = True
        span = self.tracer.span(
            f"window{self.windows.index(w)}:payload-wait+consume",
            pid=self.pid, tid=90) if self.tracer else None
        if span:
            span.__enter__()
        for work in w.payload_works:
            work.wait()
        w.payload_works = []
        off = 0
        boff = [0] * len(self.bytes_cols)
        for (src, p, rows, bts) in w.my_meta or []:
            cols = []
            for c, rc in enumerate(w.recv_cols):
                if isinstance(rc, tuple):
                    j = self.bytes_cols.index(c)
                    lens, data = rc
                    l = lens[off:off + rows]
                    offs = torch.zeros(rows + 1, dtype=torch.int64,
                                       device=l.device)
                    torch.cumsum(l, 0, out=offs[1:])
                    cols.append(BytesColumn(
                        data[boff[j]:boff[j] + bts[j]], offs))
                    boff[j] += bts[j]
                else:
                    cols.append(rc[off:off + rows])
            self.consume_cb(p, Frame(cols, self.schema.prefix))
            off += rows
        w.recv_cols = []
        if span:
            span.__exit__(None, None, None)
