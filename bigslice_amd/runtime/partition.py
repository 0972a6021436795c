"""Partitioning task output into shuffle buckets (kernel K4).

Role-parity: the reference's defaultPartitioner (exec/compile.go:20-24,
hash%nshard) and scatter loop (exec/bigmachine.go:960-996).  The MI355X
path is a fused hash+scatter: compute murmur3 of the key prefix, stable-sort
rows by partition id, and slice the reordered frame into zero-copy
contiguous buckets — exactly the layout an RCCL all-to-allv wants.  On GPU
this dispatches the HIP partition kernel when available; the torch fallback
(argsort+gather) is used on CPU.
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch

from ..frame import Frame


def partition_ids(frame: Frame, num_partitions: int,
                  partitioner: Optional[Callable] = None) -> torch.Tensor:
    """Shard id per row (int64)."""
    if partitioner is not None:
        p = partitioner(frame, num_partitions)
        if not isinstance(p, torch.Tensor):
            p = torch.tensor(list(p), dtype=torch.int64)
        return p.to(torch.int64)
    h = frame.hash(0)
    return (h.to(torch.int64) & 0xFFFFFFFF) % num_partitions


def split_frame(frame: Frame, num_partitions: int,
                partitioner: Optional[Callable] = None) -> List[Optional[Frame]]:
    """Split a frame into per-partition sub-frames (None when empty)."""
    if num_partitions == 1:
        return [frame]
    if len(frame) == 0:
        return [None] * num_partitions
    if frame.device != "cpu" and not frame.has_objects:
        from .. import kernels
        # the LDS histogram kernel supports up to 4096 partitions; wider
        # fans fall back to the sort-based split below
        if num_partitions <= 4096 and kernels.partition_supported(frame):
            return kernels.partition_frame(frame, num_partitions,
                                           partitioner)
    p = partition_ids(frame, num_partitions, partitioner)
    order = torch.argsort(p, stable=True)
    sorted_f = frame.select(order)
    counts = torch.bincount(p, minlength=num_partitions)
    out: List[Optional[Frame]] = []
    off = 0
    for c in counts.tolist():
        out.append(sorted_f.slice(off, off + c) if c else None)
        off += c
    return out


class SharedPhaseCombiner:
    """Per-GPU shared combiner for one shuffle phase (the reference's
    machine-combiners mode, exec/session.go:166-176 + bigmachine.go
    combine keys): every producer task of the phase inserts into ONE
    hash-aggregate table, and the last task to finish compacts,
    partitions and stores the combined output once.

    Known limitation carried from the reference: no error recovery —
    the local executor only enables this when no fault injection is
    active (tasks cannot be LOST mid-phase).
    """

    def __init__(self, schema, combiner, device: str, num_partitions: int,
                 expected_tasks: int, chunk: int):
        import threading

        from ..ops.aggregate import make_aggregator
        self.agg = make_aggregator(schema, combiner, device)
        self.device = device
        self.num_partitions = num_partitions
        self.chunk = chunk
        self.remaining = expected_tasks
        self.lock = threading.Lock()
        self.events = []
        self.rows = 0

    def add(self, frame: Frame) -> None:
        with self.lock:
            self.rows += len(frame)
            self.agg.add(frame)

    def task_done(self) -> bool:
        """Record this task's inserts (stream event) and return True for
        the last task, which must then call finish_buckets."""
        if self.device.startswith("cuda"):
            ev = torch.cuda.Event()
            ev.record()
        else:
            ev = None
        with self.lock:
            if ev is not None:
                self.events.append(ev)
            self.remaining -= 1
            return self.remaining == 0

    def finish_buckets(self) -> List[List[Frame]]:
        # all other tasks' insert kernels must complete first
        if self.device.startswith("cuda"):
            cur = torch.cuda.current_stream()
            for ev in self.events:
                cur.wait_event(ev)
        buckets: List[List[Frame]] = \
            [[] for _ in range(self.num_partitions)]
        for f in self.agg.result_frames(self.chunk * 4):
            for pi, pf in enumerate(split_frame(f, self.num_partitions,
                                                None)):
                if pf is not None and len(pf):
                    buckets[pi].append(pf)
        return buckets


class PartitionWriter:
    """Accumulates a task's output into per-partition frame lists, with
    optional producer-side pre-combine (the reference's combiner
    machinery, exec/bigmachine.go:1084-1210)."""

    def __init__(self, num_partitions: int, partitioner, combiner,
                 schema, device: str, chunk: int):
        self.num_partitions = num_partitions
        self.partitioner = partitioner
        self.schema = schema
        self.device = device
        self.chunk = chunk
        self.rows = 0
        self.agg = None
        self.aggs = None
        self.buckets: List[List[Frame]] = \
            [[] for _ in range(num_partitions)]
        if combiner is not None:
            from ..ops.aggregate import make_aggregator
            if partitioner is None:
                # Combine-first: the default partitioner is a pure key
                # function, so pre-combining the whole shard THEN
                # splitting the (much smaller) combined result is
                # equivalent to the reference's per-partition combiners
                # and turns 8 small hash tables + scatters into one big
                # streaming insert — the GPU-friendly order.
                self.agg = make_aggregator(schema, combiner, device)
            else:
                # Custom partitioners may not be key-pure: keep the
                # partition-then-combine order.
                self.aggs = [make_aggregator(schema, combiner, device)
                             for _ in range(num_partitions)]

    def add(self, frame: Frame) -> None:
        self.rows += len(frame)
        if self.agg is not None:
            self.agg.add(frame)
            return
        parts = split_frame(frame, self.num_partitions, self.partitioner)
        from ..frame import over_high_water
        tier = frame.device != "cpu" and over_high_water()
        for pi, pf in enumerate(parts):
            if pf is None or len(pf) == 0:
                continue
            if self.aggs is not None:
                self.aggs[pi].add(pf)
            elif tier:
                # accumulating past the HBM high-water mark: hold the
                # bucket in pinned host DRAM (e.g. a sort whose output
                # cannot coexist with its input in HBM)
                self.buckets[pi].append(pf.to_pinned_host())
            else:
                self.buckets[pi].append(pf)

    def finish(self) -> List[List[Frame]]:
        """Per-partition frame lists."""
        if self.agg is not None:
            for f in self.agg.result_frames(self.chunk * 4):
                parts = split_frame(f, self.num_partitions, None)
                for pi, pf in enumerate(parts):
                    if pf is not None and len(pf):
                        self.buckets[pi].append(pf)
            return self.buckets
        if self.aggs is not None:
            return [list(a.result_frames(self.chunk)) for a in self.aggs]
        return self.buckets
