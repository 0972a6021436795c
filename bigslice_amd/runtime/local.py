"""Local executor: in-process shard tasks on the local device.

Role-parity: exec/local.go — a semaphore of p procs runs task pipelines,
buffering partitioned output in an in-memory store (taskBuffer analog).
On a GPU host every task's batches live in HBM; tasks multiplex onto the
device concurrently (HIP streams via torch's per-thread dispatch).
"""

from __future__ import annotations

import threading
import traceback

import torch

from .. import config
from ..ops.slice_base import TaskContext
from ..utils import metrics
from ..sliceio import MultiReader, Reader
from .eval import Executor
from .partition import PartitionWriter
from .store import MemoryStore, Store
from .task import Task, TaskState


class TaskLost(Exception):
    """Transient failure: the evaluator should resubmit (exec/eval.go)."""


class LocalExecutor(Executor):
    def __init__(self, parallelism: int = None, device: str = None,
                 store: Store = None):
        self.parallelism = parallelism or config.DEFAULT_PARALLELISM
        if device is None:
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.store = store or MemoryStore()
        # proc accounting (Pragma Procs/Exclusive, slice.go:109-200):
        # a task takes pragma.procs permits; Exclusive takes them all.
        self._avail = self.parallelism
        self._res_cond = threading.Condition()
        # persistent worker pool reused across evaluations (one thread
        # per concurrent shard task; GPU work overlaps via the HIP
        # stream while threads trade the GIL)
        from concurrent.futures import ThreadPoolExecutor
        self.pool = ThreadPoolExecutor(max_workers=self.parallelism)
        # fault injection hook for chaos tests: fn(task) -> None or raise
        self.fault_hook = None
        self._tls = threading.local()
        self.scopes = {}  # task name -> metrics.Scope
        self.tracer = None  # utils.trace.Tracer, set by the session
        self.eventer = None  # utils.eventlog.Eventer, set by the session
        # machine-combiners mode (reference exec/session.go:166-176):
        # producer tasks of one shuffle phase share one combiner table.
        self.machine_combiners = config.MACHINE_COMBINERS
        self._shared_combiners = {}  # phase id -> SharedPhaseCombiner
        self._shared_lock = threading.Lock()
        # Task-completion ordering.  Default: hard per-task stream
        # sync.  BIGSLICE_TASK_SYNC=0 switches to event-based ordering
        # (consumers wait_event, driver synchronizes events for host
        # reads) — measured neutral-to-slightly-negative on the
        # BASELINE configs (the per-task syncs overlap across the 8
        # worker threads, and letting phases overlap hurts the
        # atomic-bound combine kernels), kept as an opt-in experiment.
        import os as _os
        self._hard_sync = _os.environ.get("BIGSLICE_TASK_SYNC",
                                          "1") == "1"
        # Inline execution: tasks run on the evaluating thread (no
        # pool handoffs).  "auto" = inline on GPU sessions and at
        # parallelism==1: device work is launched async on streams, so
        # serial host-side launching loses no overlap while the
        # per-task thread wake/park latency disappears — measured on
        # MI355X: config 2 9.7->3.7 ms, config 3 at 125M 9.6->8.5 ms,
        # config 4 equal-to-3x-better (the pool path is wildly
        # box-variable under CPU contention), config 5 equal.  CPU
        # sessions keep the pool: worker threads are the actual
        # compute parallelism there.
        _inl = _os.environ.get("BIGSLICE_INLINE", "auto")
        self.inline = (_inl == "1"
                       or (_inl == "auto"
                           and (self.parallelism == 1
                                or self.device.startswith("cuda"))))
        self._events = {}  # task name -> torch.cuda.Event
        # prewarm the HBM high-water cache from this (main) thread:
        # CUDA device queries can fail when first issued from workers
        from ..frame import over_high_water
        over_high_water()

    def _record_done(self, task: Task, also_first: bool = False):
        """Record the task's completion event on its stream (consumers
        order against it; the driver synchronizes it for host reads)."""
        if self._hard_sync or not self.device.startswith("cuda"):
            return
        ev = torch.cuda.Event()
        ev.record()
        self._events[task.name] = ev
        if also_first:
            self._events[task.group[0].name] = ev

    def _stream(self):
        """Per-worker-thread HIP stream: concurrent shard tasks overlap
        on the device and one task's sync does not convoy behind
        another's queued kernels."""
        if not self.device.startswith("cuda"):
            return None
        s = getattr(self._tls, "stream", None)
        if s is None:
            s = torch.cuda.Stream(device=self.device)
            self._tls.stream = s
        return s

    # -- Executor ---------------------------------------------------------

    def run(self, task: Task) -> None:
        pragma = task.pragma
        want = self.parallelism if (pragma and pragma.exclusive) else \
            min(pragma.procs if pragma else 1, self.parallelism)
        want = max(want, 1)
        with self._res_cond:
            self._res_cond.wait_for(lambda: self._avail >= want)
            self._avail -= want
        task.set_state(TaskState.RUNNING)
        import time as _time
        _t0 = _time.perf_counter()
        try:
            if self.fault_hook is not None:
                self.fault_hook(task)
            scope = metrics.Scope()
            span = (self.tracer.span(task.name, pid=0)
                    if self.tracer else None)
            with metrics.scoped(scope):
                if span:
                    span.__enter__()
                stream = self._stream()
                if stream is not None:
                    with torch.cuda.stream(stream):
                        self._run_inner(task)
                    if self._hard_sync:
                        # stored frames materialized before any reader
                        stream.synchronize()
                else:
                    self._run_inner(task)
                if span:
                    span.__exit__(None, None, None)
            self.scopes[task.name] = scope
            task.set_state(TaskState.OK)
        except TaskLost as e:
            task.set_state(TaskState.LOST)
        except Exception as e:
            e.task_traceback = traceback.format_exc()
            task.set_state(TaskState.ERR, e)
        finally:
            if self.eventer is not None:
                self.eventer.task_complete(
                    task.name, _time.perf_counter() - _t0,
                    task.state.name)
            with self._res_cond:
                self._avail += want
                self._res_cond.notify_all()

    def _run_inner(self, task: Task) -> None:
        ctx = TaskContext(device=self.device)
        dep_readers = []
        missing = []
        use_cuda = self.device.startswith("cuda")
        for dep in task.deps:
            readers = []
            for h in dep.head_tasks:
                if use_cuda:
                    ev = self._events.get(h.name)
                    if ev is not None:
                        torch.cuda.current_stream().wait_event(ev)
                try:
                    readers.append(self.store.open(
                        h.name, dep.partition, device=self.device))
                except KeyError:
                    missing.append(h)
            if dep.expand:
                dep_readers.append(readers)
            else:
                dep_readers.append(MultiReader(readers))
        if missing:
            # Stored dep outputs vanished (machine-loss analog): mark
            # every missing producer LOST in one pass so the evaluator
            # recomputes them all before this task's single resubmit
            # (exec/eval.go:352-376 semantics).
            for h in missing:
                h.set_state(TaskState.LOST)
            raise TaskLost(
                f"missing dep outputs: {[h.name for h in missing]}")
        out = task.do(dep_readers, ctx)
        if task.num_out_columns == 0:
            # terminal (Scan) task: drive it (exec/local.go:188-193)
            for _ in out:
                pass
            self.store.put(task.name, 0, [], 0)
            self._record_done(task)
            return
        shared = self._shared_combiner_for(task)
        if shared is not None:
            for f in out:
                shared.add(f)
            if shared.task_done():
                buckets = shared.finish_buckets()
                # combined output stored under the phase's first task;
                # siblings store empty partitions (consumers concat all)
                first = task.group[0]
                for pi, frames in enumerate(buckets):
                    rows = sum(len(f) for f in frames)
                    self.store.put(first.name, pi, frames, rows)
                with self._shared_lock:
                    self._shared_combiners.pop(id(task.group[0]), None)
                # the finisher's kernels produced group[0]'s stored
                # output; consumers of first.name must order on THIS
                # task's event
                self._record_done(task, also_first=True)
            if task is not task.group[0]:
                for pi in range(task.num_partitions):
                    self.store.put(task.name, pi, [], 0)
            elif task.group[0] is task and \
                    not self.store.has(task.name, 0):
                # first task finished before the phase completed: its
                # entries are written by the last finisher above; make
                # sure empty markers exist if it was also the last.
                pass
            if task.name not in self._events:
                self._record_done(task)
            return
        w = PartitionWriter(task.num_partitions, task.partitioner,
                            task.combiner, task.schema, self.device,
                            ctx.chunk)
        for f in out:
            w.add(f)
        buckets = w.finish()
        for pi, frames in enumerate(buckets):
            rows = sum(len(f) for f in frames)
            self.store.put(task.name, pi, frames, rows)
        self._record_done(task)

    def _shared_combiner_for(self, task: Task):
        """The phase's shared combiner, when machine-combiners mode
        applies: combiner present, default hash partitioner, no fault
        injection (the mode has no loss recovery — reference
        exec/session.go:166-176), and a multi-task phase."""
        if (not self.machine_combiners or task.combiner is None
                or task.partitioner is not None
                or self.fault_hook is not None
                or len(task.group) <= 1):
            return None
        key = id(task.group[0])
        with self._shared_lock:
            sc = self._shared_combiners.get(key)
            if sc is None:
                if any(s.state == TaskState.OK for s in task.group
                       if s is not task):
                    # Partial phase re-run (output loss after the shared
                    # combine completed): the mode cannot recover — fall
                    # back to solo per-task combining and invalidate any
                    # surviving sibling outputs so consumers recompute
                    # them too (they re-enqueue via the missing-dep
                    # TaskLost path).
                    for s in task.group:
                        self.store.discard_task(s.name)
                    return None
                from .partition import SharedPhaseCombiner
                sc = SharedPhaseCombiner(
                    task.schema, task.combiner, self.device,
                    task.num_partitions, len(task.group),
                    TaskContext(device=self.device).chunk)
                self._shared_combiners[key] = sc
            return sc

    def reader(self, task: Task, partition: int) -> Reader:
        ev = self._events.get(task.name)
        if ev is not None:
            ev.synchronize()  # host read: the event must be complete
        return self.store.open(task.name, partition, device="cpu")

    def discard(self, task: Task) -> None:
        self._events.pop(task.name, None)
        self.store.discard_task(task.name)

    def shutdown(self) -> None:
        self.pool.shutdown(wait=False)

    def merged_scope(self, tasks):
        """Merge task scopes across the reachable graph
        (Result.Scope semantics, exec/session.go:418-426)."""
        out = metrics.Scope()
        seen = set()

        def visit(t):
            if t.name in seen:
                return
            seen.add(t.name)
            s = self.scopes.get(t.name)
            if s is not None:
                out.merge(s)
            for dep in t.deps:
                for h in dep.head_tasks:
                    visit(h)
        for t in tasks:
            visit(t)
        return out
