"""Sessions, Funcs, Invocations and Results.

Role-parity: func.go (deterministic Func registry with race detection and
cross-process registry verification via location diffs) and exec/session.go
(Start -> Session; Session.Run: Invocation -> Invoke -> compile -> Eval;
Result is a Slice usable as an argument to further Funcs, enabling
iterative computing).
"""

from __future__ import annotations

import hashlib
import os
import threading
from typing import Callable, List, Sequence

from ..ops.slice_base import Name, Slice, TaskContext
from ..sliceio import MultiReader, Reader, Scanner
from .compile import CompileEnv, Compiler
from .eval import Executor, evaluate
from .task import Task

# -- Func registry (func.go:160-343) ------------------------------------

_funcs: List["FuncValue"] = []
_funcs_lock = threading.Lock()
_funcs_busy = 0  # count of sessions currently running (func.go:26-28)


class FuncValue:
    """A registered slice-constructor function."""

    def __init__(self, fn: Callable, index: int, location: str,
                 exclusive: bool = False):
        self.fn = fn
        self.index = index
        self.location = location
        self.exclusive = exclusive

    def invocation(self, args: Sequence) -> "Invocation":
        return Invocation(self.index, args)

    def invoke(self, args: Sequence) -> Slice:
        s = self.fn(*args)
        if not isinstance(s, Slice):
            raise TypeError(
                f"Func at {self.location} returned {type(s)}, not a Slice")
        return s

    def __call__(self, *args) -> Slice:
        return self.invoke(args)


def func(fn: Callable, exclusive: bool = False) -> FuncValue:
    """Register a slice-constructor (bigslice.Func).  Registration order
    must be deterministic across worker processes (func.go:26-28); the
    registry digest is verified at distributed start."""
    global _funcs_busy
    with _funcs_lock:
        if _funcs_busy > 0:
            raise RuntimeError(
                "bigslice_amd.func called while a session is running; "
                "Funcs must be registered at module init "
                "(reference func.go:175-181)")
        import inspect
        frame = inspect.currentframe().f_back
        # skip internal frames
        while frame is not None and "bigslice_amd" in frame.f_code.co_filename:
            frame = frame.f_back
        loc = (f"{frame.f_code.co_filename}:{frame.f_lineno}"
               if frame else "<unknown>")
        fv = FuncValue(fn, len(_funcs), loc, exclusive)
        _funcs.append(fv)
        return fv


def func_locations() -> List[str]:
    with _funcs_lock:
        return [f.location for f in _funcs]


def registry_digest() -> str:
    """Digest of the Func registry; compared across worker processes
    (the FuncLocations diff check, exec/slicemachine.go:689-702)."""
    h = hashlib.sha256()
    for loc in func_locations():
        h.update(loc.encode())
        h.update(b"\0")
    return h.hexdigest()


def _mark_busy(busy: bool):
    global _funcs_busy
    with _funcs_lock:
        _funcs_busy += 1 if busy else -1


class Invocation:
    """A serializable record of a Func application (func.go:218-258)."""

    def __init__(self, func_index: int, args: Sequence):
        self.func_index = func_index
        self.args = list(args)

    def invoke(self) -> Slice:
        return _funcs[self.func_index].invoke(self.args)


# -- Result (exec/session.go:394-434) ------------------------------------

class Result(Slice):
    """The computed output of an invocation; a Slice over the root tasks'
    stored partitions, reusable as an argument to further Funcs."""

    def __init__(self, session: "Session", slice_: Slice,
                 tasks: List[Task]):
        self.session = session
        self.slice = slice_
        self.tasks = tasks
        super().__init__(slice_.schema, slice_.num_shards,
                         name=Name("result"))

    def reader(self, shard: int, dep_readers, ctx: TaskContext) -> Reader:
        return self.session.executor.reader(self.tasks[shard], 0)

    def open(self) -> Reader:
        """Reader over all shards' outputs.  With a distributed executor
        this is a collective call: every rank must call it; rank 0 sees
        all rows, other ranks see none."""
        gather = getattr(self.session.executor, "gather_result", None)
        if gather is not None:
            frames = gather(self.tasks, self.schema)
            from ..sliceio import IterReader
            return IterReader(iter(frames))
        # Scan-time fault tolerance (reference evalOpenerAt,
        # exec/bigmachine.go:1500-1513): a task whose stored output
        # vanished since the run is re-evaluated before reading.
        session = self.session

        def open_task(t):
            try:
                return session.executor.reader(t, 0)
            except KeyError:
                from ..runtime.task import TaskState
                from .eval import evaluate
                t.set_state(TaskState.LOST)
                evaluate(session.executor, [t])
                return session.executor.reader(t, 0)

        from ..sliceio import Reader

        class _LazyTaskReader(Reader):
            def __init__(self, t):
                self.t = t
                self.r = None

            def read(self):
                if self.r is None:
                    self.r = open_task(self.t)
                return self.r.read()

        return MultiReader([_LazyTaskReader(t) for t in self.tasks])

    def scanner(self) -> Scanner:
        return Scanner(self.open())

    def __iter__(self):
        return self.scan()

    def scan(self):
        """Iterate result rows."""
        return self.scanner().rows()

    def scope(self):
        """Merged user-metric scope over this result's task graph
        (Result.Scope, exec/session.go:418-426).  Collective with a
        distributed executor."""
        ms = getattr(self.session.executor, "merged_scope", None)
        if ms is None:
            from ..utils import metrics
            return metrics.Scope()
        return ms(self.tasks)

    def discard(self) -> None:
        """Free stored task outputs (session.go:231-251)."""
        seen = set()

        def visit(t: Task):
            if t.name in seen:
                return
            seen.add(t.name)
            self.session.executor.discard(t)
            for dep in t.deps:
                for h in dep.head_tasks:
                    visit(h)
        for t in self.tasks:
            visit(t)


def _is_comm_failure(e: BaseException) -> bool:
    """Transport-level failure (a peer died) vs an application error
    that the engine already surfaced collectively on live ranks."""
    try:
        import torch.distributed as dist
        if isinstance(e, dist.DistBackendError):
            return True
    except Exception:
        pass
    if not isinstance(e, RuntimeError):
        return False
    msg = str(e)
    return any(s in msg for s in (
        "onnection", "Socket", "Timed out", "imed out", "peer",
        "NCCL", "Gloo", "aborted"))


# -- Session ---------------------------------------------------------------

class Session:
    """An execution session bound to an executor (exec/session.go)."""

    def __init__(self, executor: Executor, parallelism: int = None,
                 trace_path: str = None, eventlog_path: str = None):
        self.executor = executor
        self.parallelism = parallelism
        self._inv_counter = 0
        self._lock = threading.Lock()
        self.env = CompileEnv()
        self.trace_path = trace_path
        from ..utils.eventlog import Eventer, NOP
        self.eventer = Eventer(eventlog_path) if eventlog_path else NOP
        if hasattr(executor, "eventer"):
            executor.eventer = self.eventer
        self.eventer.session_start()
        if trace_path is not None:
            from ..utils.trace import Tracer
            self.tracer = Tracer()
            if hasattr(executor, "tracer"):
                executor.tracer = self.tracer
        else:
            self.tracer = None

    def run(self, funcv: FuncValue, *args) -> Result:
        if not isinstance(funcv, FuncValue):
            raise TypeError("Session.run takes a registered Func "
                            "(bigslice_amd.func)")
        _mark_busy(True)
        try:
            with self._lock:
                self._inv_counter += 1
                inv_index = self._inv_counter
            slice_ = funcv.invoke(args)
            # fresh CompileEnv per run: cache decisions are re-checked
            # each invocation (exec/compile.go recomputes presence) and
            # concurrent runs must not share frozen state.  In
            # distributed mode rank 0 computes all cache decisions and
            # broadcasts the frozen env; other ranks compile against it
            # without touching the filesystem (exec/compile.go:125-184)
            # so a racing cache write cannot yield divergent graphs.
            env = CompileEnv()
            compiler = Compiler(inv_index, env)
            comm = getattr(self.executor, "comm", None)
            if comm is not None and comm.world > 1:
                if comm.rank == 0:
                    tasks = compiler.compile(slice_)
                    comm.broadcast_obj(env.cached, src=0)
                else:
                    env.seal(comm.broadcast_obj(None, src=0))
                    tasks = compiler.compile(slice_)
            else:
                tasks = compiler.compile(slice_)
            if funcv.exclusive:
                # Exclusive Funcs (func.go Exclusive): their tasks do
                # not share the executor with others (locally: take all
                # proc permits; reference: a dedicated cluster).
                from ..ops.slice_base import Pragma
                seen = set()

                def mark(t):
                    if id(t) in seen:
                        return
                    seen.add(id(t))
                    p = t.pragma or Pragma()
                    t.pragma = Pragma(procs=p.procs, exclusive=True,
                                      materialize=p.materialize)
                    for dep in t.deps:
                        for h in dep.head_tasks:
                            mark(h)
                for t in tasks:
                    mark(t)
            env.freeze()
            ev = getattr(self.executor, "evaluate", None)
            if ev is not None:
                # In-run rank-loss recovery: a transport failure (a
                # peer died inside a collective) triggers a shrink-
                # rebuild + re-evaluate when the executor has
                # persistent checkpoints; completed phases skip, lost
                # partitions recompute on the survivors.
                attempts = 0
                while True:
                    try:
                        ev(tasks)
                        break
                    except BaseException as e:
                        rec = getattr(self.executor, "recoverable",
                                      False)
                        comm_fail = _is_comm_failure(e)
                        if os.environ.get("BIGSLICE_RECOVERY_DEBUG"):
                            import sys as _s
                            print(f"[recovery-gate] {type(e).__name__} "
                                  f"recoverable={rec} "
                                  f"comm_fail={comm_fail} "
                                  f"attempts={attempts}: "
                                  f"{str(e)[:120]}",
                                  file=_s.stderr, flush=True)
                        if rec and comm_fail and attempts < 3:
                            attempts += 1
                            self.executor.recover(tasks)
                            continue
                        raise
            else:
                evaluate(self.executor, tasks)
            return Result(self, slice_, tasks)
        finally:
            _mark_busy(False)

    def must(self, funcv: FuncValue, *args) -> Result:
        return self.run(funcv, *args)

    def __enter__(self) -> "Session":
        return self

    def __exit__(self, *exc):
        self.shutdown()
        return False

    def shutdown(self):
        if self.tracer is not None and self.trace_path:
            self.tracer.write(self.trace_path)
        sd = getattr(self.executor, "shutdown", None)
        if sd is not None:
            sd()

    def status(self):
        """Task-state counts by invocation (the status display's
        rollup, exec/slicestatus.go)."""
        return dict(self._status) if hasattr(self, "_status") else {}


def start(parallelism: int = None, device: str = None,
          executor: Executor = None, distributed: bool = None,
          trace_path: str = None, eventlog_path: str = None,
          checkpoint_dir: str = None,
          machine_combiners: bool = None) -> Session:
    """Create a session (exec.Start analog; option set parity with
    exec/session.go:98-176 — parallelism, executor choice, trace path,
    eventer, machine combiners, checkpointing).

    distributed=True (or WORLD_SIZE>1 in the environment) starts the SPMD
    executor: one process per GPU over RCCL; every rank must call the
    same Session methods in the same order.  Otherwise the local executor
    runs on cuda:0 when a GPU is visible, else CPU.
    """
    import os
    if distributed is None:
        distributed = int(os.environ.get("WORLD_SIZE", "1")) > 1
    if executor is None:
        if distributed:
            from ..parallel.comm import init_comm
            from .dist import DistExecutor
            comm = init_comm(device=device)
            comm.check_registry(registry_digest())
            store = None
            if checkpoint_dir is not None:
                from .store import FileStore
                store = FileStore(
                    os.path.join(checkpoint_dir,
                                 f"rank{comm.rank:03d}"),
                    peers_dir=checkpoint_dir)
            executor = DistExecutor(comm, store=store)
            if checkpoint_dir is not None:
                # enables in-run shrink-recovery after a rank loss
                executor.recovery_dir = os.path.join(checkpoint_dir,
                                                     "recovery")
        else:
            from .local import LocalExecutor
            store = None
            if checkpoint_dir is not None:
                from .store import FileStore
                store = FileStore(checkpoint_dir)
            executor = LocalExecutor(parallelism=parallelism,
                                     device=device, store=store)
    if machine_combiners is not None and hasattr(executor,
                                                 "machine_combiners"):
        executor.machine_combiners = machine_combiners
    return Session(executor, parallelism, trace_path=trace_path,
                   eventlog_path=eventlog_path)
