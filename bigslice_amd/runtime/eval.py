"""Evaluator: schedules the task graph to an executor.

Role-parity: exec/eval.go:80-176 — enqueue roots, run runnable tasks
concurrently, resubmit LOST tasks together with their now-missing
dependencies (:112-115, :352-376), give up after maxConsecutiveLost
(:30, :139-159).  Multiple concurrent Evals coordinate through task
state: a task already WAITING/RUNNING is watched, not re-run.

Implementation: callback-driven over a persistent thread pool (no
watcher-thread-per-task, no polling): completion of a task decrements its
consumers' pending counts and submits newly-runnable tasks.
"""

from __future__ import annotations

import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Sequence, Set

from .. import config
from .task import Task, TaskState


class TooManyTriesError(RuntimeError):
    def __init__(self, task: Task):
        super().__init__(
            f"task {task.name} lost {task.consecutive_lost} consecutive "
            f"times; giving up")
        self.task = task


class Executor:
    """Executor interface (exec/eval.go:42-71)."""

    def run(self, task: Task) -> None:
        """Run the task, setting its state to RUNNING then OK/ERR/LOST.
        Called from an evaluator pool thread; may block."""
        raise NotImplementedError

    def reader(self, task: Task, partition: int):
        """Open a reader over a completed task's output partition."""
        raise NotImplementedError

    def discard(self, task: Task) -> None:
        pass


def _reachable(roots: Sequence[Task]) -> List[Task]:
    out: List[Task] = []
    seen: Set[int] = set()

    def visit(t: Task):
        if id(t) in seen:
            return
        seen.add(id(t))
        for dep in t.deps:
            for h in dep.head_tasks:
                visit(h)
        out.append(t)

    for r in roots:
        visit(r)
    return out


def evaluate(executor: Executor, roots: Sequence[Task]) -> None:
    """Evaluate all root tasks to OK (or raise)."""
    tasks = _reachable(roots)
    consumers: Dict[int, List[Task]] = {}
    for t in tasks:
        for dep in t.deps:
            for h in dep.head_tasks:
                consumers.setdefault(id(h), []).append(t)

    lock = threading.Lock()
    done = threading.Condition(lock)
    errors: List[BaseException] = []
    # tasks this evaluation has claimed (submitted or is watching)
    active: Set[int] = set()

    pool = getattr(executor, "pool", None)
    own_pool = None
    if pool is None:
        own_pool = ThreadPoolExecutor(
            max_workers=getattr(executor, "parallelism", None)
            or config.DEFAULT_PARALLELISM)
        pool = own_pool

    def deps_ok(t: Task) -> bool:
        return all(h.state == TaskState.OK
                   for dep in t.deps for h in dep.head_tasks)

    def fail(e: BaseException):
        with lock:
            errors.append(e)
            done.notify_all()

    def submit(t: Task):
        """Called with lock held.  Schedule t if runnable, else make its
        missing deps runnable first."""
        if id(t) in active:
            return
        st = t.state
        if st == TaskState.OK:
            return
        if st == TaskState.ERR:
            errors.append(t.error or RuntimeError(f"{t.name} failed"))
            done.notify_all()
            return
        if t.consecutive_lost >= config.MAX_CONSECUTIVE_LOST:
            errors.append(TooManyTriesError(t))
            done.notify_all()
            return
        if st in (TaskState.WAITING, TaskState.RUNNING):
            # claimed by a concurrent evaluation: watch it
            active.add(id(t))
            threading.Thread(target=_watch, args=(t,), daemon=True).start()
            return
        if deps_ok(t):
            active.add(id(t))
            t.set_state(TaskState.WAITING)
            pool.submit(_run, t)
        else:
            for dep in t.deps:
                for h in dep.head_tasks:
                    if h.state != TaskState.OK:
                        submit(h)

    def _watch(t: Task):
        t.wait_state(TaskState.OK)
        _finish(t)

    def _run(t: Task):
        try:
            executor.run(t)
        except BaseException as e:
            t.set_state(TaskState.ERR, e)
        _finish(t)

    def _finish(t: Task):
        st = t.state
        with lock:
            active.discard(id(t))
            if st == TaskState.OK:
                for c in consumers.get(id(t), ()):  # wake consumers
                    if id(c) not in active and c.state in (
                            TaskState.INIT, TaskState.LOST) and deps_ok(c):
                        submit(c)
                # a consumer may be waiting only on this task at root
                done.notify_all()
            elif st == TaskState.LOST:
                submit(t)  # resubmit (re-walks lost deps)
                done.notify_all()
            else:
                errors.append(t.error or RuntimeError(f"{t.name} failed"))
                done.notify_all()

    try:
        with lock:
            for r in roots:
                submit(r)
            while not errors:
                if all(r.state == TaskState.OK for r in roots):
                    return
                if not active:
                    # quiescent but incomplete: re-walk from roots
                    # (lost interior tasks)
                    progressed = False
                    for r in roots:
                        if r.state != TaskState.OK:
                            submit(r)
                            progressed = True
                    if not progressed or not active:
                        if errors:
                            break
                        if all(r.state == TaskState.OK for r in roots):
                            return
                done.wait(timeout=1.0)
            raise errors[0]
    finally:
        if own_pool is not None:
            own_pool.shutdown(wait=False)
