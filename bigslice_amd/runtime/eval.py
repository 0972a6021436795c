"""Evaluator: schedules the task graph to an executor.

Role-parity: exec/eval.go:80-176 — enqueue roots, run runnable tasks
concurrently, watch running tasks, resubmit LOST tasks together with their
now-missing dependencies (:112-115, :352-376), give up after
maxConsecutiveLost (:30, :139-159).  Multiple concurrent Evals coordinate
through task state: a task already WAITING/RUNNING is watched, not re-run.
"""

from __future__ import annotations

import threading
from typing import List, Sequence

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
        Called from an evaluator goroutine; may block."""
        raise NotImplementedError

    def reader(self, task: Task, partition: int):
        """Open a reader over a completed task's output partition."""
        raise NotImplementedError

    def discard(self, task: Task) -> None:
        pass


def evaluate(executor: Executor, roots: Sequence[Task]) -> None:
    """Evaluate all root tasks to OK (or raise)."""
    pending_lock = threading.Lock()
    done_event = threading.Event()
    errors: List[BaseException] = []
    # Tasks this evaluation is responsible for watching.
    watched = set()
    inflight = [0]

    def note_error(e: BaseException):
        with pending_lock:
            errors.append(e)
        done_event.set()

    def runnable(task: Task) -> bool:
        return all(t.state == TaskState.OK
                   for dep in task.deps for t in dep.head_tasks)

    def enqueue(task: Task):
        """Walk the graph; run tasks whose deps are satisfied; recurse
        into deps otherwise (phase-aware recomputation: a LOST dep is
        re-enqueued; exec/eval.go:255-451 semantics, simplified to
        O(edges))."""
        with pending_lock:
            if task in watched and task.state in (
                    TaskState.WAITING, TaskState.RUNNING):
                return
        st = task.state
        if st == TaskState.OK:
            return
        if st == TaskState.ERR:
            note_error(task.error or RuntimeError(f"{task.name} failed"))
            return
        if task.consecutive_lost >= config.MAX_CONSECUTIVE_LOST:
            note_error(TooManyTriesError(task))
            return
        if runnable(task):
            start(task)
        else:
            # watch deps; when they complete we revisit this task
            for dep in task.deps:
                for h in dep.head_tasks:
                    enqueue(h)
            watch_until_deps_ready(task)

    def start(task: Task):
        with pending_lock:
            if task in watched and task.state in (
                    TaskState.WAITING, TaskState.RUNNING):
                return
            watched.add(task)
            inflight[0] += 1
        task.set_state(TaskState.WAITING)

        def runner():
            try:
                executor.run(task)
            except BaseException as e:  # executor bug; treat as ERR
                task.set_state(TaskState.ERR, e)
            finally:
                finish(task)

        threading.Thread(target=runner, daemon=True).start()

    def watch_until_deps_ready(task: Task):
        with pending_lock:
            if task in watched:
                return
            watched.add(task)
            inflight[0] += 1

        def waiter():
            try:
                for dep in task.deps:
                    for h in dep.head_tasks:
                        h.wait_state(TaskState.OK)
                with pending_lock:
                    watched.discard(task)
                enqueue(task)
            finally:
                finish_watch()
        threading.Thread(target=waiter, daemon=True).start()

    def finish_watch():
        with pending_lock:
            inflight[0] -= 1
            if inflight[0] == 0:
                done_event.set()

    def finish(task: Task):
        st = task.state
        with pending_lock:
            watched.discard(task)
            inflight[0] -= 1
        if st == TaskState.OK:
            pass
        elif st == TaskState.LOST:
            # resubmit: deps may also be lost; enqueue re-walks
            enqueue(task)
        elif st == TaskState.ERR:
            note_error(task.error or RuntimeError(f"{task.name} failed"))
        with pending_lock:
            if inflight[0] == 0:
                done_event.set()

    for r in roots:
        enqueue(r)
    while True:
        if all(r.state == TaskState.OK for r in roots):
            return
        done_event.wait(timeout=0.05)
        with pending_lock:
            if errors:
                raise errors[0]
            if inflight[0] == 0:
                if all(r.state == TaskState.OK for r in roots):
                    return
                # quiescent but incomplete: re-enqueue (lost deps)
                done_event.clear()
                needs = [r for r in roots if r.state != TaskState.OK]
            else:
                done_event.clear()
                continue
        for r in needs:
            enqueue(r)
