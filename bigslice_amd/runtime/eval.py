"""Evaluator: schedules the task graph to an executor.

Role-parity: exec/eval.go:80-176 — enqueue roots, run runnable tasks
concurrently, resubmit LOST tasks together with their now-missing
dependencies (:112-115, :352-376), give up after maxConsecutiveLost
(:30, :139-159).  Multiple concurrent Evals coordinate through task
state: a task already WAITING/RUNNING is watched, not re-run.

Implementation: callback-driven over a persistent thread pool with the
reference's phase-aware dependency counting (exec/eval.go:255-451): a
shuffle consumer depends on the producer PHASE, so bookkeeping is
O(tasks), not O(edges) — a phase keeps one remaining-task counter and
one consumer list, and the last task of a phase wakes all consumers.
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
    seen_lists: Set[int] = set()  # consumers share per-phase head lists
    stack: List[Task] = []
    for t in roots:
        if id(t) not in seen:
            seen.add(id(t))
            stack.append(t)
    while stack:
        t = stack.pop()
        out.append(t)
        for dep in t.deps:
            heads = dep.head_tasks
            if id(heads) in seen_lists:
                continue
            seen_lists.add(id(heads))
            for h in heads:
                if id(h) not in seen:
                    seen.add(id(h))
                    stack.append(h)
    return out


def evaluate(executor: Executor, roots: Sequence[Task]) -> None:
    """Evaluate all root tasks to OK (or raise)."""
    tasks = _reachable(roots)

    def gid(t: Task) -> int:
        return id(t.group[0])

    # Phase bookkeeping over reachable tasks (exec/eval.go state):
    phase_tasks: Dict[int, List[Task]] = {}
    for t in tasks:
        phase_tasks.setdefault(gid(t), []).append(t)
    # consumers of each phase (tasks with a dep whose heads are in it).
    # Consumers share per-phase head LISTS, so the group scan runs once
    # per unique list (O(tasks + heads)), not per consumer x heads.
    phase_consumers: Dict[int, List[Task]] = {}
    dep_groups: Dict[int, List[int]] = {}  # task -> dep phase ids
    list_groups: Dict[int, List[int]] = {}  # id(head list) -> group ids

    def groups_of(heads) -> List[int]:
        got = list_groups.get(id(heads))
        if got is None:
            got = []
            sg = set()
            # heads normally share one group; scan defensively for
            # hand-built graphs (tests) with ungrouped heads
            for h in heads:
                g = gid(h)
                if g not in sg:
                    sg.add(g)
                    got.append(g)
            list_groups[id(heads)] = got
        return got

    for t in tasks:
        groups = []
        seen_g = set()
        for dep in t.deps:
            for g in groups_of(dep.head_tasks):
                if g not in seen_g:
                    seen_g.add(g)
                    groups.append(g)
                    phase_consumers.setdefault(g, []).append(t)
        dep_groups[id(t)] = groups

    lock = threading.Lock()
    done = threading.Condition(lock)
    errors: List[BaseException] = []
    enqueued: Set[int] = set()   # visited, not yet finalized
    ok_counted: Set[int] = set()  # tasks currently counted complete
    walked_groups: Set[int] = set()  # phases whose members were walked
    phase_remaining: Dict[int, int] = {}
    for g, members in phase_tasks.items():
        rem = 0
        for m in members:
            if m.state == TaskState.OK:
                ok_counted.add(id(m))
            else:
                rem += 1
        phase_remaining[g] = rem

    # Driver wakeups are counter-based: notify only when the last root
    # completes (or on error/lost), not on every task finish — at 5000
    # shards the per-finish notify_all was the scheduler's top cost
    # (one main-thread wakeup + O(roots) rescan per task).  The 1 s
    # timed wait below remains the backstop for quiescent re-walks and
    # state flipped by concurrent evaluations.
    root_ids: Set[int] = {id(r) for r in roots}
    roots_left = sum(1 for i in root_ids if i not in ok_counted)

    pool = getattr(executor, "pool", None)
    own_pool = None
    if pool is None:
        own_pool = ThreadPoolExecutor(
            max_workers=getattr(executor, "parallelism", None)
            or config.DEFAULT_PARALLELISM)
        pool = own_pool

    def deps_ready(t: Task) -> bool:
        return all(phase_remaining.get(g, 0) == 0
                   for g in dep_groups[id(t)])

    def visit(t: Task):
        """Lock held: walk t once; start if its dep phases are complete,
        else walk incomplete dep tasks (their completion wakes t)."""
        if id(t) in enqueued:
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
        enqueued.add(id(t))
        if st in (TaskState.WAITING, TaskState.RUNNING):
            # claimed by a concurrent evaluation: watch it
            threading.Thread(target=_watch, args=(t,),
                             daemon=True).start()
            return
        if deps_ready(t):
            start(t)
        else:
            # walk each incomplete dep PHASE once (O(tasks), not
            # O(consumers x heads))
            for g in dep_groups[id(t)]:
                if phase_remaining.get(g, 0) == 0 or g in walked_groups:
                    continue
                walked_groups.add(g)
                for h in phase_tasks.get(g, ()):
                    if h.state != TaskState.OK:
                        note_incomplete(h)
                        visit(h)

    def note_incomplete(h: Task):
        """A task previously counted OK went LOST/stale: restore its
        phase counter so consumers wait again."""
        nonlocal roots_left
        if id(h) in ok_counted and h.state != TaskState.OK:
            ok_counted.discard(id(h))
            phase_remaining[gid(h)] = phase_remaining.get(gid(h), 0) + 1
            if id(h) in root_ids:
                roots_left += 1

    # Inline mode (executor.inline): tasks run on THIS thread from a
    # deferred queue instead of pool workers — the per-task cond-var
    # wake/park round trips (measured ~3.5 ms/step at the 125M bench
    # scale) disappear.  Correct only because start() callers hold the
    # lock: the queue defers execution to the drain loop outside it.
    inline = bool(getattr(executor, "inline", False))
    inline_queue: List[Task] = []

    def start(t: Task):
        t.set_state(TaskState.WAITING)
        if inline:
            inline_queue.append(t)
        else:
            pool.submit(_run, t)

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
        nonlocal roots_left
        st = t.state
        with lock:
            if st == TaskState.OK:
                enqueued.discard(id(t))
                if id(t) not in ok_counted:
                    ok_counted.add(id(t))
                    g = gid(t)
                    phase_remaining[g] -= 1
                    if phase_remaining[g] == 0:
                        for c in phase_consumers.get(g, ()):
                            if id(c) in enqueued and c.state in (
                                    TaskState.INIT, TaskState.LOST) and \
                                    deps_ready(c):
                                start(c)
                    if id(t) in root_ids:
                        roots_left -= 1
                if id(t) in root_ids and roots_left <= 0:
                    # notify outside the freshly-counted check too: a
                    # root re-finishing after an external LOST/re-run
                    # may still be in ok_counted.  A root flipped to OK
                    # entirely outside this evaluator's callbacks (a
                    # concurrent evaluate sharing the task) is only
                    # picked up by the 1s timed-wait backstop below —
                    # up to ~1s extra latency on that rare path, traded
                    # for keeping completion notification lock-free.
                    done.notify_all()
            elif st == TaskState.LOST:
                # resubmit, re-walking deps that were also lost:
                # reconcile phase counters for dep members whose OK was
                # revoked (marked LOST by the failed run)
                enqueued.discard(id(t))
                note_incomplete(t)
                for g in dep_groups[id(t)]:
                    walked_groups.discard(g)
                    for h in phase_tasks.get(g, ()):
                        note_incomplete(h)
                visit(t)
                done.notify_all()
            else:
                errors.append(t.error or RuntimeError(f"{t.name} failed"))
                done.notify_all()

    try:
        with lock:
            for r in roots:
                visit(r)
            while not errors:
                if inline and inline_queue:
                    # drain deferred inline tasks with the lock
                    # RELEASED around execution (the task's _finish
                    # re-acquires it and may queue successors)
                    t = inline_queue.pop()
                    lock.release()
                    try:
                        _run(t)
                    finally:
                        lock.acquire()
                    continue
                if all(r.state == TaskState.OK for r in roots):
                    return
                if not enqueued:
                    # quiescent but incomplete: re-walk from roots
                    # (e.g. interior tasks lost after completion)
                    for r in roots:
                        if r.state != TaskState.OK:
                            visit(r)
                    if errors:
                        break
                    if all(r.state == TaskState.OK for r in roots):
                        return
                    if inline and inline_queue:
                        continue
                done.wait(timeout=1.0)
            raise errors[0]
    finally:
        if own_pool is not None:
            own_pool.shutdown(wait=False)
