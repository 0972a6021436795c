"""Evaluator tests with fake executors (reference exec/eval_test.go:
error propagation, lost-task resubmission, TooManyTries, multi-phase
scheduling, random-loss stress)."""

import random
import threading

import pytest

from bigslice_amd import config
from bigslice_amd.runtime.eval import (Executor, TooManyTriesError,
                                       evaluate)
from bigslice_amd.runtime.task import Task, TaskDep, TaskState


def make_task(name, deps=(), group=None):
    return Task(name=name, invocation_index=1,
                do=lambda dr, ctx: None, deps=list(deps),
                group=group)


class FakeExecutor(Executor):
    """Runs tasks instantly; per-task behavior configurable."""

    def __init__(self):
        self.runs = {}
        self.behavior = {}  # name -> list of states to emit per attempt
        self.lock = threading.Lock()

    def run(self, task):
        with self.lock:
            self.runs[task.name] = self.runs.get(task.name, 0) + 1
            plan = self.behavior.get(task.name)
            if plan:
                st = plan.pop(0)
            else:
                st = TaskState.OK
        task.set_state(TaskState.RUNNING)
        if st == TaskState.ERR:
            task.set_state(TaskState.ERR, RuntimeError(f"{task.name} boom"))
        else:
            task.set_state(st)

    def reader(self, task, partition):
        raise NotImplementedError


def chain(n):
    """t0 <- t1 <- ... (t0 is root)."""
    tasks = []
    prev = None
    for i in reversed(range(n)):
        deps = [TaskDep([prev], 0)] if prev is not None else []
        t = make_task(f"t{i}", deps)
        prev = t
        tasks.append(t)
    return list(reversed(tasks))  # [root, ..., leaf]


def test_simple_chain():
    tasks = chain(3)
    ex = FakeExecutor()
    evaluate(ex, [tasks[0]])
    assert all(t.state == TaskState.OK for t in tasks)
    assert ex.runs == {"t0": 1, "t1": 1, "t2": 1}


def test_error_propagates():
    tasks = chain(2)
    ex = FakeExecutor()
    ex.behavior["t1"] = [TaskState.ERR]
    with pytest.raises(RuntimeError, match="t1 boom"):
        evaluate(ex, [tasks[0]])


def test_lost_task_resubmitted():
    tasks = chain(2)
    ex = FakeExecutor()
    ex.behavior["t1"] = [TaskState.LOST, TaskState.LOST]
    evaluate(ex, [tasks[0]])
    assert tasks[0].state == TaskState.OK
    assert ex.runs["t1"] == 3


def test_interior_loss_recomputes_deps():
    # root depends on mid depends on leaf; mid reports LOST after leaf OK:
    # the evaluator must re-run mid (and leaf results are still OK).
    tasks = chain(3)
    ex = FakeExecutor()
    ex.behavior["t1"] = [TaskState.LOST]
    evaluate(ex, [tasks[0]])
    assert tasks[0].state == TaskState.OK
    assert ex.runs["t1"] == 2


def test_too_many_tries():
    tasks = chain(1)
    ex = FakeExecutor()
    ex.behavior["t0"] = [TaskState.LOST] * (config.MAX_CONSECUTIVE_LOST + 2)
    with pytest.raises(TooManyTriesError):
        evaluate(ex, [tasks[0]])


def test_multiphase_fanin():
    # 4 producers feeding 2 consumers (a shuffle phase shape).
    producers = [make_task(f"p{i}") for i in range(4)]
    consumers = [make_task(f"c{i}", [TaskDep(producers, i)])
                 for i in range(2)]
    ex = FakeExecutor()
    evaluate(ex, consumers)
    assert all(t.state == TaskState.OK for t in producers + consumers)


def test_stress_random_loss():
    # evalstress_test.go analog: random losses still converge.
    rng = random.Random(42)
    producers = [make_task(f"p{i}") for i in range(8)]
    mids = [make_task(f"m{i}", [TaskDep(producers, i)]) for i in range(4)]
    roots = [make_task(f"r{i}", [TaskDep(mids, i)]) for i in range(2)]

    class LossyExecutor(FakeExecutor):
        def run(self, task):
            with self.lock:
                self.runs[task.name] = self.runs.get(task.name, 0) + 1
            task.set_state(TaskState.RUNNING)
            # never exceed the consecutive-lost bound
            if task.consecutive_lost < 2 and rng.random() < 0.3:
                task.set_state(TaskState.LOST)
            else:
                task.set_state(TaskState.OK)

    evaluate(LossyExecutor(), roots)
    assert all(t.state == TaskState.OK for t in roots)


def test_stress_random_loss_inline():
    """The same convergence under the INLINE drain loop (the GPU
    default): losses requeue on the evaluating thread with no pool."""
    rng = random.Random(7)
    producers = [make_task(f"ip{i}") for i in range(8)]
    mids = [make_task(f"im{i}", [TaskDep(producers, i)])
            for i in range(4)]
    roots = [make_task(f"ir{i}", [TaskDep(mids, i)]) for i in range(2)]

    class LossyInline(FakeExecutor):
        inline = True

        def run(self, task):
            with self.lock:
                self.runs[task.name] = self.runs.get(task.name, 0) + 1
            task.set_state(TaskState.RUNNING)
            if task.consecutive_lost < 2 and rng.random() < 0.35:
                task.set_state(TaskState.LOST)
            else:
                task.set_state(TaskState.OK)

    ex = LossyInline()
    evaluate(ex, roots)
    assert all(t.state == TaskState.OK for t in roots)
    # everything ran on this thread: more runs than tasks (losses)
    assert sum(ex.runs.values()) >= 14


def test_concurrent_evals_share_tasks():
    # Two Evals over the same graph coordinate through task state.
    tasks = chain(3)
    ex = FakeExecutor()
    errs = []

    def go():
        try:
            evaluate(ex, [tasks[0]])
        except BaseException as e:  # pragma: no cover
            errs.append(e)
    ts = [threading.Thread(target=go) for _ in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs
    assert tasks[0].state == TaskState.OK


def test_roots_already_ok_no_rerun():
    # All tasks OK at entry: evaluate returns without running anything
    # (counter-based root accounting must handle pre-counted roots).
    tasks = chain(3)
    for t in tasks:
        t.set_state(TaskState.OK)
    ex = FakeExecutor()
    evaluate(ex, [tasks[0]])
    assert ex.runs == {}


def test_root_revoked_ok_recounted():
    # r1 (a root) depends on r0 (also a root).  r1's first attempt
    # discovers r0's output missing: it marks r0 LOST and reports
    # itself LOST.  Both must re-run and evaluate must complete
    # promptly (roots_left reconciliation in note_incomplete).
    r0 = make_task("r0")
    r1 = make_task("r1", [TaskDep([r0], 0)])

    class RevokingExecutor(FakeExecutor):
        def run(self, task):
            with self.lock:
                self.runs[task.name] = self.runs.get(task.name, 0) + 1
                first_r1 = task.name == "r1" and self.runs["r1"] == 1
            task.set_state(TaskState.RUNNING)
            if first_r1:
                r0.set_state(TaskState.LOST)  # dep output vanished
                task.set_state(TaskState.LOST)
            else:
                task.set_state(TaskState.OK)

    evaluate(RevokingExecutor(), [r0, r1])
    assert r0.state == TaskState.OK and r1.state == TaskState.OK
