"""Tasks: the schedulable unit (one shard of one fused pipeline phase).

Role-parity: exec/task.go — Task{Name, Do, Deps, Partitioner, NumPartition,
Combiner, Group, state machine INIT->WAITING->RUNNING->OK/ERR/LOST with
broadcast waiters (task.go:41-86, :325-418)}.
"""

from __future__ import annotations

import enum
import threading
from typing import Callable, List, Optional, Sequence


class TaskState(enum.IntEnum):
    INIT = 0
    WAITING = 1
    RUNNING = 2
    OK = 3
    ERR = 4
    LOST = 5


class TaskDep:
    """Dependency on a phase of producer tasks (task.go:91-128).

    head_tasks: the tasks of the producing phase (all shards).
    partition: which partition of the producers' output this task reads.
    expand: hand each producer stream to the consumer unmerged.
    combiner: the consumer's combiner (producers pre-combine partitions).
    """

    __slots__ = ("head_tasks", "partition", "expand", "combiner")

    def __init__(self, head_tasks: Sequence["Task"], partition: int,
                 expand: bool = False, combiner=None):
        # alias list inputs: all consumers of a phase then share ONE
        # head list, letting the evaluator dedupe phase scans by list
        # identity (O(tasks+heads) bookkeeping instead of
        # O(consumers x heads); callers pass completed phase lists)
        self.head_tasks = (head_tasks if isinstance(head_tasks, list)
                           else list(head_tasks))
        self.partition = partition
        self.expand = expand
        self.combiner = combiner


class Task:
    """One shard of a compiled phase."""

    def __init__(self, name: str, invocation_index: int,
                 do: Callable, deps: List[TaskDep],
                 num_partitions: int = 1, partitioner=None,
                 combiner=None, shuffle_out: bool = False,
                 group: Optional[List["Task"]] = None,
                 num_out_columns: int = 1, pragma=None,
                 schema=None, shard: int = 0, num_shards: int = 1):
        self.name = name
        self.invocation_index = invocation_index
        self.do = do  # do(dep_readers, ctx) -> sliceio.Reader
        self.deps = deps
        self.num_partitions = num_partitions
        self.partitioner = partitioner
        self.combiner = combiner
        self.shuffle_out = shuffle_out
        self.group: List[Task] = group if group is not None else [self]
        self.num_out_columns = num_out_columns
        self.pragma = pragma
        self.schema = schema
        self.shard = shard
        self.num_shards = num_shards

        self._state = TaskState.INIT
        self._err: Optional[BaseException] = None
        self._cond = threading.Condition()
        self.consecutive_lost = 0

    # -- state machine ----------------------------------------------------

    @property
    def state(self) -> TaskState:
        # Lock-free: a single attribute read is atomic under the GIL
        # and the value is an immutable enum.  Writers still lock (the
        # condition ordering matters only for wait_state, which takes
        # the lock).  State reads are the evaluator's hottest call
        # (~3 per task per schedule pass); the lock here contended
        # directly with every set_state notify.  Caveat: on a
        # free-threaded (PEP 703, no-GIL) build the err-before-state
        # store ordering below is not a synchronization edge; if this
        # ever runs under such a build, read state/error via the cond
        # lock instead.
        return self._state

    @property
    def error(self) -> Optional[BaseException]:
        return self._err

    def set_state(self, s: TaskState, err: BaseException = None) -> None:
        with self._cond:
            if s == TaskState.ERR:
                self._err = err  # published before state for lock-free readers
            self._state = s
            if s == TaskState.OK:
                self.consecutive_lost = 0
            elif s == TaskState.LOST:
                self.consecutive_lost += 1
            self._cond.notify_all()

    def wait_state(self, min_state: TaskState,
                   timeout: Optional[float] = None) -> TaskState:
        """Block until state >= min_state (task.go WaitState)."""
        with self._cond:
            self._cond.wait_for(lambda: self._state >= min_state,
                                timeout=timeout)
            return self._state

    def __repr__(self):
        return f"Task({self.name}, {self.state.name})"


def phase_of(task: Task) -> List[Task]:
    """All sibling tasks of the task's phase (task.go Group semantics)."""
    return task.group


def all_deps_ok(task: Task) -> bool:
    return all(t.state == TaskState.OK
               for dep in task.deps for t in dep.head_tasks)


def graph_string(roots: Sequence[Task], detail: bool = False) -> str:
    """Debug rendering of the task graph (task.go GraphString), used by
    compiler golden tests.  detail=True annotates each first visit with
    the shuffle/partition/combiner attributes so golden diffs catch
    fusion, naming, partitioning and combine-spec regressions
    (exec/compile_test.go:23-137)."""
    lines: List[str] = []
    seen = set()

    def annot(t: Task) -> str:
        if not detail:
            return ""
        bits = []
        if t.shuffle_out:
            bits.append(f"shuffle nparts={t.num_partitions}")
        if t.combiner is not None:
            bits.append(f"comb={getattr(t.combiner, 'name', '?')}")
        if t.pragma is not None and getattr(t.pragma, "materialize",
                                            False):
            bits.append("materialize")
        return f"  [{' '.join(bits)}]" if bits else ""

    def visit(t: Task, depth: int):
        if t.name in seen:
            lines.append("  " * depth + t.name)
            return
        seen.add(t.name)
        lines.append("  " * depth + t.name + annot(t))
        for dep in t.deps:
            for h in dep.head_tasks:
                visit(h, depth + 1)

    for r in roots:
        visit(r, 0)
    return "\n".join(lines)
