"""Live status rollup (reference base/status + exec/slicestatus.go):
task-state counts per invocation, with an optional TTY ticker thread.
"""

from __future__ import annotations

import sys
import threading
from collections import Counter
from typing import Dict, Sequence

from ..runtime.task import Task


def rollup(roots: Sequence[Task]) -> Dict[str, Dict[str, int]]:
    """Per-phase task-state counts (slicestatus.go:84-178 rollup)."""
    seen = set()
    out: Dict[str, Counter] = {}

    def visit(t: Task):
        if id(t) in seen:
            return
        seen.add(id(t))
        base = t.name.rsplit(":", 1)[0]
        out.setdefault(base, Counter())[t.state.name] += 1
        for dep in t.deps:
            for h in dep.head_tasks:
                visit(h)
    for r in roots:
        visit(r)
    return {k: dict(v) for k, v in out.items()}


def format_status(roots: Sequence[Task]) -> str:
    lines = []
    for phase, counts in rollup(roots).items():
        total = sum(counts.values())
        done = counts.get("OK", 0)
        states = " ".join(f"{k}:{v}" for k, v in sorted(counts.items()))
        lines.append(f"{phase:50s} {done}/{total}  {states}")
    return "\n".join(lines)


class Ticker:
    """Background status printer (the reference's TTY status display,
    docs/index.md:240-287).  Use as a context manager around a run."""

    def __init__(self, roots: Sequence[Task], interval: float = 2.0,
                 out=sys.stderr):
        self.roots = roots
        self.interval = interval
        self.out = out
        self._stop = threading.Event()
        self._thread = None

    def __enter__(self):
        def loop():
            while not self._stop.wait(self.interval):
                print(format_status(self.roots), file=self.out)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()
        return self

    def __exit__(self, *exc):
        self._stop.set()
        self._thread.join(timeout=1)
        print(format_status(self.roots), file=self.out)
        return False


def top_n(counts, n: int):
    """Largest-n (key, count) pairs (reference exec/topn.go:36-57's
    diagnostics heap): used by the debug pages to show the heaviest
    task groups / counters."""
    import heapq
    return heapq.nlargest(n, counts.items(), key=lambda kv: (kv[1], kv[0]))
