"""User-facing metrics: counters aggregated across tasks.

Role-parity: metrics/metrics.go + metrics/scope.go — globally registered
counters, per-task Scopes merged across the task graph into the driver's
Result (exec/bigmachine.go:688-695, exec/session.go:418-426).  The active
scope rides a thread-local (the reference carries it in ctx).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

_counters: List["Counter"] = []
_lock = threading.Lock()
_tls = threading.local()


class Counter:
    """A named integer counter (metrics.go:58-93)."""

    def __init__(self, name: str):
        self.name = name
        with _lock:
            self.index = len(_counters)
            _counters.append(self)

    def incr(self, n: int = 1):
        scope = current_scope()
        if scope is not None:
            scope.incr(self, n)

    def value(self, scope: "Scope") -> int:
        return scope.values.get(self.index, 0)


def counter(name: str) -> Counter:
    return Counter(name)


class Scope:
    """A set of metric instances (scope.go:17-135)."""

    def __init__(self):
        self.values: Dict[int, int] = {}
        self._lock = threading.Lock()

    def incr(self, c: Counter, n: int):
        with self._lock:
            self.values[c.index] = self.values.get(c.index, 0) + n

    def merge(self, other: "Scope"):
        with self._lock:
            for k, v in other.values.items():
                self.values[k] = self.values.get(k, 0) + v

    def snapshot(self) -> Dict[str, int]:
        with _lock:
            names = {c.index: c.name for c in _counters}
        with self._lock:
            return {names.get(k, f"#{k}"): v
                    for k, v in self.values.items()}

    def to_dict(self) -> Dict[int, int]:
        with self._lock:
            return dict(self.values)

    @staticmethod
    def from_dict(d: Dict[int, int]) -> "Scope":
        s = Scope()
        s.values = {int(k): int(v) for k, v in d.items()}
        return s


def current_scope() -> Optional[Scope]:
    return getattr(_tls, "scope", None)


class scoped:
    """Context manager installing a scope on the current thread."""

    def __init__(self, scope: Scope):
        self.scope = scope

    def __enter__(self):
        self.prev = getattr(_tls, "scope", None)
        _tls.scope = self.scope
        return self.scope

    def __exit__(self, *exc):
        _tls.scope = self.prev
        return False
