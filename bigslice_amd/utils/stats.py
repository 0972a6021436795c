"""Internal named atomic counters (reference stats/stats.go): task
read/write rows and durations, shuffle bytes, spill counts."""

from __future__ import annotations

import threading
from typing import Dict


class Map:
    def __init__(self):
        self._vals: Dict[str, int] = {}
        self._lock = threading.Lock()

    def add(self, name: str, n: int = 1):
        with self._lock:
            self._vals[name] = self._vals.get(name, 0) + n

    def values(self) -> Dict[str, int]:
        with self._lock:
            return dict(self._vals)

    def merge(self, other: Dict[str, int]):
        with self._lock:
            for k, v in other.items():
                self._vals[k] = self._vals.get(k, 0) + v

    def __str__(self):
        return " ".join(f"{k}={v}"
                        for k, v in sorted(self.values().items()))


DEFAULT = Map()
