"""Chrome-trace-format event tracer.

Role-parity: exec/tracer.go + internal/trace — B/E task events coalesced
to X complete events, one "pid" per worker (rank), written as Chrome
trace JSON viewable in chrome://tracing / Perfetto.  GPU kernel-level
detail comes from rocprofv3 (profiles/); this tracer covers the
task/phase control plane.
"""

from __future__ import annotations

import json
import threading
import time
from typing import List


class Tracer:
    def __init__(self):
        self.events: List[dict] = []
        self._lock = threading.Lock()
        self._t0 = time.perf_counter()

    def _us(self) -> float:
        return (time.perf_counter() - self._t0) * 1e6

    def emit(self, name: str, ph: str, pid: int = 0, tid: int = 0,
             args: dict = None, ts: float = None):
        ev = {"name": name, "ph": ph, "pid": pid, "tid": tid,
              "ts": self._us() if ts is None else ts}
        if args:
            ev["args"] = args
        with self._lock:
            self.events.append(ev)

    def span(self, name: str, pid: int = 0, tid: int = None,
             args: dict = None):
        return _Span(self, name, pid,
                     threading.get_ident() % 10000 if tid is None else tid,
                     args)

    def write(self, path: str):
        with self._lock:
            evs = list(self.events)
        with open(path, "w") as fp:
            json.dump({"traceEvents": evs, "displayTimeUnit": "ms"}, fp)


class _Span:
    def __init__(self, tracer, name, pid, tid, args):
        self.tracer = tracer
        self.name, self.pid, self.tid, self.args = name, pid, tid, args

    def __enter__(self):
        self.start = self.tracer._us()
        return self

    def __exit__(self, *exc):
        self.tracer.events.append({
            "name": self.name, "ph": "X", "pid": self.pid,
            "tid": self.tid, "ts": self.start,
            "dur": self.tracer._us() - self.start,
            **({"args": self.args} if self.args else {})})
        return False
