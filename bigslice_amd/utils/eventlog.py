"""Structured event log (reference base/eventlog usage: sessionStart
with argv, taskComplete with duration; exec/session.go:256-261,
exec/eval.go:160-164).  Events append as JSON lines to a file or a
callback sink."""

from __future__ import annotations

import json
import sys
import threading
import time
from typing import Callable, Optional


class Eventer:
    def __init__(self, path: Optional[str] = None,
                 sink: Optional[Callable[[dict], None]] = None):
        self.path = path
        self.sink = sink
        self._lock = threading.Lock()
        self._fp = open(path, "a") if path else None

    def event(self, typ: str, **fields):
        ev = {"time": time.time(), "event": typ, **fields}
        with self._lock:
            if self._fp is not None:
                self._fp.write(json.dumps(ev) + "\n")
                self._fp.flush()
            if self.sink is not None:
                self.sink(ev)

    def session_start(self):
        self.event("bigslice:sessionStart", argv=sys.argv)

    def task_complete(self, task_name: str, duration_s: float,
                      state: str):
        self.event("bigslice:taskComplete", task=task_name,
                   duration_s=duration_s, state=state)

    def close(self):
        with self._lock:
            if self._fp is not None:
                self._fp.close()
                self._fp = None


NOP = Eventer()
