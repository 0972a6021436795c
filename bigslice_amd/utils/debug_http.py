"""HTTP debug endpoints (reference exec/graph.go + session.go:376-389):
/debug/tasks (states), /debug/tasks/graph (JSON nodes+links),
/debug/trace (Chrome trace JSON), /debug/metrics.
"""

from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import List


def serve_session(session, port: int) -> "ThreadingHTTPServer":
    """Start the debug HTTP server on a background thread."""
    results: List = []
    orig_run = session.run

    def run_hook(*a, **kw):
        res = orig_run(*a, **kw)
        results.append(res)
        return res
    session.run = run_hook

    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def _json(self, obj, code=200):
            body = json.dumps(obj, indent=2, default=str).encode()
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_GET(self):
            if self.path == "/debug/tasks":
                out = []
                for res in results:
                    seen = set()

                    def visit(t):
                        if t.name in seen:
                            return
                        seen.add(t.name)
                        out.append({"task": t.name,
                                    "state": t.state.name})
                        for dep in t.deps:
                            for h in dep.head_tasks:
                                visit(h)
                    for t in res.tasks:
                        visit(t)
                self._json(out)
            elif self.path == "/debug/tasks/graph":
                nodes, links, seen = [], [], set()
                for res in results:
                    def visit(t):
                        if t.name in seen:
                            return
                        seen.add(t.name)
                        nodes.append({"id": t.name,
                                      "state": t.state.name})
                        for dep in t.deps:
                            for h in dep.head_tasks:
                                links.append({"source": h.name,
                                              "target": t.name})
                                visit(h)
                    for t in res.tasks:
                        visit(t)
                self._json({"nodes": nodes, "links": links})
            elif self.path == "/debug/trace":
                tr = session.tracer
                self._json({"traceEvents": tr.events if tr else []})
            elif self.path == "/debug/metrics":
                merged = {}
                for res in results:
                    merged.update(res.scope().snapshot())
                self._json(merged)
            else:
                self._json({"endpoints": ["/debug/tasks",
                                          "/debug/tasks/graph",
                                          "/debug/trace",
                                          "/debug/metrics"]}, 404)

    server = ThreadingHTTPServer(("127.0.0.1", port), Handler)
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    session._debug_server = server
    return server
