"""Offline trace analyzer (reference cmd/slicetrace): per-op duration
quartile summaries from a Chrome trace file written by the session
tracer (trace_path=...).

  python -m bigslice_amd.tools.slicetrace trace.json
"""

from __future__ import annotations

import json
import re
import sys
from collections import defaultdict


def quartiles(xs):
    xs = sorted(xs)
    n = len(xs)

    def q(p):
        i = min(int(p * (n - 1)), n - 1)
        return xs[i]
    return q(0.25), q(0.5), q(0.75)


def analyze(path: str):
    with open(path) as fp:
        data = json.load(fp)
    per_op = defaultdict(list)
    for ev in data.get("traceEvents", []):
        if ev.get("ph") != "X":
            continue
        # task names look like inv1_const_map@8:3 -> op key inv1_const_map
        op = re.sub(r"@\d+:\d+$", "", ev["name"])
        per_op[op].append(ev.get("dur", 0) / 1000.0)
    print(f"{'op':40s} {'n':>5s} {'p25':>9s} {'p50':>9s} {'p75':>9s} "
          f"{'total':>10s}")
    for op, durs in sorted(per_op.items(),
                           key=lambda kv: -sum(kv[1])):
        q1, q2, q3 = quartiles(durs)
        print(f"{op:40s} {len(durs):5d} {q1:8.2f}m {q2:8.2f}m "
              f"{q3:8.2f}m {sum(durs):9.2f}m")


def main():
    if len(sys.argv) != 2:
        print(__doc__)
        sys.exit(2)
    analyze(sys.argv[1])


if __name__ == "__main__":
    main()
