"""Compare fresh benchmark JSON lines against the recorded numbers in
profiles/ and flag regressions.

  python -m bigslice_amd.tools.benchdiff NEW.json [NEW2.json ...] \
      [--baseline-dir profiles] [--threshold 0.15]

Matching is by (metric, config.model, config.rows_total); exits 1 when
any matched metric regressed more than the threshold.
"""

from __future__ import annotations

import argparse
import json
import os
import sys


def _load(path):
    out = []
    with open(path) as fp:
        for line in fp:
            line = line.strip()
            if line.startswith("{"):
                try:
                    out.append(json.loads(line))
                except json.JSONDecodeError:
                    pass
    return out


def _key(d):
    c = d.get("config", {})
    return (d.get("metric"), c.get("model"), c.get("rows_total"))


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("new", nargs="+")
    ap.add_argument("--baseline-dir", default="profiles")
    ap.add_argument("--threshold", type=float, default=0.15)
    args = ap.parse_args(argv)

    base = {}
    for fn in sorted(os.listdir(args.baseline_dir)):
        if fn.endswith(".json"):
            for d in _load(os.path.join(args.baseline_dir, fn)):
                base[_key(d)] = d

    worst = 0.0
    rc = 0
    for path in args.new:
        for d in _load(path):
            ref = base.get(_key(d))
            if ref is None:
                print(f"  new      {d['metric'][:50]:<52} "
                      f"{d['value']:.3g} {d['unit']}")
                continue
            hib = d.get("higher_is_better", True)
            ratio = (d["value"] / ref["value"] if hib
                     else ref["value"] / d["value"])
            tag = "ok"
            if ratio < 1 - args.threshold:
                tag = "REGRESSION"
                rc = 1
            elif ratio > 1 + args.threshold:
                tag = "improved"
            worst = min(worst or ratio, ratio)
            print(f"  {tag:<10} {d['metric'][:46]:<48} "
                  f"{ref['value']:.4g} -> {d['value']:.4g} "
                  f"({(ratio - 1) * 100:+.1f}%)")
    return rc


if __name__ == "__main__":
    sys.exit(main())
