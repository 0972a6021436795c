"""Test harness for user code (reference slicetest/run.go, print.go):
run slices in a local session, scan results, deterministic printing."""

from __future__ import annotations

from typing import List

from .runtime.session import FuncValue, func, start


def run(builder_or_func, *args, device: str = "cpu",
        parallelism: int = 2):
    """Run a slice builder in a fresh local session and return the
    Result (slicetest.Run)."""
    fv = builder_or_func if isinstance(builder_or_func, FuncValue) \
        else func(builder_or_func)
    sess = start(parallelism=parallelism, device=device)
    return sess.run(fv, *args)


def run_err(builder_or_func, *args, **kw):
    """Run and return the raised error (slicetest.RunErr), or None."""
    try:
        run(builder_or_func, *args, **kw)
        return None
    except Exception as e:
        return e


def scan_all(builder_or_func, *args, **kw) -> List:
    """Run and return all rows (slicetest.ScanAll/RunAndScan)."""
    return list(run(builder_or_func, *args, **kw).scan())


def print_rows(builder_or_func, *args, **kw):
    """Run and print rows in deterministic (sorted) order
    (slicetest.Print, print.go:20-57)."""
    for row in sorted(scan_all(builder_or_func, *args, **kw), key=repr):
        if isinstance(row, tuple):
            print(" ".join(str(v) for v in row))
        else:
            print(row)
