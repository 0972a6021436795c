"""Kernel micro-benchmarks (for rocprofv3/PMC runs and A/B timing).

  python -m bigslice_amd.tools.microprof groupby --rows 125000000
  python -m bigslice_amd.tools.microprof partition --rows 125000000
  python -m bigslice_amd.tools.microprof hash --rows 125000000

Times each kernel phase with hipEvents over --iters iterations.  Keep
runs short: this is the target for `rocprofv3 --pmc ... -- ...`.
"""

from __future__ import annotations

import argparse
import json
import os

import torch

from bigslice_amd import kernels
from bigslice_amd.frame import Frame


def timeit(fn, iters):
    """Median of per-iteration times (2 warmups).  Median, not mean:
    on leases where a prior process ran, the FIRST touch of each
    freshly-committed GPU allocation pays a one-time ~40-80 ms
    driver-side cost (measured: iteration k hits it when the caching
    allocator grows, all later iterations are clean) — a mean smears
    that one-time cost into a phantom 3-4x 'regression'."""
    import time
    for _ in range(2):
        fn()  # warmups absorb allocator-growth first-touch costs
    torch.cuda.synchronize()
    ts = []
    for _ in range(max(iters, 3)):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1000)
    ts.sort()
    return ts[len(ts) // 2]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("which", choices=["groupby", "groupby_pp",
                                      "insert", "insert_rep",
                                      "partition", "hash", "compact",
                                      "sortcombine", "sortcombine2", "hashbytes",
                                      "runs"])
    ap.add_argument("--nrep", type=int, default=8)
    ap.add_argument("--cap", type=int, default=0,
                    help="table capacity (0 = 2x nkeys rounded up)")
    ap.add_argument("--rows", type=int, default=125_000_000)
    ap.add_argument("--nkeys", type=int, default=1_000_000)
    ap.add_argument("--nparts", type=int, default=8)
    ap.add_argument("--iters", type=int, default=3)
    args = ap.parse_args()
    assert torch.cuda.is_available() and kernels.have_extension()
    dev = "cuda:0"
    g = torch.Generator(device=dev)
    g.manual_seed(1)
    keys = torch.randint(0, args.nkeys, (args.rows,), dtype=torch.int64,
                         device=dev, generator=g)
    vals = torch.ones(args.rows, dtype=torch.int64, device=dev)

    out = {"which": args.which, "rows": args.rows, "nkeys": args.nkeys,
           "packed": os.environ.get("BIGSLICE_GB_PACKED", "1")}
    if args.which == "groupby":
        def run():
            t = kernels.GroupTable([torch.int64], ["sum"], dev)
            t.insert(keys, [vals])
            t.finish()
        ms = timeit(run, args.iters)
    elif args.which == "groupby_pp":
        # partition-first: scatter rows by table-slot range, then insert
        # bucket-by-bucket so atomics have TCC temporal locality
        P = int(os.environ.get("PP_PARTS", "64"))

        def run():
            t = kernels.GroupTable([torch.int64], ["sum"], dev)
            from bigslice_amd import config as cfg
            cap = 1024
            while cap < min(2 * keys.shape[0], cfg.GROUPBY_INITIAL_CAP):
                cap <<= 1
            t._alloc(cap)
            pids = kernels._C.slot_pids(keys, cap, P)
            (rk, rv), counts = kernels._C.scatter_by_partition(
                [keys, vals], pids, P)
            off = 0
            t.rows = keys.shape[0]
            t._mode = "global"
            t.batches.append((keys, [vals]))
            for c in counts.tolist():
                if c:
                    kernels._C.groupby_insert(
                        rk[off:off + c], [rv[off:off + c]], t.codes,
                        t.tkeys, t.tabs, t.flags, kernels.MAX_PROBES)
                off += c
            t.finish()
        ms = timeit(run, args.iters)
    elif args.which in ("insert", "insert_rep"):
        # steady-state insert throughput into a presized packed table
        # (all keys present after warmup: pure probe+atomicAdd path);
        # insert_rep = the sub-table replication experiment
        cap = args.cap
        if not cap:
            cap = 1024
            while cap < 2 * args.nkeys:
                cap <<= 1
        nrep = args.nrep if args.which == "insert_rep" else 1
        table = kernels._C.alloc_packed_table(
            cap * nrep + nrep - 1,  # nrep*(cap+1) slots
            torch.empty(0, dtype=torch.int64, device=dev))
        flags = torch.zeros(2, dtype=torch.int32, device=dev)
        out["cap"] = cap
        out["nrep"] = nrep

        def run():
            if nrep > 1:
                kernels._C.groupby_insert_packed_rep(
                    keys, vals, table, nrep, flags, 4096)
            else:
                kernels._C.groupby_insert_packed(
                    keys, vals, table, flags, 4096)
        ms = timeit(run, args.iters)
        assert int(flags[1].item()) == 0, "table overflow"
    elif args.which == "compact":
        t = kernels.GroupTable([torch.int64], ["sum"], dev)
        t.insert(keys, [vals])

        def run():
            cursor = torch.zeros(1, dtype=torch.int64, device=dev)
            if t._packed:
                kernels._C.groupby_compact_packed(t.table, cursor)
            else:
                kernels._C.groupby_compact(t.tkeys, t.tabs, cursor)
        ms = timeit(run, args.iters)
    elif args.which == "runs":
        # K18 run-boundary compaction over sorted keys (PMC target)
        sk = torch.sort(keys).values.contiguous()

        def run():
            kernels._C.runs_sorted(sk)
        ms = timeit(run, args.iters)
    elif args.which == "partition":
        f = Frame([keys, vals], prefix=1)

        def run():
            kernels.partition_frame(f, args.nparts, None)
        ms = timeit(run, args.iters)
    elif args.which == "hash":
        def run():
            kernels.hash_columns_device([keys], 0)
        ms = timeit(run, args.iters)
    else:
        ms = None
    if ms is not None:
        out["ms"] = ms
        out["grows_per_sec"] = args.rows / ms / 1e6
    if args.which == "sortcombine":
        from bigslice_amd.kernels import _C

        def run():
            ks, vs = _C.radix_sort_kv(keys, vals)
            uq, sm, cnt = _C.segment_reduce_sorted(ks, vs, 0)
            int(cnt.item())
        out["ms"] = timeit(run, args.iters)
        out["grows_per_s"] = args.rows / out["ms"] / 1e6
    if args.which == "sortcombine2":
        # sort + one k_runs boundary pass + thread-per-run segsum
        from bigslice_amd.kernels import _C

        def run():
            ks, vs = _C.radix_sort_kv(keys, vals)
            uq, starts, cnt = _C.runs_sorted(ks)
            g = _C.runs_guard(starts, cnt, ks.shape[0]).cpu()
            _C.segment_reduce_runs(vs, starts[:int(g[0])],
                                   ks.shape[0], 0)
        out["ms"] = timeit(run, args.iters)
        out["grows_per_s"] = args.rows / out["ms"] / 1e6
    if args.which == "hashbytes":
        import random

        from bigslice_amd import strings
        rng = random.Random(2)
        words = ["w%05d" % rng.randrange(args.nkeys)
                 for _ in range(min(args.rows, 10_000_000))]
        data, offs = strings.pack_strings(words)
        b, o = strings.to_device(data, offs, "cuda:0")
        from bigslice_amd.kernels import _C

        def run():
            _C.hash_bytes64(b, o, 0x9ACB0442, 0x85EBCA6B)
        out["ms"] = timeit(run, args.iters)
        out["rows"] = len(words)
        out["mwords_per_s"] = len(words) / out["ms"] / 1e3
    print(json.dumps(out))


if __name__ == "__main__":
    main()
