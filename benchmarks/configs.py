"""Secondary BASELINE.json measurement configs (bench.py covers the
north-star config 3):

  1: Wordcount (Map->Reshuffle->Reduce), local executor, CPU-only
  2: Map+Filter scan over a 100M-row frame, 1 GPU (per-row kernel path)
  4: Cogroup two-slice join, 2 x keyed rows (partitioned sort-merge join)
  5: External sort with spill to host DRAM

Usage: python benchmarks/configs.py --config 2 [--rows N] [--steps K]
Prints one JSON line per run (same shape as bench.py).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import bigslice_amd as bs

_DATA = {}


_WC_LINES = {}


def gen_wc_corpus(nlines: int) -> int:
    """Pre-generate the synthetic corpus (outside the timed region,
    like the tensor configs pre-generate their columns)."""
    import random
    words = ["alpha", "beta", "gamma", "delta", "epsilon", "zeta",
             "eta", "theta"]
    rng = random.Random(1)
    _WC_LINES[nlines] = [
        " ".join(rng.choice(words) for _ in range(8))
        for _ in range(nlines)]
    return nlines


def build_wordcount(nshard, corpus):
    src = bs.ScanReader(nshard, lambda: iter(_WC_LINES[corpus]))

    def split_batch(col):  # vectorized: one call per batch of lines
        out = []
        ext = out.extend
        for s in col:
            ext(s.split())
        return (out,)

    toks = bs.Flatmap(src, split_batch, out_schema=(str,))
    counts = bs.Map(toks, lambda col: (col, [1] * len(col)),
                    out_schema=(str, int))
    return bs.Reduce(counts, "sum")


def build_mapfilter(nshard):
    def gen(shard, ctx):
        x = _DATA[("mf", shard)]
        for off in range(0, x.shape[0], ctx.chunk):
            yield (x[off:off + ctx.chunk],)
    src = bs.ReaderFunc(nshard, gen, bs.schema_of(int))
    mapped = bs.Map(src, lambda x: (x * 3 + 1,))
    return bs.Filter(mapped, lambda x: (x & 7) != 0)


def build_cogroup(nshard):
    def gen_a(shard, ctx):
        yield _DATA[("cga", shard)]

    def gen_b(shard, ctx):
        yield _DATA[("cgb", shard)]
    a = bs.ReaderFunc(nshard, gen_a, bs.schema_of(int, int))
    b = bs.ReaderFunc(nshard, gen_b, bs.schema_of(int, int))
    return bs.Cogroup(a, b)


def build_sort(nshard):
    def gen(shard, ctx):
        entry = _DATA[("sort", shard)]
        if entry[0] == "lazy":
            # very large inputs: generate chunks inside the timed
            # region (EXTRA work, kept honest) so 160+ GB of synthetic
            # input is never resident beside the sorted output.  Plain
            # randint (no explicit Generator: per-thread Generator
            # construction is unreliable under worker streams).
            _, per, hi, seed, device = entry
            for off in range(0, per, ctx.chunk):
                n = min(ctx.chunk, per - off)
                yield (torch.randint(0, hi, (n,), dtype=torch.int64,
                                     device=device),
                       torch.randint(0, 1 << 30, (n,),
                                     dtype=torch.int64, device=device))
            return
        keys, vals = entry
        for off in range(0, keys.shape[0], ctx.chunk):
            yield (keys[off:off + ctx.chunk], vals[off:off + ctx.chunk])
    src = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))

    # terminal external sort per shard via WriterFunc-free custom slice
    class SortSlice(bs.Slice):
        def reader(self, shard, dep_readers, ctx):
            from bigslice_amd.sortio import SortReader
            from bigslice_amd import config as cfg
            return SortReader(dep_readers[0],
                              run_bytes=int(os.environ.get(
                                  "SORT_RUN_BYTES", 2 << 30)),
                              device=ctx.device, chunk=ctx.chunk)

    return SortSlice(src.schema, nshard, deps=[bs.Dep(src)])


FV_WC = bs.func(build_wordcount)
FV_MF = bs.func(build_mapfilter)
FV_CG = bs.func(build_cogroup)
FV_SORT = bs.func(build_sort)


def emit(metric, rows, elapsed_s, steps, warmup, cfg, device):
    ms = elapsed_s * 1000 / steps
    print(json.dumps({
        "metric": metric, "value": rows / (ms / 1000), "unit": "rows/sec",
        "n_gpus": 1, "steps": steps, "warmup": warmup,
        "ms_per_step": ms, "higher_is_better": True, "scaling": "weak",
        "vs_baseline": None, "dtype": "int64", "data": "synthetic",
        "config": cfg | {"device": device},
    }))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, required=True,
                    choices=[1, 2, 4, 5])
    ap.add_argument("--rows", type=int, default=None)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--shards", type=int, default=8)
    ap.add_argument("--run-bytes", type=int, default=None,
                    help="external-sort run size (forces spill when "
                         "shard bytes exceed it)")
    args = ap.parse_args()
    if args.run_bytes:
        os.environ["SORT_RUN_BYTES"] = str(args.run_bytes)
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    nshard = args.shards
    sess = bs.start(parallelism=nshard, device=device)

    def g(shape_rows, hi, seed):
        gen = torch.Generator(device=device)
        gen.manual_seed(seed)
        return torch.randint(0, hi, (shape_rows,), dtype=torch.int64,
                             device=device, generator=gen)

    if args.config == 1:
        # CPU plumbing config: correctness + rows/sec through the whole
        # engine on the host path (4 shards per BASELINE).
        nshard = 4
        sess = bs.start(parallelism=nshard, device="cpu")
        nlines = args.rows or 20_000
        rows = nlines * 8
        gen_wc_corpus(nlines)

        def step():
            res = sess.run(FV_WC, nshard, nlines)
            total = sum(c for _, c in res.scan())
            assert total == rows, (total, rows)
            res.discard()
        for _ in range(args.warmup):
            step()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            step()
        emit("rows/sec wordcount (CPU plumbing)", rows,
             time.perf_counter() - t0, args.steps, args.warmup,
             {"model": "Wordcount local CPU (BASELINE config 1)",
              "rows_total": rows, "global_batch": rows,
              "shards": nshard, "parallelism": "cpu"}, "cpu")
        return
    if args.config == 2:
        rows = args.rows or 100_000_000
        per = rows // nshard
        for s in range(nshard):
            _DATA[("mf", s)] = g(per, 1 << 40, s)
        fv, cfg = FV_MF, {"model": "Map+Filter scan (BASELINE config 2)",
                          "rows_total": per * nshard,
                          "global_batch": per * nshard,
                          "shards": nshard, "parallelism": "dp1"}
        metric = "rows/sec Map+Filter scan"
    elif args.config == 4:
        rows = args.rows or 500_000_000
        per = rows // nshard
        for s in range(nshard):
            _DATA[("cga", s)] = (g(per, 1 << 20, 100 + s), g(per, 1 << 30, 200 + s))
            _DATA[("cgb", s)] = (g(per, 1 << 20, 300 + s), g(per, 1 << 30, 400 + s))
        fv, cfg = FV_CG, {"model": "Cogroup join (BASELINE config 4)",
                          "rows_total": 2 * per * nshard,
                          "global_batch": 2 * per * nshard,
                          "shards": nshard, "parallelism": "dp1"}
        rows = 2 * per * nshard
        metric = "rows/sec Cogroup join"
    else:
        rows = args.rows or 1_000_000_000
        per = rows // nshard
        for s in range(nshard):
            lazy_over = int(os.environ.get(
                "BIGSLICE_BENCH_LAZY_OVER", str(100 << 30)))
            if per * 16 * nshard > lazy_over:
                _DATA[("sort", s)] = ("lazy", per, 1 << 62, 500 + s,
                                      device)
            else:
                _DATA[("sort", s)] = (g(per, 1 << 62, 500 + s),
                                      g(per, 1 << 30, 600 + s))
        fv, cfg = FV_SORT, {"model": "External sort (BASELINE config 5)",
                            "rows_total": per * nshard,
                            "global_batch": per * nshard,
                            "shards": nshard, "parallelism": "dp1"}
        metric = "rows/sec external sort"

    def step():
        res = sess.run(fv, nshard)
        res.discard()

    for _ in range(args.warmup):
        step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    emit(metric, rows, time.perf_counter() - t0, args.steps, args.warmup,
         cfg, device)


if __name__ == "__main__":
    main()
