"""Large-scale integration stress workloads with self-verifying
invariants (reference cmd/slicer: reduce / cogroup / memiter / oom).

  python -m bigslice_amd.tools.slicer reduce --nshard 16 --nkey 100000
  python -m bigslice_amd.tools.slicer cogroup --nshard 8 --nkey 10000
  python -m bigslice_amd.tools.slicer memiter --iters 5

Each subcommand asserts its invariant and exits non-zero on violation.
"""

from __future__ import annotations

import argparse
import time

import torch

import bigslice_amd as bs


def reduce_stress(nshard: int, nkey: int, device: str):
    """cmd/slicer/reduce.go invariant: nshard x nkey rows (keys permuted
    per shard); after Reduce(sum of ones) every key appears exactly once
    with count == nshard."""

    def build(nshard, nkey):
        def gen(shard, ctx):
            g = torch.Generator()
            g.manual_seed(shard)
            keys = torch.randperm(nkey, generator=g).to(torch.int64)
            if device != "cpu":
                keys = keys.to(device)
            yield (keys, torch.ones_like(keys))
        src = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
        return bs.Reduce(src, "sum")

    fv = bs.func(build)
    sess = bs.start(parallelism=8, device=device)
    t0 = time.perf_counter()
    res = sess.run(fv, nshard, nkey)
    n = 0
    for k, c in res.scan():
        assert c == nshard, f"key {k}: count {c} != {nshard}"
        n += 1
    assert n == nkey, f"{n} keys != {nkey}"
    print(f"reduce OK: {nshard}x{nkey} rows in "
          f"{time.perf_counter()-t0:.2f}s")


def cogroup_stress(nshard: int, nkey: int, device: str):
    """cmd/slicer/cogroup.go invariant: shard s contributes value
    (s<<24)|k for key k; the join must reconstruct the exact per-key
    value set."""

    def build(nshard, nkey):
        def gen(shard, ctx):
            keys = torch.arange(nkey, dtype=torch.int64)
            vals = (shard << 24) | keys
            if device != "cpu":
                keys, vals = keys.to(device), vals.to(device)
            yield (keys, vals)
        a = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
        b = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
        return bs.Cogroup(a, b)

    fv = bs.func(build)
    sess = bs.start(parallelism=8, device=device)
    t0 = time.perf_counter()
    res = sess.run(fv, nshard, nkey)
    n = 0
    for k, va, vb in res.scan():
        want = sorted((s << 24) | k for s in range(nshard))
        assert sorted(va) == want, f"key {k}: bad A values"
        assert sorted(vb) == want, f"key {k}: bad B values"
        n += 1
    assert n == nkey
    print(f"cogroup OK: {nshard}x{nkey} rows in "
          f"{time.perf_counter()-t0:.2f}s")


def memiter_stress(iters: int, device: str):
    """Iterative invocations reusing prior Results (session memory
    behavior under iteration, cmd/slicer memiter analog)."""
    def base():
        keys = torch.arange(100_000, dtype=torch.int64) % 1000
        if device != "cpu":
            keys = keys.to(device)
        return bs.Reduce(bs.Const(4, keys, torch.ones_like(keys)), "sum")

    def step(prev):
        return bs.Map(prev, lambda k, v: (k, v + 1))

    fv0, fv1 = bs.func(base), bs.func(step)
    sess = bs.start(parallelism=4, device=device)
    res = sess.run(fv0)
    for i in range(iters):
        nxt = sess.run(fv1, res)
        res.discard()
        res = nxt
    rows = dict(res.scan())
    assert rows[0] == 100 + iters
    print(f"memiter OK: {iters} iterations")


def oom_stress(size: int, device: str):
    """OOM *reporting* check (cmd/slicer/oom.go:20-31: the reference
    deliberately OOMs a worker and only cares that the error is
    reported cleanly).  MI355X analog: one absurd single-tensor
    allocation inside a Map UDF — the HIP/host allocator rejects it
    immediately (no gradual memory pressure, box-safe) — and the
    session must surface it as an ordinary task error naming the
    allocation, not hang or die."""

    if size < (1 << 45):
        raise SystemExit(
            f"--size {size} is small enough that the allocation could "
            "succeed (and the UDF would then fail with a misleading "
            "'not reached'); use >= 2**45 bytes for a guaranteed "
            "immediate allocator rejection")

    def build(nbytes):
        src = bs.Const(2, torch.arange(4, dtype=torch.int64))

        def boom(k):
            torch.empty(nbytes, dtype=torch.uint8,
                        device=device if device != "cpu" else "cpu")
            raise AssertionError("not reached")
        return bs.Map(src, boom, out_schema=(int,))

    fv = bs.func(build)
    sess = bs.start(parallelism=2, device=device)
    try:
        sess.run(fv, size)
    except Exception as e:
        msg = str(e).lower()
        assert ("memory" in msg or "alloc" in msg), (
            f"OOM surfaced as unrelated error: {e!r}")
        print(f"oom OK: allocation of {size} bytes reported as task "
              f"error: {type(e).__name__}")
        return
    raise AssertionError("oom run unexpectedly succeeded")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("cmd", choices=["reduce", "cogroup", "memiter", "oom"])
    ap.add_argument("--nshard", type=int, default=8)
    ap.add_argument("--nkey", type=int, default=100_000)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--size", type=int, default=1 << 48,
                    help="oom allocation size in bytes")
    ap.add_argument("--device", type=str, default=None)
    args = ap.parse_args()
    device = args.device or (
        "cuda:0" if torch.cuda.is_available() else "cpu")
    if args.cmd == "reduce":
        reduce_stress(args.nshard, args.nkey, device)
    elif args.cmd == "cogroup":
        cogroup_stress(args.nshard, args.nkey, device)
    elif args.cmd == "oom":
        oom_stress(args.size, device)
    else:
        memiter_stress(args.iters, device)


if __name__ == "__main__":
    main()
