"""End-to-end operator tests through the local executor (CPU), mirroring
the reference's dual-executor slice tests (slice_test.go harness)."""

import operator

import pytest
import torch

import bigslice_amd as bs


def run_slice(builder, *args, parallelism=2):
    fv = bs.func(builder)
    sess = bs.start(parallelism=parallelism, device="cpu")
    return sess.run(fv, *args)


def sorted_rows(res):
    return sorted(res.scan())


def test_const_roundtrip():
    res = run_slice(lambda: bs.Const(3, torch.arange(10, dtype=torch.int64)))
    assert sorted_rows(res) == list(range(10))


def test_map_vectorized():
    res = run_slice(lambda: bs.Map(
        bs.Const(2, torch.arange(5, dtype=torch.int64)),
        lambda x: (x, x * 2)))
    assert sorted_rows(res) == [(i, 2 * i) for i in range(5)]


def test_filter_vectorized():
    res = run_slice(lambda: bs.Filter(
        bs.Const(2, torch.arange(10, dtype=torch.int64)),
        lambda x: x % 2 == 0))
    assert sorted_rows(res) == [0, 2, 4, 6, 8]


def test_map_rowwise_strings():
    res = run_slice(lambda: bs.Map(
        bs.Const(2, ["a", "bb", "ccc"]),
        lambda s: (s, len(s)), out_schema=(str, int), rowwise=True))
    assert sorted_rows(res) == [("a", 1), ("bb", 2), ("ccc", 3)]


def test_flatmap_rowwise():
    res = run_slice(lambda: bs.Flatmap(
        bs.Const(2, ["a b", "c d e"]),
        lambda s: [(w,) for w in s.split()],
        out_schema=(str,), rowwise=True))
    assert sorted_rows(res) == ["a", "b", "c", "d", "e"]


def test_head():
    res = run_slice(lambda: bs.Head(
        bs.Const(1, torch.arange(100, dtype=torch.int64)), 5))
    assert sorted_rows(res) == [0, 1, 2, 3, 4]


def test_reduce_sum():
    keys = torch.tensor([1, 2, 1, 3, 2, 1], dtype=torch.int64)
    vals = torch.tensor([10, 20, 30, 40, 50, 60], dtype=torch.int64)
    res = run_slice(lambda: bs.Reduce(bs.Const(3, keys, vals), "sum"))
    assert sorted_rows(res) == [(1, 100), (2, 70), (3, 40)]


def test_reduce_operator_add():
    keys = torch.tensor([1, 2, 1], dtype=torch.int64)
    vals = torch.tensor([1.0, 2.0, 3.0], dtype=torch.float64)
    res = run_slice(lambda: bs.Reduce(bs.Const(2, keys, vals), operator.add))
    assert sorted_rows(res) == [(1, 4.0), (2, 2.0)]


def test_reduce_custom_fn_host_path():
    res = run_slice(lambda: bs.Reduce(
        bs.Const(2, ["a", "b", "a"], torch.tensor([1, 2, 3],
                                                  dtype=torch.int64)),
        lambda a, b: a + b))
    assert sorted_rows(res) == [("a", 4), ("b", 2)]


def test_reshuffle_preserves_rows_and_colocates_keys():
    # parity with reference reshuffle_test.go:23-84: all rows preserved;
    # each key lands in exactly one shard.
    n = 1000
    keys = torch.arange(n, dtype=torch.int64) % 17
    vals = torch.arange(n, dtype=torch.int64)

    fv = bs.func(lambda m: bs.Reshuffle(bs.Const(m, keys, vals)))
    for m in range(1, 6):
        sess = bs.start(parallelism=2, device="cpu")
        res = sess.run(fv, m)
        # multiset preserved
        rows = sorted(res.scan())
        assert rows == sorted(zip(keys.tolist(), vals.tolist()))
        # co-location: read each shard's partition separately
        shard_keys = []
        for t in res.tasks:
            r = sess.executor.reader(t, 0)
            got = set()
            for f in r:
                got.update(f.columns[0].tolist())
            shard_keys.append(got)
        for i in range(len(shard_keys)):
            for j in range(i + 1, len(shard_keys)):
                assert not (shard_keys[i] & shard_keys[j])


def test_reshard():
    s = bs.Const(4, torch.arange(10, dtype=torch.int64))
    same = bs.Reshard(s, 4)
    assert same is s
    res = run_slice(lambda: bs.Reshard(
        bs.Const(4, torch.arange(10, dtype=torch.int64)), 2))
    assert res.slice.num_shards == 2
    assert sorted_rows(res) == list(range(10))


def test_fold():
    keys = torch.tensor([1, 1, 2], dtype=torch.int64)
    vals = torch.tensor([5, 6, 7], dtype=torch.int64)
    res = run_slice(lambda: bs.Fold(
        bs.Const(2, keys, vals),
        lambda acc, v: (acc or 0) + v, out_schema=(int,)))
    assert sorted_rows(res) == [(1, 11), (2, 7)]


def test_cogroup():
    a_keys = ["x", "y", "x"]
    a_vals = torch.tensor([1, 2, 3], dtype=torch.int64)
    b_keys = ["y", "z"]
    b_vals = torch.tensor([20, 30], dtype=torch.int64)
    res = run_slice(lambda: bs.Cogroup(
        bs.Const(2, a_keys, a_vals), bs.Const(2, b_keys, b_vals)))
    rows = sorted_rows(res)
    assert rows == [
        ("x", [1, 3], []),
        ("y", [2], [20]),
        ("z", [], [30]),
    ]


def test_scan_terminal():
    seen = {}

    def scan_fn(shard, rows):
        seen[shard] = sorted(rows)

    res = run_slice(lambda: bs.Scan(
        bs.Const(2, torch.arange(6, dtype=torch.int64)), scan_fn))
    allrows = sorted(v for rows in seen.values() for v in rows)
    assert allrows == list(range(6))


def test_writerfunc():
    written = []
    res = run_slice(lambda: bs.WriterFunc(
        bs.Const(1, torch.arange(4, dtype=torch.int64)),
        lambda shard, f: written.append((shard, len(f)))))
    assert sorted_rows(res) == [0, 1, 2, 3]
    assert sum(n for _, n in written) == 4


def test_prefixed_multi_key_reduce():
    k1 = torch.tensor([1, 1, 2, 1], dtype=torch.int64)
    k2 = torch.tensor([1, 1, 2, 2], dtype=torch.int64)
    v = torch.tensor([10, 20, 30, 40], dtype=torch.int64)
    res = run_slice(lambda: bs.Reduce(
        bs.Prefixed(bs.Const(2, k1, k2, v), 2), "sum"))
    assert sorted_rows(res) == [(1, 1, 30), (1, 2, 40), (2, 2, 30)]


def test_wordcount_end_to_end():
    # The canonical acceptance program (docs/index.md:93-155):
    # ScanReader -> Flatmap(split) -> Map((w,1)) -> Reduce(+)
    text = ["the quick brown fox", "jumps over the lazy dog",
            "the fox"]

    def build(nshard):
        lines = bs.ScanReader(nshard, lambda: iter(text))
        words = bs.Flatmap(lines, lambda s: [(w,) for w in s.split()],
                           out_schema=(str,), rowwise=True)
        counts = bs.Map(words, lambda w: (w, 1), out_schema=(str, int),
                        rowwise=True)
        return bs.Reduce(counts, "sum")

    res = run_slice(build, 4)
    got = dict(res.scan())
    assert got == {"the": 3, "quick": 1, "brown": 1, "fox": 2,
                   "jumps": 1, "over": 1, "lazy": 1, "dog": 1}


def test_reader_func():
    def gen(shard, ctx):
        yield (torch.arange(3, dtype=torch.int64) + 10 * shard,)

    res = run_slice(lambda: bs.ReaderFunc(
        3, gen, bs.schema_of(int)))
    assert sorted_rows(res) == [0, 1, 2, 10, 11, 12, 20, 21, 22]


def test_result_reuse_iterative():
    # Result passed back into another Func reuses tasks
    # (exec/session.go:40-43 iterative computing).
    fv1 = bs.func(lambda: bs.Reduce(
        bs.Const(2, torch.tensor([1, 2, 1], dtype=torch.int64),
                 torch.tensor([1, 1, 1], dtype=torch.int64)), "sum"))
    fv2 = bs.func(lambda prev: bs.Map(prev, lambda k, c: (k, c * 10)))
    sess = bs.start(parallelism=2, device="cpu")
    r1 = sess.run(fv1)
    r2 = sess.run(fv2, r1)
    assert sorted(r2.scan()) == [(1, 20), (2, 10)]


def test_custom_ops_length_hash_colocation():
    # Parity with the reference's ONLY in-repo use of RegisterOps
    # (reshuffle_test.go:86-94): a key type hashed by its string LENGTH
    # must land every key of one length in exactly one shard while the
    # full multiset of rows is preserved, for shard counts 1..6.
    from bigslice_amd import hashing

    class LengthKey(str):
        pass

    hashing.register_ops(LengthKey,
                         hash_fn=lambda v, seed: seed + len(v),
                         less_key=lambda v: (len(v), str(v)))

    words = [LengthKey(w) for w in
             ["a", "bb", "cc", "ddd", "e", "ffff", "gg", "hhh"]]
    vals = torch.arange(len(words), dtype=torch.int64)
    fv = bs.func(lambda m: bs.Reshuffle(
        bs.Const(m, list(words), vals)))
    for m in range(1, 7):
        sess = bs.start(parallelism=2, device="cpu")
        res = sess.run(fv, m)
        rows = sorted((str(k), v) for k, v in res.scan())
        assert rows == sorted((str(k), v.item())
                              for k, v in zip(words, vals))
        lengths_by_shard = []
        for t in res.tasks:
            ls = set()
            for f in sess.executor.reader(t, 0):
                ls.update(len(k) for k in f.columns[0])
            lengths_by_shard.append(ls)
        for i in range(len(lengths_by_shard)):
            for j in range(i + 1, len(lengths_by_shard)):
                assert not (lengths_by_shard[i] & lengths_by_shard[j])


def test_reduce_keys_only_distinct():
    # Reduce with zero value columns = distinct keys per the combiner
    keys = torch.tensor([3, 1, 3, 2, 1, 1], dtype=torch.int64)
    res = run_slice(lambda: bs.Reduce(bs.Const(3, keys), "sum"))
    assert sorted_rows(res) == [1, 2, 3]


def test_fast_wordcount_recipe():
    from bigslice_amd.recipes import fast_wordcount
    text = ["the quick brown fox", "jumps over the lazy dog",
            "the fox"] * 5
    fv = bs.func(lambda n: fast_wordcount(n, lambda: iter(text)))
    res = bs.start(parallelism=2, device="cpu").run(fv, 4)
    got = dict(res.scan())
    assert got == {"the": 15, "quick": 5, "brown": 5, "fox": 10,
                   "jumps": 5, "over": 5, "lazy": 5, "dog": 5}


def test_precombined_passthrough_correctness():
    # The single-stream passthrough must yield identical results to
    # full re-aggregation (machine-combiners on and off).
    import os
    keys = torch.randint(0, 97, (5000,), dtype=torch.int64)
    vals = torch.randint(0, 10, (5000,), dtype=torch.int64)

    fv = bs.func(lambda: bs.Reduce(bs.Const(6, keys, vals), "sum"))
    on = bs.start(parallelism=3, device="cpu").run(fv)
    from bigslice_amd.runtime.local import LocalExecutor
    from bigslice_amd.runtime.session import Session
    ex = LocalExecutor(parallelism=3, device="cpu")
    ex.machine_combiners = False
    off = Session(ex).run(fv)
    assert sorted(on.scan()) == sorted(off.scan())
    # oracle
    want = {}
    for k, v in zip(keys.tolist(), vals.tolist()):
        want[k] = want.get(k, 0) + v
    assert dict(on.scan()) == want


def test_mixed_combiners_over_shared_producer():
    # regression: Reduce("sum") and Reduce("max") branching from ONE
    # slice must not share pre-combined producer output.
    keys = torch.tensor([1, 1, 2], dtype=torch.int64)
    vals = torch.tensor([5, 7, 3], dtype=torch.int64)

    def build():
        base = bs.Map(bs.Const(2, keys, vals), lambda k, v: (k, v))
        return bs.Cogroup(bs.Reduce(base, "sum"), bs.Reduce(base, "max"))

    res = bs.slicetest.run(build)
    assert sorted(res.scan()) == [(1, [12], [7]), (2, [3], [3])]


def test_reshard_up_and_down():
    k = torch.arange(100, dtype=torch.int64) % 7
    v = torch.ones(100, dtype=torch.int64)
    for nshard in (1, 3, 16):
        res = run_slice(lambda: bs.Reduce(
            bs.Reshard(bs.Const(2, k, v, prefix=1), nshard), "sum"))
        rows = sorted_rows(res)
        assert len(rows) == 7 and sum(c for _, c in rows) == 100
