"""Property-based differential tests (hypothesis; the reference uses
gofuzz for the same purpose): the tensor aggregation path vs a plain
dict oracle, merge-sort invariants, and partition totality."""

import pytest
import torch
from hypothesis import given, settings, strategies as st

from bigslice_amd.frame import Frame
from bigslice_amd.ops.aggregate import (Aggregation, DictAggregator,
                                        TensorAggregator)
from bigslice_amd.runtime.partition import split_frame
from bigslice_amd.schema import Schema
from bigslice_amd.sliceio import IterReader, read_all
from bigslice_amd.sortio import MergeReader, SortReader


rows_strategy = st.lists(
    st.tuples(st.integers(-100, 100), st.integers(-1000, 1000)),
    min_size=0, max_size=300)


@settings(max_examples=40, deadline=None)
@given(rows=rows_strategy,
       agg=st.sampled_from(["sum", "min", "max"]),
       nbatches=st.integers(1, 4))
def test_tensor_aggregator_matches_dict_oracle(rows, agg, nbatches):
    schema = Schema([torch.int64, torch.int64], 1)
    ta = TensorAggregator(schema, Aggregation([agg]), "cpu")
    da = DictAggregator(schema, Aggregation([agg]))
    # split rows into batches
    per = max(1, len(rows) // nbatches)
    for off in range(0, len(rows), per):
        part = rows[off:off + per]
        if not part:
            continue
        f = Frame([torch.tensor([r[0] for r in part], dtype=torch.int64),
                   torch.tensor([r[1] for r in part], dtype=torch.int64)])
        ta.add(f)
        da.add(f)
    got = {}
    for f in ta.result_frames(64):
        for k, v in zip(f.columns[0].tolist(), f.columns[1].tolist()):
            assert k not in got
            got[k] = v
    want = {}
    for f in da.result_frames(64):
        for k, v in zip(f.columns[0].tolist(), f.columns[1].tolist()):
            want[k] = v
    assert got == want


@settings(max_examples=30, deadline=None)
@given(runs=st.lists(
    st.lists(st.integers(-50, 50), min_size=0, max_size=80),
    min_size=1, max_size=5),
    chunk=st.integers(2, 32))
def test_merge_reader_property(runs, chunk):
    readers = []
    expect = []
    for run in runs:
        run = sorted(run)
        expect.extend(run)
        t = torch.tensor(run, dtype=torch.int64)

        def gen(t=t):
            for off in range(0, t.shape[0], 7):
                yield Frame([t[off:off + 7]])
        readers.append(IterReader(gen()))
    m = MergeReader(readers, chunk=chunk)
    out = read_all(m)
    got = out.columns[0].tolist() if out is not None else []
    assert got == sorted(expect)


@settings(max_examples=30, deadline=None)
@given(vals=st.lists(st.integers(-10**9, 10**9), min_size=0,
                     max_size=200),
       run_bytes=st.integers(16, 4096))
def test_sort_reader_property(vals, run_bytes):
    t = torch.tensor(vals, dtype=torch.int64)

    def gen():
        for off in range(0, t.shape[0], 13):
            yield Frame([t[off:off + 13]])
    sr = SortReader(IterReader(gen()), run_bytes=run_bytes)
    out = read_all(sr)
    got = out.columns[0].tolist() if out is not None else []
    assert got == sorted(vals)


@settings(max_examples=30, deadline=None)
@given(keys=st.lists(st.integers(-2**62, 2**62), min_size=1,
                     max_size=300),
       nparts=st.integers(1, 9))
def test_partition_property(keys, nparts):
    f = Frame([torch.tensor(keys, dtype=torch.int64)])
    parts = split_frame(f, nparts, None)
    got = sorted(v for p in parts if p is not None
                 for v in p.columns[0].tolist())
    assert got == sorted(keys)
    # determinism: same key -> same partition
    seen = {}
    for pi, p in enumerate(parts):
        if p is None:
            continue
        for k in p.columns[0].tolist():
            assert seen.setdefault(k, pi) == pi


@given(rows=st.lists(st.tuples(st.integers(0, 30),
                               st.integers(-100, 100)),
                     min_size=0, max_size=150),
       ops=st.lists(st.sampled_from(["mapadd", "filter", "reshuffle",
                                     "reshard", "flatmap"]), max_size=4),
       nshard=st.integers(1, 4))
@settings(max_examples=40, deadline=None)
def test_random_pipeline_matches_oracle(rows, ops, nshard):
    """Random op compositions agree with a plain-Python oracle."""
    import bigslice_amd as bs

    keys = torch.tensor([k for k, _ in rows], dtype=torch.int64)
    vals = torch.tensor([v for _, v in rows], dtype=torch.int64)

    def build():
        s = bs.Const(nshard, keys, vals, prefix=1)
        for i, op in enumerate(ops):
            if op == "mapadd":
                s = bs.Map(s, lambda k, v: (k, v + 3))
            elif op == "filter":
                s = bs.Filter(s, lambda k, v: (k & 3) != 1)
            elif op == "flatmap":
                # vectorized: each row (k,v) -> (k,v),(k,-v)
                s = bs.Flatmap(s, lambda k, v: (torch.cat([k, k]),
                                                torch.cat([v, -v])))
            elif op == "reshuffle":
                s = bs.Reshuffle(s)
            else:
                s = bs.Reshard(s, 2)
        return bs.Reduce(s, "sum")

    # oracle
    data = list(rows)
    for op in ops:
        if op == "mapadd":
            data = [(k, v + 3) for k, v in data]
        elif op == "filter":
            data = [(k, v) for k, v in data if (k & 3) != 1]
        elif op == "flatmap":
            data = [p for k, v in data for p in ((k, v), (k, -v))]
    oracle = {}
    for k, v in data:
        oracle[k] = oracle.get(k, 0) + v

    sess = bs.start(parallelism=3, device="cpu")
    got = dict(sess.run(bs.func(build)).scan())
    assert got == oracle, (ops, nshard)
