"""External sort / merge tests over randomized frames (reference
sortio/sort_test.go shapes, fuzzFrame analog)."""

import operator

import pytest
import torch

from bigslice_amd.frame import Frame
from bigslice_amd.schema import Schema
from bigslice_amd.sliceio import IterReader, read_all
from bigslice_amd.sortio import MergeReader, SortReader, reduce_reader


def frames_of(tensors, prefix=1, batch=1000):
    def gen():
        n = tensors[0].shape[0]
        for off in range(0, n, batch):
            yield Frame([t[off:off + batch] for t in tensors], prefix)
    return IterReader(gen())


def test_sortreader_sorts_across_spilled_runs():
    g = torch.Generator().manual_seed(0)
    keys = torch.randint(0, 10_000, (50_000,), generator=g,
                         dtype=torch.int64)
    vals = torch.arange(50_000, dtype=torch.int64)
    # tiny run budget forces many spilled runs
    sr = SortReader(frames_of([keys, vals]), run_bytes=100_000)
    out = read_all(sr)
    assert len(out) == 50_000
    ok = out.columns[0]
    assert bool((ok[1:] >= ok[:-1]).all())
    # rows intact
    got = sorted(zip(ok.tolist(), out.columns[1].tolist()))
    want = sorted(zip(keys.tolist(), vals.tolist()))
    assert got == want


def test_merge_reader_two_runs():
    a = torch.arange(0, 100, 2, dtype=torch.int64)
    b = torch.arange(1, 101, 2, dtype=torch.int64)
    m = MergeReader([frames_of([a.sort().values], batch=7),
                     frames_of([b.sort().values], batch=13)], chunk=16)
    out = read_all(m)
    assert out.columns[0].tolist() == list(range(100))


def test_merge_reader_multi_key_ties_across_batches():
    # run 1 ends a batch mid-way through first-key 5; secondary keys of
    # the next batch are smaller than run 2's: the merge must wait.
    k1 = torch.tensor([1, 5, 5, 5, 5], dtype=torch.int64)
    s1 = torch.tensor([9, 0, 1, 2, 3], dtype=torch.int64)
    k2 = torch.tensor([5, 6], dtype=torch.int64)
    s2 = torch.tensor([1, 0], dtype=torch.int64)
    m = MergeReader([frames_of([k1, s1], prefix=2, batch=2),
                     frames_of([k2, s2], prefix=2, batch=2)], chunk=2)
    out = read_all(m)
    rows = list(zip(out.columns[0].tolist(), out.columns[1].tolist()))
    assert rows == sorted(rows)
    assert sorted(rows) == sorted(
        list(zip(k1.tolist(), s1.tolist())) +
        list(zip(k2.tolist(), s2.tolist())))


def test_merge_reader_windows_stay_bounded():
    # Regression: the window-growth loop used to grow the MINIMUM run
    # on every window (the minimum is at its own cutoff by definition),
    # buffering entire runs — a 10B-row external sort OOMed on it.
    # With unique keys every window must stay near the chunk size.
    runs = []
    for r in range(4):
        keys = torch.arange(r, 400_000, 4, dtype=torch.int64)
        runs.append(frames_of([keys], batch=4096))
    m = MergeReader(runs, chunk=8192)
    total = 0
    expected = None
    while True:
        f = m.read()
        if f is None:
            break
        assert len(f) <= 8 * 8192, len(f)
        col = f.columns[0]
        if expected is not None:
            assert int(col[0]) >= expected
        expected = int(col[-1])
        assert torch.equal(col, col.sort().values)
        total += len(f)
    assert total == 4 * 100_000


def test_merge_reader_giant_duplicate_block():
    # One run is a single giant equal-key block spanning many batches:
    # the merge must grow ONLY that run (bounded by the block) and keep
    # ordering with the other run's interleaved keys.
    k1 = torch.full((50_000,), 7, dtype=torch.int64)
    k2 = torch.tensor([1, 7, 7, 9, 12], dtype=torch.int64)
    m = MergeReader([frames_of([k1], batch=1000),
                     frames_of([k2], batch=2)], chunk=64)
    out = read_all(m)
    assert len(out) == 50_005
    col = out.columns[0]
    assert torch.equal(col, col.sort().values)


def test_reduce_reader_combines_across_streams():
    schema = Schema([torch.int64, torch.int64], 1)
    a_k = torch.tensor([1, 1, 2, 3], dtype=torch.int64)
    a_v = torch.tensor([1, 2, 3, 4], dtype=torch.int64)
    b_k = torch.tensor([2, 3, 3, 9], dtype=torch.int64)
    b_v = torch.tensor([10, 20, 30, 40], dtype=torch.int64)
    from bigslice_amd.ops.aggregate import Aggregation
    r = reduce_reader([frames_of([a_k, a_v], batch=2),
                       frames_of([b_k, b_v], batch=2)],
                      schema, Aggregation(["sum"]), chunk=4)
    out = read_all(r)
    got = dict(zip(out.columns[0].tolist(), out.columns[1].tolist()))
    assert got == {1: 3, 2: 13, 3: 54, 9: 40}


def test_sortreader_object_keys():
    keys = ["banana", "apple", "cherry", "apple"]
    vals = torch.tensor([1, 2, 3, 4], dtype=torch.int64)
    sr = SortReader(IterReader(iter([Frame([list(keys), vals])])),
                    run_bytes=10)
    out = read_all(sr)
    assert out.columns[0] == ["apple", "apple", "banana", "cherry"]


@pytest.mark.gpu
def test_sortreader_gpu_large():
    g = torch.Generator(device="cuda:0").manual_seed(1)
    keys = torch.randint(0, 1 << 40, (5_000_000,), dtype=torch.int64,
                         device="cuda:0", generator=g)
    vals = torch.arange(5_000_000, dtype=torch.int64, device="cuda:0")
    sr = SortReader(frames_of([keys, vals], batch=1_000_000),
                    run_bytes=20_000_000, device="cuda:0")
    out = read_all(sr)
    assert len(out) == 5_000_000
    ok = out.columns[0]
    assert bool((ok[1:] >= ok[:-1]).all())
    assert int(out.columns[1].sum()) == int(vals.sum())


def test_merge_reader_mixed_eof_states():
    # regression: one run at EOF while others still buffered must not
    # mix scalar types in the cutoff computation.
    a = torch.arange(0, 10, dtype=torch.int64)         # short run (EOF early)
    b = torch.arange(5, 40, dtype=torch.int64)
    c = torch.arange(20, 60, dtype=torch.int64)
    m = MergeReader([frames_of([a], batch=3), frames_of([b], batch=4),
                     frames_of([c], batch=5)], chunk=6)
    out = read_all(m)
    got = out.columns[0].tolist()
    assert got == sorted(a.tolist() + b.tolist() + c.tolist())


@pytest.mark.gpu
def test_sortreader_gpu_forced_spill():
    g = torch.Generator(device="cuda:0").manual_seed(3)
    keys = torch.randint(0, 1 << 50, (8_000_000,), dtype=torch.int64,
                         device="cuda:0", generator=g)
    vals = torch.arange(8_000_000, dtype=torch.int64, device="cuda:0")
    sr = SortReader(frames_of([keys, vals], batch=1_000_000),
                    run_bytes=8_000_000,  # ~0.5M rows per run -> 16 runs
                    device="cuda:0")
    out = read_all(sr)
    assert len(out) == 8_000_000
    ok = out.columns[0]
    assert bool((ok[1:] >= ok[:-1]).all())
    assert int(out.columns[1].sum()) == int(vals.sum())
