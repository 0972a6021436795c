"""Frame semantics tests (reference frame/frame_test.go shapes)."""

import io

import pytest
import torch

from bigslice_amd.frame import Frame
from bigslice_amd.schema import OBJECT, Schema
from bigslice_amd.sliceio import FrameReader, MultiReader, read_all
from bigslice_amd.sliceio.codec import (CorruptionError, decode_frame,
                                        encode_frame)
from bigslice_amd.sliceio.spiller import Spiller


def make_frame(n=10):
    return Frame([torch.arange(n, dtype=torch.int64),
                  torch.arange(n, dtype=torch.float64) * 0.5],
                 prefix=1)


def test_slice_and_len():
    f = make_frame(10)
    s = f.slice(2, 5)
    assert len(s) == 3
    assert s.columns[0].tolist() == [2, 3, 4]


def test_concat():
    f = Frame.concat([make_frame(3), make_frame(2)])
    assert len(f) == 5
    assert f.columns[0].tolist() == [0, 1, 2, 0, 1]


def test_mask_and_select():
    f = make_frame(6)
    m = f.mask(torch.tensor([True, False, True, False, True, False]))
    assert m.columns[0].tolist() == [0, 2, 4]
    g = f.select(torch.tensor([5, 0]))
    assert g.columns[0].tolist() == [5, 0]


def test_object_columns():
    f = Frame([["a", "b", "c"], torch.tensor([1, 2, 3],
                                             dtype=torch.int64)])
    assert f.has_objects
    s = f.slice(1, 3)
    assert s.columns[0] == ["b", "c"]
    m = f.mask(torch.tensor([True, False, True]))
    assert m.columns[0] == ["a", "c"]


def test_sort_by_prefix_multi():
    f = Frame([torch.tensor([2, 1, 2, 1], dtype=torch.int64),
               torch.tensor([1, 2, 0, 1], dtype=torch.int64),
               torch.tensor([10, 20, 30, 40], dtype=torch.int64)],
              prefix=2)
    s = f.sort_by_prefix()
    assert s.columns[0].tolist() == [1, 1, 2, 2]
    assert s.columns[1].tolist() == [1, 2, 0, 1]
    assert s.columns[2].tolist() == [40, 20, 30, 10]


def test_schema_roundtrip():
    sch = Schema([torch.int64, OBJECT, torch.float32], prefix=2)
    assert Schema.from_key(sch.key()) == sch


def test_codec_roundtrip():
    f = Frame([torch.arange(100, dtype=torch.int64),
               ["s%d" % i for i in range(100)],
               torch.randn(100, dtype=torch.float32)], prefix=1)
    buf = io.BytesIO()
    encode_frame(f, buf)
    buf.seek(0)
    g = decode_frame(buf)
    assert g.columns[0].tolist() == f.columns[0].tolist()
    assert g.columns[1] == f.columns[1]
    assert torch.equal(g.columns[2], f.columns[2])
    assert g.prefix == 1
    assert decode_frame(buf) is None  # EOF


def test_codec_detects_corruption():
    f = make_frame(50)
    buf = io.BytesIO()
    encode_frame(f, buf)
    raw = bytearray(buf.getvalue())
    raw[30] ^= 0xFF
    with pytest.raises(CorruptionError):
        decode_frame(io.BytesIO(bytes(raw)))


def test_spiller_roundtrip_memory_and_disk():
    sp = Spiller(host_budget_bytes=200)  # force disk after ~1 batch
    frames = [make_frame(10), make_frame(20), make_frame(30)]
    for f in frames:
        sp.spill(f)
    assert sp.rows == 60
    got = read_all(sp.reader())
    assert len(got) == 60
    sp.close()


def test_spiller_global_host_accountant(monkeypatch):
    """The process-wide host accountant (utils.hostmem) caps the SUM of
    all spillers' host tiers: once exhausted, batches overflow to disk
    even under a generous per-spiller budget, and close() releases the
    reservation (VERDICT task 3 / reference spiller.go unbounded disk
    tier)."""
    from bigslice_amd.sliceio.spiller import _DiskBatch
    from bigslice_amd.utils import hostmem
    monkeypatch.setenv("BIGSLICE_HOST_BUDGET_BYTES", "600")
    base = hostmem.used_bytes()
    sp1 = Spiller(host_budget_bytes=1 << 30)
    sp2 = Spiller(host_budget_bytes=1 << 30)
    for sp in (sp1, sp2):
        for _ in range(3):
            sp.spill(make_frame(16))  # 16 rows x 2 cols x 8B = 256 B
    # global budget 600 B admits ~2 host batches total; the rest are
    # disk batches despite the large per-spiller budgets
    disk = sum(isinstance(b, _DiskBatch)
               for sp in (sp1, sp2) for b in sp.batches)
    assert disk >= 4, disk
    assert hostmem.used_bytes() - base <= 600
    assert read_all(sp1.reader()) and read_all(sp2.reader())
    sp1.close()
    sp2.close()
    assert hostmem.used_bytes() == base  # reservations released


def test_frame_reader_chunks():
    f = make_frame(10)
    r = FrameReader(f, chunk=3)
    sizes = [len(b) for b in r]
    assert sizes == [3, 3, 3, 1]


def test_fuzz_codec_sort_partition_roundtrip():
    # randomized multi-dtype frames (reference fuzzFrame-style tests):
    # codec roundtrip, prefix sort order, partition row preservation.
    import io
    import random

    from bigslice_amd.runtime.partition import split_frame

    rng = random.Random(11)
    for trial in range(10):
        n = rng.randint(1, 400)
        cols = []
        dts = [torch.int64, torch.int32, torch.float64, torch.float32,
               torch.bool]
        rng.shuffle(dts)
        ncols = rng.randint(1, 4)
        for dt in dts[:ncols]:
            if dt == torch.bool:
                cols.append(torch.randint(0, 2, (n,)).to(torch.bool))
            elif dt.is_floating_point:
                cols.append(torch.randn(n, dtype=dt))
            else:
                cols.append(torch.randint(-50, 50, (n,), dtype=dt))
        if rng.random() < 0.5:
            cols.append(["s%d" % rng.randint(0, 20) for _ in range(n)])
        prefix = 1
        f = Frame(cols, prefix)
        # codec roundtrip
        buf = io.BytesIO()
        encode_frame(f, buf)
        buf.seek(0)
        g = decode_frame(buf)
        assert g.schema == f.schema and len(g) == n
        for a, b in zip(f.columns, g.columns):
            if isinstance(a, torch.Tensor):
                assert torch.equal(a, b)
            else:
                assert a == b
        # sort by prefix is ordered and row-preserving
        s = f.sort_by_prefix()
        k = s.columns[0]
        if isinstance(k, torch.Tensor) and not k.dtype.is_floating_point:
            assert bool((k[1:] >= k[:-1]).all())
        # partition preserves the multiset of first-column values
        if isinstance(f.columns[0], torch.Tensor) and \
                f.columns[0].dtype == torch.int64 and not f.has_objects:
            parts = split_frame(f, 4, None)
            got = sorted(v for p in parts if p is not None
                         for v in p.columns[0].tolist())
            assert got == sorted(f.columns[0].tolist())


def test_unpicklable_object_column_errors():
    # reference slice_test.go:983 (non-gob-encodable data): encoding a
    # frame with an unpicklable value must raise, and the error must
    # surface from a Cache write (task ERR, not a hang).
    import io
    f = Frame([[lambda: 1, lambda: 2]])
    buf = io.BytesIO()
    with pytest.raises(Exception):
        encode_frame(f, buf)
