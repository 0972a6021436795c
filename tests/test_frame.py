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


def test_frame_reader_chunks():
    f = make_frame(10)
    r = FrameReader(f, chunk=3)
    sizes = [len(b) for b in r]
    assert sizes == [3, 3, 3, 1]
