"""First-class device varlen BYTES columns (frame.BytesColumn):
frame ops, codec, partition colocation, keyed reduce, and the
distributed exchange (VERDICT task 5; reference
frame/ops_builtin.go:143-164 string ops)."""

import io

import pytest
import torch

import bigslice_amd as bs
from bigslice_amd.frame import BytesColumn, Frame
from bigslice_amd.hashing import murmur3_bytes
from bigslice_amd.schema import BYTES
from bigslice_amd.sliceio import codec

WORDS = ["a", "bb", "ccc", "dd", "a", "eeeee", "", "bb"] * 25


def make_col(words=None):
    return BytesColumn.from_list(words or WORDS)


def test_bytes_column_ops():
    c = make_col()
    assert len(c) == len(WORDS)
    assert c.tolists() == [w.encode() for w in WORDS]
    s = c[3:7]
    assert s.tolists() == [w.encode() for w in WORDS[3:7]]
    sel = c.select(torch.tensor([7, 0, 5]))
    assert sel.tolists() == [b"bb", b"a", b"eeeee"]
    # hash is bit-identical to the host murmur3 over raw bytes
    h = c.hash32(11).tolist()
    assert h[:4] == [murmur3_bytes(w.encode(), 11) for w in WORDS[:4]]


def test_bytes_frame_roundtrips():
    f = Frame([make_col(), torch.arange(len(WORDS),
                                        dtype=torch.int64)], 1)
    assert f.schema.dtypes[0] == BYTES
    assert not f.has_objects  # device-capable column
    # codec roundtrip incl. a sliced (non-zero-based) column
    buf = io.BytesIO()
    codec.encode_frame(f.slice(5, 60), buf)
    buf.seek(0)
    back = codec.decode_frame(buf)
    assert back.column_lists() == f.slice(5, 60).column_lists()
    # concat + sort by (dictionary-id) prefix groups equal keys
    cc = Frame.concat([f.slice(0, 10), f.slice(10, 30)])
    srt = cc.sort_by_prefix()
    ks = srt.columns[0].tolists()
    assert sorted(ks) == sorted(k.encode() for k in WORDS[:30])
    for i in range(1, len(ks)):  # equal keys adjacent
        if ks[i] in ks[:i]:
            assert ks[i - 1] == ks[i] or ks[i] not in ks[i - 1:i]


def test_bytes_partition_colocation():
    # same key -> same partition; rows preserved (the reshuffle
    # colocation invariant, reshuffle_test.go:23-84)
    from bigslice_amd.runtime.partition import split_frame
    f = Frame([make_col(), torch.arange(len(WORDS),
                                        dtype=torch.int64)], 1)
    parts = split_frame(f, 5)
    seen = {}
    total = 0
    for pi, pf in enumerate(parts):
        if pf is None:
            continue
        total += len(pf)
        for k in pf.columns[0].tolists():
            assert seen.setdefault(k, pi) == pi, k
    assert total == len(WORDS)
    assert len(seen) == len(set(WORDS))


def test_bytes_keyed_reduce_local():
    def build(m):
        def gen(shard, ctx):
            yield (BytesColumn.from_list(WORDS),
                   torch.ones(len(WORDS), dtype=torch.int64))
        src = bs.ReaderFunc(m, gen, bs.schema_of(bytes, int))
        return bs.Reduce(src, "sum")

    fv = bs.func(build)
    sess = bs.start(parallelism=2, device="cpu")
    res = sess.run(fv, 3)
    got = {k: v for k, v in res.scan()}
    want = {}
    for w in WORDS * 3:
        want[w.encode()] = want.get(w.encode(), 0) + 1
    assert got == want


def _bytes_dist_worker(rank, world, port, q):
    from tests.test_dist import _init
    _init(rank, world, port)
    import bigslice_amd as bs
    from bigslice_amd.frame import BytesColumn

    def build(m):
        def gen(shard, ctx):
            ws = [f"w{(i * 7 + shard) % 41:03d}" for i in range(2000)]
            yield (BytesColumn.from_list(ws),
                   torch.ones(len(ws), dtype=torch.int64))
        src = bs.ReaderFunc(m, gen, bs.schema_of(bytes, int))
        return bs.Reduce(src, "sum")

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu")
    res = sess.run(fv, 4)
    q.put((rank, sorted(res.scan())))


@pytest.mark.parametrize("world", [2, 3])
def test_bytes_dist_exchange(world):
    from tests.test_exchange import _run_workers
    results = _run_workers(_bytes_dist_worker, world=world,
                           env={"BIGSLICE_EXCHANGE_WINDOW_BYTES":
                                "4096"})
    want = {}
    for shard in range(4):
        for i in range(2000):
            k = f"w{(i * 7 + shard) % 41:03d}".encode()
            want[k] = want.get(k, 0) + 1
    assert results[0] == sorted(want.items())
