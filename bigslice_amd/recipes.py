"""Recipes: composed patterns over the public API.

fast_wordcount: the reference's headline wordcount
(docs/index.md:93-155) restructured so the aggregation runs on the
GPU: the rowwise tokenizer (unavoidably host-side in Python)
dictionary-encodes words to 64-bit murmur hashes AND emits each
shard's (id -> word) mapping only once per new word, so the count
Reduce runs device-native over int64 ids and only the small distinct
mapping flows on the host path.

MEASURED verdict (3M words, 20k vocabulary, 1x MI355X): the plain
string-keyed wordcount wins against fast_wordcount's HOST-side id
encoding (3.5 s vs 8.6 s) — Python tokenization dominates both and a
C-level dict count beats the extra per-word hash pass.  The K17
DEVICE-side variant flips this: gpu_wordcount (batch byte packing +
device murmur ids + device count) measures 7.5 M words/s vs 5.2 M for
the string-keyed path at 4M words (benchmarks/wc_ab.py).

Collision note: ids are murmur3-64 (two murmur3-32 lanes); distinct
words colliding would merge counts with probability ~V^2/2^65 —
negligible for real vocabularies, but this recipe is opt-in; the exact
string-keyed Reduce remains the default wordcount.
"""

from __future__ import annotations

from typing import Callable, Iterable

from . import hashing
from .ops import Cogroup, Flatmap, Map, Reduce, ScanReader
from .ops.slice_base import Slice, materialize


def _word_id(w: str) -> int:
    b = w.encode("utf-8")
    lo = hashing.murmur3_bytes(b, 0)
    hi = hashing.murmur3_bytes(b, 0x9E3779B9)
    v = (hi << 32) | lo
    return v - (1 << 64) if v >= (1 << 63) else v


def fast_wordcount(nshard: int, open_fn: Callable[[], Iterable[str]]
                   ) -> Slice:
    """Slice<word: str, count: int> with the count aggregation on
    device.  Build inside a bigslice_amd.func."""
    lines = ScanReader(nshard, open_fn)

    # per-shard tokenizer with a new-word side channel: emits
    # (id, count_marker, word_or_None); the word rides along only on
    # first sight within the shard (fresh seen-set per shard reader)
    def tok_factory():
        seen = set()

        def tok(line: str):
            out = []
            for w in line.split():
                i = _word_id(w)
                if i in seen:
                    out.append((i, 1, None))
                else:
                    seen.add(i)
                    out.append((i, 1, w))
            return out
        return tok

    tokens = Flatmap(lines, None, out_schema=(int, int, object),
                     rowwise=True, fn_factory=tok_factory)
    # two branches consume tokens; materialize so the (host-side)
    # tokenization pass runs once, not once per branch
    materialize(tokens)

    counts = Reduce(Map(tokens, lambda i, c, w: (i, c),
                        out_schema=(int, int)), "sum")
    names = Reduce(
        Map(Flatmap(tokens,
                    lambda i, c, w: [(i, w)] if w is not None else [],
                    out_schema=(int, object), rowwise=True),
            lambda i, w: (i, w), out_schema=(int, object), rowwise=True),
        lambda a, b: a)  # arbitrary-pick combine (values identical)

    joined = Cogroup(names, counts)

    def resolve(i, words, cs):
        return (words[0], sum(cs))

    return Map(joined, resolve, out_schema=(str, int), rowwise=True)


# -- gpu_wordcount: device dictionary ids via K17 ------------------------

_GPU_WC_RUNS = {}
_GPU_WC_SEQ = [0]


def _build_gpu_wordcount(nshard, token):
    import torch

    from . import strings
    from .ops import ReaderFunc
    from .ops.elementwise import schema_of

    lines_list, device, state = _GPU_WC_RUNS[token]

    def gen(shard, ctx):
        words_all = []
        state[shard] = words_all
        base = shard << 40

        def emit(buf):
            ids = strings.string_ids(buf, device)
            start = base + len(words_all)
            words_all.extend(buf)
            n = len(buf)
            ones = torch.ones(n, dtype=torch.int64, device=ids.device)
            idx = torch.arange(start, start + n, dtype=torch.int64,
                               device=ids.device)
            return (ids, ones, idx)

        buf = []
        for li in range(shard, len(lines_list), nshard):
            buf.extend(lines_list[li].split())
            if len(buf) >= 500_000:
                yield emit(buf)
                buf = []
        if buf:
            yield emit(buf)

    src = ReaderFunc(nshard, gen, schema_of(int, int, int, prefix=1))
    from .ops import Reduce
    return Reduce(src, ("sum", "min"))


def _register_gpu_wc():
    from .runtime.session import func
    return func(_build_gpu_wordcount)


_gpu_wc_func = _register_gpu_wc()


def gpu_wordcount(sess, nshard: int, lines, device) -> dict:
    """word -> count with hashing AND counting on device: the host
    only tokenizes and packs bytes; K17 (csrc/strings.hip) computes
    64-bit dictionary ids on the GPU and the count runs as a
    device-native int64 Reduce(("sum","min")) — "min" keeps a
    representative row index per id so readback recovers the word
    without any host-side id map on the hot path.

    Single-process sessions (the retained per-shard token lists are
    shared-memory state); see fast_wordcount's side-channel pattern
    for the distributed variant.  Collision note as fast_wordcount."""
    _GPU_WC_SEQ[0] += 1
    token = _GPU_WC_SEQ[0]
    state = {}
    _GPU_WC_RUNS[token] = (list(lines), device, state)
    try:
        res = sess.run(_gpu_wc_func, nshard, token)
        out = {}
        mask = (1 << 40) - 1
        for _id, cnt, gidx in res.scan():
            out[state[gidx >> 40][gidx & mask]] = cnt
        res.discard()
    finally:
        _GPU_WC_RUNS.pop(token, None)
    return out
