"""Recipes: composed patterns over the public API.

fast_wordcount: the reference's headline wordcount
(docs/index.md:93-155) restructured so the aggregation runs on the
GPU: the rowwise tokenizer (unavoidably host-side in Python)
dictionary-encodes words to 64-bit murmur hashes AND emits each
shard's (id -> word) mapping only once per new word, so the count
Reduce runs device-native over int64 ids and only the small distinct
mapping flows on the host path.

MEASURED verdict (3M words, 20k vocabulary, 1x MI355X): the plain
string-keyed wordcount wins (3.5 s vs 8.6 s) — Python tokenization
dominates both, and a C-level dict count is cheaper than the extra
encode pass.  Use this pattern when the per-key aggregation itself is
heavy (many value columns, large key counts, downstream device
compute), not for simple counting.

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
