"""The canonical wordcount demo (reference cmd/urls + docs/index.md
wordcount): ScanReader -> Flatmap(split) -> Map((w,1)) -> Reduce(+).

  python -m bigslice_amd.tools.wordcount [FILE] [--shards N] [--top K]
"""

from __future__ import annotations

import argparse
import sys

import bigslice_amd as bs


def build_wordcount(nshard, path):
    def lines():
        with open(path) as fp:
            yield from fp
    lines_slice = bs.ScanReader(nshard, lines)
    words = bs.Flatmap(lines_slice,
                       lambda s: [(w,) for w in s.split()],
                       out_schema=(str,), rowwise=True)
    counts = bs.Map(words, lambda w: (w, 1), out_schema=(str, int),
                    rowwise=True)
    return bs.Reduce(counts, "sum")


wordcount = bs.func(build_wordcount)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("file")
    ap.add_argument("--shards", type=int, default=8)
    ap.add_argument("--top", type=int, default=20)
    args, rest = ap.parse_known_args()
    sess, _ = bs.sliceconfig.parse(rest)
    res = sess.run(wordcount, args.shards, args.file)
    rows = sorted(res.scan(), key=lambda kv: (-kv[1], kv[0]))
    for w, c in rows[: args.top]:
        print(f"{c:8d}  {w}")


if __name__ == "__main__":
    main()
