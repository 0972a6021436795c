"""The canonical wordcount demo (reference cmd/urls + docs/index.md
wordcount): ScanReader -> Flatmap(split) -> Map((w,1)) -> Reduce(+).

  python -m bigslice_amd.tools.wordcount [FILE] [--shards N] [--top K]
"""

from __future__ import annotations

import argparse

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


def build_fast_wordcount(nshard, path):
    from bigslice_amd.recipes import fast_wordcount

    def lines():
        with open(path) as fp:
            yield from fp
    return fast_wordcount(nshard, lines)


wordcount = bs.func(build_wordcount)
wordcount_fast = bs.func(build_fast_wordcount)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("file")
    ap.add_argument("--shards", type=int, default=8)
    ap.add_argument("--top", type=int, default=20)
    ap.add_argument("--fast", action="store_true",
                    help="dictionary-encoded device-reduce recipe "
                         "(see recipes.fast_wordcount verdict)")
    args, rest = ap.parse_known_args()
    sess, _ = bs.sliceconfig.parse(rest)
    fv = wordcount_fast if args.fast else wordcount
    res = sess.run(fv, args.shards, args.file)
    rows = sorted(res.scan(), key=lambda kv: (-kv[1], kv[0]))
    for w, c in rows[: args.top]:
        print(f"{c:8d}  {w}")


if __name__ == "__main__":
    main()
