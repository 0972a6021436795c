"""The canonical wordcount (reference docs/index.md:93-155) as a
standalone example; see also bigslice_amd.tools.wordcount.

  python examples/wordcount.py FILE [--shards N]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bigslice_amd as bs


def build(nshard, path):
    lines = bs.ScanReader(nshard, lambda: open(path))
    words = bs.Flatmap(lines, lambda s: [(w,) for w in s.split()],
                       out_schema=(str,), rowwise=True)
    pairs = bs.Map(words, lambda w: (w, 1), out_schema=(str, int),
                   rowwise=True)
    return bs.Reduce(pairs, "sum")


wordcount = bs.func(build)

if __name__ == "__main__":
    path = sys.argv[1]
    sess = bs.start()
    for word, count in sorted(sess.run(wordcount, 8, path).scan(),
                              key=lambda kv: -kv[1])[:20]:
        print(f"{count:8d}  {word}")
