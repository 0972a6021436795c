"""Domain-count demo in the shape of the reference's cmd/urls
(cmd/urls/urls.go:37-96): ReaderFunc over CSV records -> Map(extract
domain) -> Reduce(sum) -> top-N scan.  There is no network in this
environment, so the CSV is synthesized; point `open_fn` at a real file
to run it on data.

  python examples/urls.py [--rows 100000]
"""

import argparse
import io
import os
import random
import sys
from urllib.parse import urlparse

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bigslice_amd as bs
from bigslice_amd.utils.status import top_n

_DOMAINS = ["example.com", "grail.com", "amd.com", "github.com",
            "news.site", "blog.net"]


def synthesize_csv(rows: int, seed: int = 7) -> str:
    rng = random.Random(seed)
    buf = io.StringIO()
    for i in range(rows):
        d = rng.choice(_DOMAINS)
        buf.write(f"{i},2026-09-13,https://{d}/p/{rng.randrange(100)}\n")
    return buf.getvalue()


def build(nshard, text):
    lines = bs.ScanReader(nshard, lambda: io.StringIO(text))

    def domain(line):
        url = line.split(",")[2]
        return (urlparse(url).netloc, 1)

    pairs = bs.Map(lines, domain, out_schema=(str, int), rowwise=True)
    return bs.Reduce(pairs, "sum")


domain_counts = bs.func(build)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=100_000)
    ap.add_argument("--top", type=int, default=10)
    args = ap.parse_args()
    text = synthesize_csv(args.rows)
    sess = bs.start(parallelism=8)
    counts = dict(sess.run(domain_counts, 8, text).scan())
    for dom, n in top_n(counts, args.top):
        print(f"{n:10d}  {dom}")


if __name__ == "__main__":
    main()
