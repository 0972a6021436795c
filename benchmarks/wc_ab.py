"""Wordcount A/B on device: host string-keyed Reduce vs gpu_wordcount
(K17 device dictionary ids + device count)."""
import sys, time
sys.path.insert(0, ".")
import bigslice_amd as bs
from bigslice_amd import recipes
from collections import Counter
import random

NLINES = int(sys.argv[1]) if len(sys.argv) > 1 else 500_000
rng = random.Random(1)
VOCAB = [f"w{i:05d}" for i in range(20_000)]
lines = [" ".join(rng.choice(VOCAB) for _ in range(8))
         for _ in range(NLINES)]
ref = Counter(w for ln in lines for w in ln.split())

import torch
dev = "cuda:0" if torch.cuda.is_available() else "cpu"
sess = bs.start(parallelism=4, device=dev)

def host_path():
    def build(nshard, token):
        src = bs.ScanReader(nshard, lambda: iter(lines))
        toks = bs.Flatmap(src, lambda col: (
            [w for s in col for w in s.split()],), out_schema=(str,))
        pairs = bs.Map(toks, lambda col: (col, [1] * len(col)),
                       out_schema=(str, int))
        return bs.Reduce(pairs, "sum")
    fv = bs.func(build)
    res = sess.run(fv, 4, 0)
    out = dict(res.scan())
    res.discard()
    return out

def bytes_path():
    """First-class BYTES column pipeline: tokens pack into a device
    BytesColumn at the source; hashing, partitioning, shuffle and the
    keyed count all run device-resident (VERDICT task 5)."""
    from bigslice_amd.frame import BytesColumn

    def build(nshard, token):
        def gen(shard, ctx):
            ws = [w for i, ln in enumerate(lines)
                  if i % nshard == shard for w in ln.split()]
            col = BytesColumn.from_list(ws, dev)
            ones = torch.ones(len(ws), dtype=torch.int64, device=dev)
            yield (col, ones)
        src = bs.ReaderFunc(nshard, gen, bs.schema_of(bytes, int))
        return bs.Reduce(src, "sum")
    fv = bs.func(build)
    res = sess.run(fv, 4, 0)
    out = {k.decode(): v for k, v in res.scan()}
    res.discard()
    return out


for name, fn in (("host-string", host_path),
                 ("bytes-col", bytes_path),
                 ("gpu-dict", lambda: recipes.gpu_wordcount(
                     sess, 4, lines, dev))):
    got = fn()
    assert got == dict(ref), name
    t0 = time.perf_counter()
    got = fn()
    dt = time.perf_counter() - t0
    nwords = NLINES * 8
    print(f"{name}: {dt*1000:8.1f} ms  {nwords/dt/1e6:6.2f} M words/s")
