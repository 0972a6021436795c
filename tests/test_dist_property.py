"""Randomized distributed-consistency property test: random pipelines
over random data run under the LOCAL executor and the SPMD executor at
random world sizes with random (often tiny) exchange windows — results
must be identical multisets.  The strongest CPU-side de-risk for the
driver's 8-GPU RCCL runs: every seed exercises a different mix of
windows, drain rounds, LPT placements, empty partitions and combine
modes through the exact code the nccl backend drives."""

import os
import random

import pytest
import torch

from tests.test_dist import _free_port, _init, _worker_entry


def build_random_program(seed):
    """Deterministic random slice program + its builder registry."""
    import bigslice_amd as bs

    rng = random.Random(seed)
    nshard = rng.choice([1, 2, 3, 5, 8])
    nrows = rng.choice([0, 1, 7, 1000, 20_000])
    nkeys = rng.choice([1, 3, 17, 300, 5000])
    skew = rng.choice([False, True])

    def build(m):
        def gen(shard, ctx):
            g = torch.Generator().manual_seed(seed * 100 + shard)
            n = nrows
            if skew and shard == 0:
                n = nrows * 5  # one hot shard
            if n == 0:
                return iter(())
            keys = torch.randint(0, nkeys, (n,), dtype=torch.int64,
                                 generator=g)
            if skew:  # zipf-ish hot key
                hot = torch.rand(n, generator=g) < 0.5
                keys = torch.where(hot, torch.zeros_like(keys), keys)
            vals = torch.randint(-50, 50, (n,), dtype=torch.int64,
                                 generator=g)
            return iter([(keys, vals)])
        s = bs.ReaderFunc(m, gen, bs.schema_of(int, int))
        op = rng.choice(["reduce_sum", "reduce_minmax", "reshuffle",
                         "reshard", "cogroup_self", "chain"])
        if op == "reduce_sum":
            return bs.Reduce(s, "sum")
        if op == "reduce_minmax":
            m2 = bs.Map(s, lambda k, v: (k, v, v * 2))
            return bs.Reduce(m2, ("min", "max"))
        if op == "reshuffle":
            return bs.Reshuffle(s)
        if op == "reshard":
            return bs.Reshard(s, rng.choice([1, 2, 7, 16]))
        if op == "cogroup_self":
            a = bs.Filter(s, lambda k, v: v >= 0)
            b = bs.Filter(s, lambda k, v: v < 10)
            return bs.Cogroup(a, b)
        red = bs.Reduce(bs.Map(s, lambda k, v: (k, v + 1)), "sum")
        return bs.Map(red, lambda k, v: (k, v * 3))

    return nshard, build


def canon(rows):
    def c(v):
        return tuple(sorted(v)) if isinstance(v, list) else v
    return sorted(tuple(c(v) for v in r) if isinstance(r, tuple) else r
                  for r in rows)


def _prop_local(seeds, q):
    import bigslice_amd as bs
    out = {}
    sess = bs.start(parallelism=3, device="cpu")
    for seed in seeds:
        nshard, build = build_random_program(seed)
        res = sess.run(bs.func(build), nshard)
        out[seed] = canon(res.scan())
        res.discard()
    q.put(("local", out))


def _prop_dist(rank, world, port, q, seeds=None):
    _init(rank, world, port)
    import bigslice_amd as bs
    out = {}
    sess = bs.start(distributed=True, device="cpu")
    for seed in seeds:
        nshard, build = build_random_program(seed)
        res = sess.run(bs.func(build), nshard)
        out[seed] = canon(res.scan())
        res.discard()
    q.put((rank, out))


@pytest.mark.parametrize("world,wseed", [(2, 0), (4, 1), (8, 2)])
def test_random_programs_consistent(world, wseed):
    import functools

    import torch.multiprocessing as mp
    rng = random.Random(1234 + wseed)
    seeds = [rng.randrange(10**6) for _ in range(6)]
    ctx = mp.get_context("spawn")
    lq = ctx.SimpleQueue()
    lp = ctx.Process(target=_prop_local, args=(seeds, lq))
    lp.start()
    tag, local = lq.get()
    lp.join(180)
    assert lp.exitcode == 0

    os.environ["BIGSLICE_EXCHANGE_WINDOW_BYTES"] = str(
        random.Random(wseed).choice([512, 8192, 1 << 20]))
    try:
        port = _free_port()
        q = ctx.SimpleQueue()
        worker = functools.partial(_prop_dist, seeds=seeds)
        procs = [ctx.Process(target=_worker_entry,
                             args=(worker, r, world, port, q))
                 for r in range(world)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world):
            rank, out = q.get()
            results[rank] = out
        for p in procs:
            p.join(180)
            assert p.exitcode == 0
    finally:
        os.environ.pop("BIGSLICE_EXCHANGE_WINDOW_BYTES", None)
    for seed in seeds:
        assert results[0][seed] == local[seed], seed
        for r in range(1, world):
            assert results[r][seed] == []
