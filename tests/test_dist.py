"""Distributed executor tests: SPMD over gloo, world_size=2, CPU.

These exercise the same code paths the RCCL/GPU path uses (phase
schedule, registry check, and — since gloo supports alltoall — the
tensor all-to-allv bucket exchange itself); only the transport differs.
Object/string-column pipelines fall back to the object exchange.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _reduce_worker(rank, world, port, q):
    _init(rank, world, port)
    import bigslice_amd as bs

    def build(nshard):
        def gen(shard, ctx):
            keys = torch.arange(100, dtype=torch.int64) % 7
            vals = torch.full((100,), shard + 1, dtype=torch.int64)
            yield (keys, vals)
        src = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
        return bs.Reduce(src, "sum")

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu")
    res = sess.run(fv, 4)
    rows = sorted(res.scan())
    q.put((rank, rows))


def _wordcount_worker(rank, world, port, q):
    _init(rank, world, port)
    import bigslice_amd as bs
    text = ["a b a", "c b a", "d"]

    def build(nshard):
        lines = bs.ScanReader(nshard, lambda: iter(text))
        words = bs.Flatmap(lines, lambda s: [(w,) for w in s.split()],
                           out_schema=(str,), rowwise=True)
        counts = bs.Map(words, lambda w: (w, 1), out_schema=(str, int),
                        rowwise=True)
        return bs.Reduce(counts, "sum")

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu")
    res = sess.run(fv, 3)
    q.put((rank, sorted(res.scan())))


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_entry(fn, rank, world, port, q):
    """Run a dist worker, then tear the process group down cleanly:
    ranks exiting at different times with a live gloo group can
    SIGABRT in its helper threads (same race as bench.py's teardown),
    tripping the exitcode assertion below."""
    try:
        fn(rank, world, port, q)
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            try:
                dist.barrier()
            except Exception:
                pass
            dist.destroy_process_group()


def _run_workers(fn, world=2, port=None):
    if port is None:
        port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_entry,
                         args=(fn, r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, rows = q.get()
        results[rank] = rows
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    return results


def test_dist_reduce_gloo():
    results = _run_workers(_reduce_worker)
    # each of 4 shards contributes 100 rows over 7 keys; values shard+1
    # total per key: sum over shards of (count of key in shard) * (shard+1)
    keys = (torch.arange(100, dtype=torch.int64) % 7)
    expect = {}
    for shard in range(4):
        for k in keys.tolist():
            expect[k] = expect.get(k, 0) + (shard + 1)
    want = sorted(expect.items())
    assert results[0] == want
    assert results[1] == []  # non-root ranks see no rows


def test_dist_wordcount_gloo():
    results = _run_workers(_wordcount_worker)
    assert results[0] == [("a", 3), ("b", 2), ("c", 1), ("d", 1)]


def _matrix_worker(rank, world, port, q):
    """Several operators through the SPMD executor in one process pair
    (the reference's dual-executor matrix, slice_test.go:64-67)."""
    _init(rank, world, port)
    import bigslice_amd as bs

    results = {}

    # 1. Reshuffle: rows preserved
    def b_reshuffle(m):
        keys = torch.arange(200, dtype=torch.int64) % 11
        vals = torch.arange(200, dtype=torch.int64)
        return bs.Reshuffle(bs.Const(m, keys, vals))
    fv1 = bs.func(b_reshuffle)

    # 2. Cogroup
    def b_cogroup(m):
        a = bs.Const(m, torch.tensor([1, 2, 1], dtype=torch.int64),
                     torch.tensor([10, 20, 30], dtype=torch.int64))
        b = bs.Const(m, torch.tensor([2, 3], dtype=torch.int64),
                     torch.tensor([5, 6], dtype=torch.int64))
        return bs.Cogroup(a, b)
    fv2 = bs.func(b_cogroup)

    # 3. Fold
    def b_fold(m):
        keys = torch.tensor([1, 1, 2, 2, 2], dtype=torch.int64)
        vals = torch.tensor([1, 2, 3, 4, 5], dtype=torch.int64)
        return bs.Fold(bs.Const(m, keys, vals),
                       lambda acc, v: (acc or 0) + v, out_schema=(int,))
    fv3 = bs.func(b_fold)

    # 4. Iterative: Result reuse
    def b_map_prev(prev):
        return bs.Map(prev, lambda k, v: (k, v * 2))
    fv4 = bs.func(b_map_prev)

    sess = bs.start(distributed=True, device="cpu")
    r1 = sess.run(fv1, 3)
    results["reshuffle"] = sorted(r1.scan())
    r2 = sess.run(fv2, 2)
    results["cogroup"] = sorted(
        (k, sorted(a), sorted(b)) for k, a, b in r2.scan())
    r3 = sess.run(fv3, 2)
    results["fold"] = sorted(r3.scan())
    r4 = sess.run(fv4, r3)
    results["iterative"] = sorted(r4.scan())
    q.put((rank, results))


def test_dist_operator_matrix_gloo():
    results = _run_workers(_matrix_worker)
    r0 = results[0]
    keys = (torch.arange(200) % 11).tolist()
    assert r0["reshuffle"] == sorted(zip(keys, range(200)))
    assert r0["cogroup"] == [(1, [10, 30], []), (2, [20], [5]),
                             (3, [], [6])]
    assert r0["fold"] == [(1, 3), (2, 12)]
    assert r0["iterative"] == [(1, 6), (2, 24)]


def _error_worker(rank, world, port, q):
    """One rank's UDF raises: EVERY rank must raise (collective error
    surfacing), and the session must stay usable for a subsequent run."""
    _init(rank, world, port)
    import bigslice_amd as bs

    def build(nshard):
        def gen(shard, ctx):
            if shard == 1:
                raise ValueError("boom on shard 1")
            yield (torch.arange(5, dtype=torch.int64),)
        return bs.ReaderFunc(nshard, gen, bs.schema_of(int))
    fv_bad = bs.func(build)
    fv_ok = bs.func(lambda: bs.Const(2, torch.arange(4,
                                                     dtype=torch.int64)))
    sess = bs.start(distributed=True, device="cpu")
    raised = False
    try:
        sess.run(fv_bad, 4)
    except Exception:
        raised = True
    res = sess.run(fv_ok)  # session still usable after the failure
    rows = sorted(res.scan())
    q.put((rank, raised, rows))


def test_dist_error_propagates_to_all_ranks():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_worker_entry,
                         args=(_error_worker, r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, raised, rows = q.get()
        results[rank] = (raised, rows)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    assert results[0][0] and results[1][0]  # both ranks raised
    assert results[0][1] == [0, 1, 2, 3]


def _restart_worker(rank, world, port, q):
    """Simulated job restart: a fresh session over the same checkpoint
    dir skips completed phases (restart-based rank-loss recovery)."""
    import tempfile
    _init(rank, world, port)
    import bigslice_amd as bs

    counter = {"n": 0}

    def build(nshard):
        def gen(shard, ctx):
            counter["n"] += 1
            keys = torch.arange(50, dtype=torch.int64) % 5
            yield (keys, torch.ones_like(keys))
        return bs.Reduce(bs.ReaderFunc(nshard, gen,
                                       bs.schema_of(int, int)), "sum")

    fv = bs.func(build)
    ckpt = "/tmp/bigslice_ckpt_test"
    sess1 = bs.start(distributed=True, device="cpu",
                     checkpoint_dir=ckpt)
    r1 = sess1.run(fv, 4)
    rows1 = sorted(r1.scan())
    n_first = counter["n"]

    # "restart": a brand-new session + executor over the same dir
    sess2 = bs.start(distributed=True, device="cpu",
                     checkpoint_dir=ckpt)
    r2 = sess2.run(fv, 4)
    rows2 = sorted(r2.scan())
    q.put((rank, (rows1, rows2, n_first, counter["n"])))


def test_dist_checkpoint_restart():
    import shutil
    shutil.rmtree("/tmp/bigslice_ckpt_test", ignore_errors=True)
    results = _run_workers(_restart_worker)
    rows1, rows2, n1, n2 = results[0]
    # each of 4 shards contributes 50 rows over 5 keys (10 each)
    expect = sorted((k, 40) for k in range(5))
    assert rows1 == expect
    assert rows2 == expect  # restart serves from checkpoints
    assert n2 == n1  # no shard recomputation on restart


def _exchange3_worker(rank, world, port, q):
    """Drive Comm.exchange_buckets directly at world 3 with asymmetric
    buckets, both metadata planes.  World 3 breaks the rank symmetry
    that world 2 can hide (the round-end scaling bench runs N=4/8 on
    this same code; only the transport differs under RCCL)."""
    _init(rank, world, port)
    from bigslice_amd.parallel.comm import Comm
    from bigslice_amd.frame import Frame
    from bigslice_amd import schema_of

    comm = Comm(rank, world, "cpu")
    assert comm.tensor_exchange_ok  # gloo drives the tensor path
    schema = schema_of(int, int)
    names = [f"t{d}" for d in range(world)]
    name_index = {n: i for i, n in enumerate(names)}

    def mk(dest, j):
        # rows encode (src, dest, j) so routing is fully checkable
        n = (rank + 1) * (j + 1)  # asymmetric sizes
        k = torch.full((n,), rank * 100 + dest * 10 + j,
                       dtype=torch.int64)
        return Frame([k, torch.arange(n, dtype=torch.int64)], 1)

    # rank r sends (r+dest) % 2 + 1 buckets to each dest
    send = []
    for dest in range(world):
        nb = (rank + dest) % 2 + 1
        send.append([(names[dest], dest * 4 + j, mk(dest, j))
                     for j in range(nb)])

    for kw in ({"name_index": name_index, "index_name": names}, {}):
        got = comm.exchange_buckets(send, schema, **kw)
        # expect: from each src, (src+rank)%2+1 buckets for me
        expect = []
        for src in range(world):
            nb = (src + rank) % 2 + 1
            for j in range(nb):
                expect.append((names[rank], rank * 4 + j,
                               src * 100 + rank * 10 + j,
                               (src + 1) * (j + 1)))
        summary = sorted((t, p, int(f.columns[0][0]), len(f))
                         for (t, p, f) in got)
        assert summary == sorted(expect), (kw, summary)
    q.put((rank, "ok"))


def test_dist_tensor_exchange_world3():
    results = _run_workers(_exchange3_worker, world=3)
    assert all(v == "ok" for v in results.values())


def _narrow_phase_worker(rank, world, port, q):
    """nshard < world: rank 2 owns no shards and no partitions of the
    2-shard phases.  The checkpoint-skip decision must be collective or
    that rank deadlocks the others' phase collectives (regression:
    vacuously-true _phase_checkpointed at world 3)."""
    _init(rank, world, port)
    import bigslice_amd as bs

    def b_cogroup(m):
        a = bs.Const(m, torch.tensor([1, 2, 1], dtype=torch.int64),
                     torch.tensor([10, 20, 30], dtype=torch.int64))
        b = bs.Const(m, torch.tensor([2, 3], dtype=torch.int64),
                     torch.tensor([5, 6], dtype=torch.int64))
        return bs.Cogroup(a, b)

    fv = bs.func(b_cogroup)
    sess = bs.start(distributed=True, device="cpu")
    r = sess.run(fv, 2)
    q.put((rank, sorted((k, sorted(x), sorted(y))
                        for k, x, y in r.scan())))


def test_dist_nshard_less_than_world():
    results = _run_workers(_narrow_phase_worker, world=3)
    assert results[0] == [(1, [10, 30], []), (2, [20], [5]),
                          (3, [], [6])]


def _compile_env_worker(rank, world, port, q):
    """Driver-broadcast CompileEnv (exec/compile.go:125-184): only rank
    0 may stat the filesystem for cache decisions; other ranks compile
    against the sealed broadcast env.  Non-zero ranks get a poisoned
    ShardCache.present to prove they never probe locally."""
    import shutil
    _init(rank, world, port)
    import bigslice_amd as bs
    from bigslice_amd.ops import cache as cache_mod

    prefix = "/tmp/bigslice_envtest/c"
    if rank == 0:
        shutil.rmtree("/tmp/bigslice_envtest", ignore_errors=True)
    else:
        # a non-zero rank consulting the filesystem is the divergent-
        # graph bug this feature exists to prevent
        def poisoned(self):
            raise AssertionError(
                "rank != 0 computed cache decisions locally")
        cache_mod.ShardCache.present = poisoned

    computed = {"n": 0}

    def build(m):
        def gen(shard, ctx):
            computed["n"] += 1
            keys = torch.arange(40, dtype=torch.int64) % 4
            yield (keys, torch.ones_like(keys))
        src = bs.ReaderFunc(m, gen, bs.schema_of(int, int))
        return bs.Reduce(bs.Cache(src, prefix), "sum")

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu")
    # barrier so rank 0's rmtree lands before anyone compiles
    import torch.distributed as dist
    dist.barrier()
    r1 = sess.run(fv, 4)
    rows1 = sorted(r1.scan())
    n_first = computed["n"]
    dist.barrier()  # all cache writes visible before the second compile
    r2 = sess.run(fv, 4)
    rows2 = sorted(r2.scan())
    q.put((rank, (rows1, rows2, n_first, computed["n"])))


def test_dist_compile_env_broadcast():
    results = _run_workers(_compile_env_worker)
    expect = sorted((k, 40) for k in range(4))
    rows1, rows2, n1, n2 = results[0]
    assert rows1 == expect
    assert rows2 == expect
    assert n2 == n1  # second run read caches, no shard recompute
    # rank 1 also recomputed nothing the second time
    assert results[1][3] == results[1][2]


def _registry_diff_worker(rank, world, port, q):
    """A divergent Func registry must fail with a location-level diff
    naming the extra Func (func.go:276-343), not just digests."""
    _init(rank, world, port)
    import bigslice_amd as bs
    from bigslice_amd.runtime.session import registry_digest

    bs.func(lambda: bs.Const(1, torch.arange(2, dtype=torch.int64)))
    if rank == 1:  # the divergence
        extra = bs.func(
            lambda: bs.Const(1, torch.arange(3, dtype=torch.int64)))
        assert extra is not None
    from bigslice_amd.parallel.comm import Comm
    comm = Comm(rank, world, "cpu")
    try:
        comm.check_registry(registry_digest())
        q.put((rank, None))
    except RuntimeError as e:
        q.put((rank, str(e)))


def test_dist_registry_mismatch_location_diff():
    results = _run_workers(_registry_diff_worker)
    for rank in (0, 1):
        msg = results[rank]
        assert msg is not None, "mismatch not detected"
        # the unified diff marks rank 1's extra registration line
        assert "rank 1 Func registry" in msg
        assert any(line.startswith("+") and "test_dist" in line
                   for line in msg.splitlines()), msg


def _float_worker(rank, world, port, q):
    """Mixed-dtype (int64 keys, float32 values) through the tensor
    exchange: per-column all_to_all_single must handle heterogeneous
    column dtypes."""
    _init(rank, world, port)
    import bigslice_amd as bs

    def build(m):
        keys = torch.arange(999, dtype=torch.int64) % 13
        vals = keys.to(torch.float32) + 0.5
        return bs.Reduce(bs.Const(m, keys, vals), "sum")

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu")
    q.put((rank, dict(sess.run(fv, 4).scan())))


def test_dist_mixed_dtype_exchange():
    results = _run_workers(_float_worker)
    keys = torch.arange(999, dtype=torch.int64) % 13
    vals = keys.to(torch.float32) + 0.5
    for k in range(13):
        expect = float(vals[keys == k].sum())
        assert abs(results[0][k] - expect) < 1e-2
