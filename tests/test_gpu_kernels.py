"""GPU kernel numerics tests: each HIP kernel vs a plain torch/numpy
reference on the same data."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def kernels():
    from bigslice_amd import kernels as k
    assert k.have_extension(), "HIP extension must be built in-tree"
    return k


def test_partition_kernel_matches_host(kernels):
    from bigslice_amd.frame import Frame
    from bigslice_amd.runtime.partition import partition_ids, split_frame
    n, nparts = 1_000_000, 8
    keys = torch.randint(-2**62, 2**62, (n,), dtype=torch.int64)
    vals = torch.randn(n, dtype=torch.float64)
    f_cpu = Frame([keys, vals], prefix=1)
    f_gpu = f_cpu.to("cuda:0")

    parts_gpu = kernels.partition_frame(f_gpu, nparts, None)
    # host oracle
    pids = partition_ids(f_cpu, nparts)
    for p in range(nparts):
        want_keys = keys[pids == p]
        got = parts_gpu[p]
        if got is None:
            assert want_keys.numel() == 0
            continue
        gk = got.columns[0].cpu()
        assert gk.shape[0] == want_keys.shape[0]
        # row order within a partition is unspecified: compare sorted
        assert torch.equal(gk.sort().values, want_keys.sort().values)
        # rows stay intact (key,val) pairs
        gv = got.columns[1].cpu()
        want_pairs = sorted(zip(keys[pids == p].tolist(),
                                vals[pids == p].tolist()))
        got_pairs = sorted(zip(gk.tolist(), gv.tolist()))
        assert got_pairs == want_pairs


def test_groupby_kernel_matches_torch(kernels):
    n, nkeys = 2_000_000, 4096
    keys = torch.randint(0, nkeys, (n,), dtype=torch.int64, device="cuda:0")
    v1 = torch.randint(-1000, 1000, (n,), dtype=torch.int64,
                       device="cuda:0")
    v2 = torch.rand(n, dtype=torch.float64, device="cuda:0")
    uk, outs = kernels.groupby(keys, [v1, v2], ["sum", "max"])
    # torch oracle
    ref_uk, inv = torch.unique(keys, return_inverse=True)
    ref_sum = torch.zeros(ref_uk.shape[0], dtype=torch.int64,
                          device="cuda:0").scatter_reduce_(
        0, inv, v1, reduce="sum", include_self=False)
    ref_max = torch.empty(ref_uk.shape[0], dtype=torch.float64,
                          device="cuda:0").scatter_reduce_(
        0, inv, v2, reduce="amax", include_self=False)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(outs[0][order], ref_sum)
    assert torch.allclose(outs[1][order], ref_max)


def test_groupby_sentinel_key(kernels):
    # int64 min is the table sentinel; must still aggregate correctly.
    sent = -(2**63)
    keys = torch.tensor([sent, 5, sent, 5], dtype=torch.int64,
                        device="cuda:0")
    vals = torch.tensor([1, 2, 3, 4], dtype=torch.int64, device="cuda:0")
    uk, outs = kernels.groupby(keys, [vals], ["sum"])
    got = dict(zip(uk.cpu().tolist(), outs[0].cpu().tolist()))
    assert got == {sent: 4, 5: 6}


def test_groupby_min_float(kernels):
    keys = torch.randint(0, 100, (100_000,), dtype=torch.int64,
                         device="cuda:0")
    vals = torch.randn(100_000, dtype=torch.float32, device="cuda:0")
    uk, outs = kernels.groupby(keys, [vals], ["min"])
    ref_uk, inv = torch.unique(keys, return_inverse=True)
    ref = torch.empty(ref_uk.shape[0], dtype=torch.float32,
                      device="cuda:0").scatter_reduce_(
        0, inv, vals, reduce="amin", include_self=False)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(outs[0][order], ref)


def test_radix_argsort_matches_torch(kernels):
    for dt in (torch.int64, torch.float32):
        if dt.is_floating_point:
            k = torch.randn(1_000_000, dtype=dt, device="cuda:0")
        else:
            k = torch.randint(-2**60, 2**60, (1_000_000,), dtype=dt,
                              device="cuda:0")
        perm = kernels.radix_argsort(k)
        assert torch.equal(k[perm], k.sort().values)


def test_hand_radix_sort_edges(kernels):
    """Hand-written LSD radix sort (csrc/radix.hip): stability,
    negatives, tail tiles, constant digits, int32, vs torch stable
    sort."""
    _C = kernels._C
    g = torch.Generator(device="cuda").manual_seed(11)
    cases = [
        torch.randint(-2**62, 2**62, (n,), dtype=torch.int64,
                      device="cuda", generator=g)
        for n in (1, 63, 8191, 8192, 8193, 1_000_000)
    ]
    cases.append(torch.randint(0, 37, (500_000,), dtype=torch.int64,
                               device="cuda", generator=g))
    cases.append(torch.full((10_000,), -5, dtype=torch.int64,
                            device="cuda"))
    cases.append(torch.randint(-2**31, 2**31, (300_000,),
                               dtype=torch.int32, device="cuda",
                               generator=g))
    for keys in cases:
        vals = torch.arange(keys.numel(), dtype=torch.int64,
                            device="cuda")
        sk, sv = _C.radix_sort_kv(keys, vals)
        ref_k, ref_i = torch.sort(keys, stable=True)
        assert torch.equal(sk, ref_k)
        assert torch.equal(sv, ref_i)  # stability
        assert torch.equal(_C.radix_sort_keys(keys), ref_k)


def test_hash_partition_large_nparts(kernels):
    from bigslice_amd.frame import Frame
    n, nparts = 300_000, 1024
    keys = torch.randint(0, 10**9, (n,), dtype=torch.int64, device="cuda:0")
    f = Frame([keys], prefix=1)
    parts = kernels.partition_frame(f, nparts, None)
    total = sum(len(p) for p in parts if p is not None)
    assert total == n


def test_end_to_end_gpu_matches_cpu():
    import bigslice_amd as bs

    def build(nshard):
        def gen(shard, ctx):
            g = torch.Generator()
            g.manual_seed(shard)
            keys = torch.randint(0, 5000, (500_000,), dtype=torch.int64,
                                 generator=g)
            vals = torch.randint(0, 100, (500_000,), dtype=torch.int64,
                                 generator=g)
            yield (keys, vals)
        return bs.Reduce(bs.ReaderFunc(nshard, gen,
                                       bs.schema_of(int, int)), "sum")

    fv = bs.func(build)
    cpu = bs.start(parallelism=2, device="cpu").run(fv, 4)
    gpu = bs.start(parallelism=2, device="cuda:0").run(fv, 4)
    assert sorted(cpu.scan()) == sorted(gpu.scan())


def test_grouptable_regrow_on_overflow(kernels):
    # Force the overflow path: tiny initial capacity, many distinct keys.
    n = 3_000_000
    keys = torch.arange(n, dtype=torch.int64, device="cuda:0")
    vals = torch.ones(n, dtype=torch.int64, device="cuda:0")
    t = kernels.GroupTable([torch.int64], ["sum"], "cuda:0", cap_hint=1024)
    t.insert(keys, [vals])
    out_keys, out_vals = t.finish()
    assert out_keys.shape[0] == n
    assert int(out_vals[0].sum()) == n


def test_grouptable_streaming_inserts(kernels):
    t = kernels.GroupTable([torch.int64], ["sum"], "cuda:0")
    for i in range(5):
        keys = torch.randint(0, 1000, (200_000,), dtype=torch.int64,
                             device="cuda:0")
        t.insert(keys, [torch.ones_like(keys)])
    out_keys, out_vals = t.finish()
    assert int(out_vals[0].sum()) == 1_000_000
    assert out_keys.shape[0] <= 1000


def test_cogroup_device_matches_cpu():
    import bigslice_amd as bs

    def build(nshard):
        a = bs.Const(nshard,
                     torch.tensor([1, 2, 1, 3], dtype=torch.int64),
                     torch.tensor([10, 20, 30, 40], dtype=torch.int64))
        b = bs.Const(nshard,
                     torch.tensor([2, 3, 3], dtype=torch.int64),
                     torch.tensor([5, 6, 7], dtype=torch.int64))
        return bs.Cogroup(a, b)

    fv = bs.func(build)
    cpu = bs.start(parallelism=2, device="cpu").run(fv, 2)
    gpu = bs.start(parallelism=2, device="cuda:0").run(fv, 2)

    def norm(rows):
        return sorted((k, sorted(va), sorted(vb)) for k, va, vb in rows)
    assert norm(cpu.scan()) == norm(gpu.scan())


def test_multikey_reduce_gpu_matches_cpu():
    import bigslice_amd as bs

    def build(nshard):
        def gen(shard, ctx):
            g = torch.Generator()
            g.manual_seed(shard)
            k1 = torch.randint(0, 50, (200_000,), dtype=torch.int64,
                               generator=g)
            k2 = torch.randint(0, 20, (200_000,), dtype=torch.int64,
                               generator=g)
            v = torch.randint(0, 100, (200_000,), dtype=torch.int64,
                              generator=g)
            yield (k1, k2, v)
        src = bs.ReaderFunc(nshard, gen,
                            bs.schema_of(int, int, int, prefix=2))
        return bs.Reduce(src, "sum")

    fv = bs.func(build)
    cpu = bs.start(parallelism=2, device="cpu").run(fv, 3)
    gpu = bs.start(parallelism=2, device="cuda:0").run(fv, 3)
    assert sorted(cpu.scan()) == sorted(gpu.scan())


def test_chained_pipeline_gpu_matches_cpu():
    # map -> filter -> reduce -> (reuse) -> map over a shuffle: one
    # integration chain compared CPU vs GPU.
    import bigslice_amd as bs

    def build(nshard):
        def gen(shard, ctx):
            g = torch.Generator()
            g.manual_seed(100 + shard)
            keys = torch.randint(0, 3000, (300_000,), dtype=torch.int64,
                                 generator=g)
            vals = torch.randint(1, 50, (300_000,), dtype=torch.int64,
                                 generator=g)
            yield (keys, vals)
        src = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
        mapped = bs.Map(src, lambda k, v: (k, v * 2))
        filt = bs.Filter(mapped, lambda k, v: (k & 1) == 0)
        red = bs.Reduce(filt, "sum")
        return bs.Map(red, lambda k, s: (k, s + 1))

    fv = bs.func(build)
    cpu = bs.start(parallelism=2, device="cpu").run(fv, 4)
    gpu = bs.start(parallelism=2, device="cuda:0").run(fv, 4)
    assert sorted(cpu.scan()) == sorted(gpu.scan())


def test_reduce_keys_only_gpu():
    import bigslice_amd as bs
    keys = torch.randint(0, 500, (1_000_000,), dtype=torch.int64)
    fv = bs.func(lambda: bs.Reduce(bs.Const(3, keys), "sum"))
    cpu = bs.start(parallelism=2, device="cpu").run(fv)
    gpu = bs.start(parallelism=2, device="cuda:0").run(fv)
    assert sorted(cpu.scan()) == sorted(gpu.scan())


def test_packed_groupby_variant(kernels, monkeypatch):
    # experimental packed-slot layout stays correct
    monkeypatch.setenv("BIGSLICE_GB_PACKED", "1")
    keys = torch.randint(0, 5000, (1_000_000,), dtype=torch.int64,
                         device="cuda:0")
    vals = torch.randint(0, 100, (1_000_000,), dtype=torch.int64,
                         device="cuda:0")
    uk, outs = kernels.groupby(keys, [vals], ["sum"])
    ref_uk, inv = torch.unique(keys, return_inverse=True)
    ref = torch.zeros(ref_uk.shape[0], dtype=torch.int64,
                      device="cuda:0").scatter_reduce_(
        0, inv, vals, reduce="sum", include_self=False)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(outs[0][order], ref)


def test_forced_lds_and_global_modes(kernels, monkeypatch):
    keys = torch.randint(0, 200, (2_000_000,), dtype=torch.int64,
                         device="cuda:0")
    vals = torch.ones_like(keys)
    for mode in ("lds", "global"):
        monkeypatch.setenv("BIGSLICE_GB_MODE", mode)
        uk, outs = kernels.groupby(keys, [vals], ["sum"])
        assert int(outs[0].sum()) == 2_000_000
        assert uk.shape[0] <= 200


def test_radix_sort_reduced_bits(kernels):
    # >1M rows with small non-negative keys triggers the reduced
    # end_bit path; negative keys must fall back to full width.
    n = 2_000_000
    for lo, hi in ((0, 1000), (0, 2**40), (-50, 50)):
        k = torch.randint(lo, hi, (n,), dtype=torch.int64, device="cuda:0")
        v = torch.arange(n, dtype=torch.int64, device="cuda:0")
        ks, vs = kernels.radix_sort_kv(k, v)
        ref_k, ref_perm = torch.sort(k, stable=True)
        assert torch.equal(ks, ref_k), (lo, hi)
        # values must travel with their keys (pairing, not order)
        assert torch.equal(k[vs], ks), (lo, hi)
        perm = kernels.radix_argsort(k)
        assert torch.equal(k[perm], ref_k), (lo, hi)
        assert torch.equal(kernels.radix_sort_keys(k), ref_k), (lo, hi)


def test_runs_sorted_matches_fallback(kernels, monkeypatch):
    # K18 one-pass run boundaries vs the diff-mask chain, including
    # all-equal, all-distinct, negative keys and the large-tile
    # geometry (>16M rows); the rocPRIM A/B arm must agree too
    from bigslice_amd.ops.cogroup import _runs
    for keys in (
            torch.randint(0, 97, (1_000_003,), dtype=torch.int64),
            torch.zeros(4097, dtype=torch.int64),
            torch.arange(5000, dtype=torch.int64),
            torch.randint(-5, 5, (300_000,), dtype=torch.int64),
            torch.randint(0, 1 << 18, (17_000_001,), dtype=torch.int64),
            torch.empty(0, dtype=torch.int64)):
        sk = torch.sort(keys).values.to("cuda:0")
        # fallback path on the same data (force via CPU then move)
        m_uniq, m_starts, m_ends = _runs(sk.cpu())
        for env in ("0", "1"):
            monkeypatch.setenv("BIGSLICE_RUNS_ROCPRIM", env)
            uniq, starts, ends = _runs(sk)
            assert torch.equal(uniq.cpu(), m_uniq), env
            assert torch.equal(starts.cpu(), m_starts), env
            assert torch.equal(ends.cpu(), m_ends), env


def test_segment_reduce_runs_fast_path(kernels, monkeypatch):
    # K16 fast path: forced sort-combine must agree with a torch
    # reference for sum/min/max, including the skewed-run case that
    # triggers the rocPRIM fallback (max run length > 4096)
    monkeypatch.setenv("BIGSLICE_GB_COMBINE", "sort")
    for skew in (False, True):
        n, nk = 2_000_000, 50_000
        g = torch.Generator().manual_seed(11 + skew)
        keys = torch.randint(0, nk, (n,), dtype=torch.int64, generator=g)
        if skew:
            hot = torch.rand(n, generator=g) < 0.5
            keys = torch.where(hot, torch.zeros_like(keys), keys)
        v1 = torch.randint(-100, 100, (n,), dtype=torch.int64,
                           generator=g)
        v2 = torch.randint(-100, 100, (n,), dtype=torch.int64,
                           generator=g)
        t = kernels.GroupTable([torch.int64, torch.int64],
                               ["sum", "max"], "cuda:0")
        t.insert(keys.cuda(), [v1.cuda(), v2.cuda()])
        uk, outs = t.finish()
        order = torch.argsort(uk.cpu())
        uk_s = uk.cpu()[order]
        ref_k = torch.unique(keys)
        assert torch.equal(uk_s, ref_k), skew
        import torch as _t
        ref_sum = _t.zeros(nk, dtype=_t.int64).scatter_add_(
            0, keys, v1)[ref_k]
        ref_max = _t.full((nk,), -(1 << 62), dtype=_t.int64)
        ref_max.scatter_reduce_(0, keys, v2, reduce="amax")
        ref_max = ref_max[ref_k]
        assert torch.equal(outs[0].cpu()[order], ref_sum), skew
        assert torch.equal(outs[1].cpu()[order], ref_max), skew


def test_grouptable_sort_combine_paths(kernels):
    # Force both combine paths and check they agree with torch.
    import os
    n, nkeys = 3_000_000, 500_000
    k = torch.randint(0, nkeys, (n,), dtype=torch.int64, device="cuda:0")
    v = torch.randint(-100, 100, (n,), dtype=torch.int64, device="cuda:0")
    ref_uk, inv = torch.unique(k, return_inverse=True)
    ref_sum = torch.zeros_like(ref_uk).index_add_(0, inv, v)
    for mode in ("sort", "hash", "auto"):
        os.environ["BIGSLICE_GB_COMBINE"] = mode
        try:
            t = kernels.GroupTable([torch.int64], ["sum"],
                                   torch.device("cuda:0"))
            for off in range(0, n, 1_000_000):
                t.insert(k[off:off + 1_000_000],
                         [v[off:off + 1_000_000]])
            uk, (us,) = t.finish()
        finally:
            del os.environ["BIGSLICE_GB_COMBINE"]
        order = torch.argsort(uk)
        assert torch.equal(uk[order], ref_uk), mode
        assert torch.equal(us[order], ref_sum), mode


def test_grouptable_mixed_provisional_sort(kernels):
    # First large batch is hash-inserted provisionally; wide key space
    # flips the decision to sort for the rest; finish merges both.
    n, nkeys = 9_000_000, 8_000_000
    k = torch.randint(0, nkeys, (n,), dtype=torch.int64, device="cuda:0")
    v = torch.ones(n, dtype=torch.int64, device="cuda:0")
    t = kernels.GroupTable([torch.int64], ["sum"], torch.device("cuda:0"))
    t.insert(k[:3_000_000], [v[:3_000_000]])     # provisional (pending)
    t.insert(k[3_000_000:6_000_000], [v[3_000_000:6_000_000]])
    t.insert(k[6_000_000:], [v[6_000_000:]])
    uk, (us,) = t.finish()
    ref_uk, counts = torch.unique(k, return_counts=True)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(us[order], counts)


def test_grouptable_hotkeys_streams_lds(kernels):
    # Hot keys: decision lands on the LDS tier; results still exact.
    n = 6_000_000
    k = torch.randint(0, 500, (n,), dtype=torch.int64, device="cuda:0")
    v = torch.randint(-10, 10, (n,), dtype=torch.int64, device="cuda:0")
    t = kernels.GroupTable([torch.int64], ["sum"], torch.device("cuda:0"))
    for off in range(0, n, 3_000_000):
        t.insert(k[off:off + 3_000_000], [v[off:off + 3_000_000]])
    assert t._mode in ("pending", "lds")
    uk, (us,) = t.finish()
    ref_uk, inv = torch.unique(k, return_inverse=True)
    ref = torch.zeros_like(ref_uk).index_add_(0, inv, v)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(us[order], ref)


def test_grouptable_sort_combine_multicol(kernels):
    import os
    n = 3_000_000
    k = torch.randint(0, 2_500_000, (n,), dtype=torch.int64,
                      device="cuda:0")
    v1 = torch.randint(-5, 5, (n,), dtype=torch.int64, device="cuda:0")
    v2 = torch.ones(n, dtype=torch.int64, device="cuda:0")
    os.environ["BIGSLICE_GB_COMBINE"] = "sort"
    try:
        t = kernels.GroupTable([torch.int64, torch.int64],
                               ["sum", "sum"], torch.device("cuda:0"))
        t.insert(k, [v1, v2])
        uk, (s1, s2) = t.finish()
    finally:
        del os.environ["BIGSLICE_GB_COMBINE"]
    ref_uk, inv = torch.unique(k, return_inverse=True)
    r1 = torch.zeros_like(ref_uk).index_add_(0, inv, v1)
    r2 = torch.zeros_like(ref_uk).index_add_(0, inv, v2)
    order = torch.argsort(uk)
    assert torch.equal(uk[order], ref_uk)
    assert torch.equal(s1[order], r1)
    assert torch.equal(s2[order], r2)


def test_grouptable_sort_combine_float_deterministic(kernels):
    # float sums via the sort path: deterministic across runs and
    # accurate against an fp64 reference; min/max exact.
    import os
    n = 2_500_000
    k = torch.randint(0, 2_000_000, (n,), dtype=torch.int64,
                      device="cuda:0")
    v = torch.randn(n, dtype=torch.float32, device="cuda:0")
    os.environ["BIGSLICE_GB_COMBINE"] = "sort"
    try:
        outs = []
        for _ in range(2):
            t = kernels.GroupTable([torch.float32], ["sum"],
                                   torch.device("cuda:0"))
            t.insert(k, [v])
            uk, (s,) = t.finish()
            o = torch.argsort(uk)
            outs.append((uk[o], s[o]))
        assert torch.equal(outs[0][1], outs[1][1])  # bitwise repeatable
        ref_uk, inv = torch.unique(k, return_inverse=True)
        ref = torch.zeros(ref_uk.shape[0], dtype=torch.float64,
                          device="cuda:0").index_add_(0, inv,
                                                      v.to(torch.float64))
        assert torch.equal(outs[0][0], ref_uk)
        assert torch.allclose(outs[0][1].to(torch.float64), ref,
                              atol=1e-3, rtol=1e-5)
        t = kernels.GroupTable([torch.float32, torch.int64],
                               ["min", "max"], torch.device("cuda:0"))
        vi = torch.randint(-10**9, 10**9, (n,), dtype=torch.int64,
                           device="cuda:0")
        t.insert(k, [v, vi])
        uk2, (mn, mx) = t.finish()
        o2 = torch.argsort(uk2)
        ref_mn = torch.full((ref_uk.shape[0],), float("inf"),
                            device="cuda:0").scatter_reduce_(
            0, inv, v, "amin").to(torch.float32)
        ref_mx = torch.full((ref_uk.shape[0],), -(2**62),
                            dtype=torch.int64,
                            device="cuda:0").scatter_reduce_(
            0, inv, vi, "amax")
        assert torch.equal(uk2[o2], ref_uk)
        assert torch.equal(mn[o2], ref_mn)
        assert torch.equal(mx[o2], ref_mx)
    finally:
        del os.environ["BIGSLICE_GB_COMBINE"]


def test_colsum16_valu_and_mfma(kernels):
    from bigslice_amd.kernels import _C
    x = torch.randn(1_000_003, 16, dtype=torch.float32, device="cuda:0")
    ref = x.to(torch.float64).sum(0)
    for mfma in (False, True):
        out = _C.colsum16(x, mfma).to(torch.float64)
        assert torch.allclose(out, ref, atol=1e-1, rtol=1e-5), mfma


def test_hash_bytes_matches_host(kernels):
    import random
    from bigslice_amd import strings
    from bigslice_amd.hashing import murmur3_bytes
    rng = random.Random(5)
    words = ["", "a", "ab", "abc", "abcd", "ünïcödé-wörd", "x" * 33]
    words += ["".join(chr(rng.randrange(33, 1000)) for _ in
                      range(rng.randrange(0, 20))) for _ in range(200)]
    for seed in (0, 7, 0x9ACB0442):
        dev = strings.hash_strings(words, "cuda:0", seed).cpu()
        ref = torch.tensor([murmur3_bytes(w.encode("utf-8"), seed)
                            for w in words], dtype=torch.int64)
        assert torch.equal(dev, ref), seed
    ids_dev = strings.string_ids(words, "cuda:0").cpu()
    ids_host = strings.string_ids(words, "cpu")
    assert torch.equal(ids_dev, ids_host)


def test_bytes_column_device_pipeline(kernels):
    """First-class BYTES columns on device: hash (K17), select,
    partition colocation, and a string-keyed Reduce through the device
    dictionary aggregator — end to end on cuda:0."""
    import bigslice_amd as bs
    from bigslice_amd.frame import BytesColumn, Frame
    from bigslice_amd.hashing import murmur3_bytes
    words = [f"k{i % 97:03d}" for i in range(20_000)]
    c = BytesColumn.from_list(words, "cuda:0")
    h = c.hash32(7).cpu().tolist()
    assert h[:5] == [murmur3_bytes(w.encode(), 7) for w in words[:5]]
    # device ids match the host fallback bit for bit
    ids_dev = c.ids64().cpu()
    ids_host = BytesColumn.from_list(words).ids64()
    assert torch.equal(ids_dev, ids_host)

    def build(m):
        def gen(shard, ctx):
            col = BytesColumn.from_list(words, ctx.device)
            yield (col, torch.ones(len(words), dtype=torch.int64,
                                   device=ctx.device))
        src = bs.ReaderFunc(m, gen, bs.schema_of(bytes, int))
        return bs.Reduce(src, "sum")

    sess = bs.start(parallelism=2, device="cuda:0")
    res = sess.run(bs.func(build), 2)
    got = {k.decode(): v for k, v in res.scan()}
    want = {}
    for w in words * 2:
        want[w] = want.get(w, 0) + 1
    assert got == want


def test_gpu_wordcount_recipe_device(kernels):
    from collections import Counter
    import bigslice_amd as bs
    from bigslice_amd import recipes
    lines = [f"alpha beta g{i % 50} delta e{i % 11}" for i in range(4000)]
    ref = Counter(w for ln in lines for w in ln.split())
    sess = bs.start(parallelism=4, device="cuda:0")
    got = recipes.gpu_wordcount(sess, 4, lines, "cuda:0")
    assert got == dict(ref)


def test_chaos_task_loss_on_device(kernels):
    # fault tolerance with device-resident frames: random transient
    # task losses + stored-output loss; results stay exact.
    import random
    import bigslice_amd as bs
    from bigslice_amd.runtime.local import LocalExecutor, TaskLost
    from bigslice_amd.runtime.session import Session
    rng = random.Random(11)

    def build():
        keys = torch.randint(0, 997, (200_000,), dtype=torch.int64,
                             device="cuda:0")
        vals = torch.ones(200_000, dtype=torch.int64, device="cuda:0")
        return bs.Reduce(bs.Const(6, keys, vals), "sum")

    ex = LocalExecutor(parallelism=4, device="cuda:0")

    def chaos(task):
        if task.consecutive_lost < 2 and rng.random() < 0.35:
            ex.store.discard_task(task.name)
            raise TaskLost(task.name)
    ex.fault_hook = chaos
    sess = Session(ex)
    fv = bs.func(build)
    res = sess.run(fv)
    got = dict(res.scan())
    assert sum(got.values()) == 200_000
    assert len(got) == 997


def test_op_matrix_gpu_matches_cpu(kernels, tmp_path):
    # Rarely-GPU-exercised ops: Fold, Head, Scan, WriterFunc, Cache,
    # Reshard, custom-partitioner Repartition, 3-dep Cogroup — each op
    # on device data must equal its CPU result.
    import bigslice_amd as bs

    def data(dev):
        g = torch.Generator()
        g.manual_seed(4)
        k = torch.randint(0, 97, (40_000,), dtype=torch.int64,
                          generator=g)
        v = torch.randint(-5, 5, (40_000,), dtype=torch.int64,
                          generator=g)
        return (k.to(dev), v.to(dev))

    def build_fold(nshard, dev):
        src = bs.Const(6, *data(dev), prefix=1)
        return bs.Fold(src, lambda acc, v: (acc or 0) + v,
                       out_schema=(int,))

    def build_head(nshard, dev):
        src = bs.Const(6, *data(dev), prefix=1)
        return bs.Head(src, 5)

    def build_reshard(nshard, dev):
        src = bs.Const(6, *data(dev), prefix=1)
        return bs.Reduce(bs.Reshard(src, 3), "sum")

    def build_repart(nshard, dev):
        def odd_even(frame, nparts):
            return (frame.columns[0] % nparts).to(torch.int64)
        src = bs.Const(6, *data(dev), prefix=1)
        return bs.Reduce(bs.Repartition(src, odd_even), "sum")

    def build_cg3(nshard, dev):
        a = bs.Const(4, *data(dev), prefix=1)
        b = bs.Const(4, *data(dev), prefix=1)
        c = bs.Const(4, *data(dev), prefix=1)
        cg = bs.Cogroup(a, b, c)
        return bs.Map(cg, lambda k, va, vb, vc: (
            k, sum(va), len(vb), len(vc)),
            out_schema=(int, int, int, int), rowwise=True)

    builders = {"fold": build_fold, "head": build_head,
                "reshard": build_reshard, "repart": build_repart,
                "cg3": build_cg3}
    for name, b in builders.items():
        fv = bs.func(b)
        results = {}
        for dev in ("cpu", "cuda:0"):
            sess = bs.start(parallelism=6, device=dev)
            rows = sorted(sess.run(fv, 6, dev).scan())
            results[dev] = rows
        if name == "head":  # per-shard head: row sets differ by split
            assert len(results["cpu"]) == len(results["cuda:0"])
        else:
            assert results["cpu"] == results["cuda:0"], name

    # Cache on device frames: second session reads the cached files
    import bigslice_amd.ops.cache as _  # noqa
    prefix = str(tmp_path / "c")

    def build_cache(nshard, dev):
        src = bs.Const(4, *data(dev), prefix=1)
        return bs.Cache(bs.Reduce(src, "sum"), prefix)
    fv = bs.func(build_cache)
    sess = bs.start(parallelism=4, device="cuda:0")
    first = sorted(sess.run(fv, 4, "cuda:0").scan())
    sess2 = bs.start(parallelism=4, device="cuda:0")
    second = sorted(sess2.run(fv, 4, "cuda:0").scan())
    assert first == second and len(first) == 97


def test_store_high_water_tiering(kernels, monkeypatch):
    # Force tiering: stored frames land in pinned host DRAM and read
    # back exactly.
    monkeypatch.setenv("BIGSLICE_STORE_HIGH_WATER", "0.0")
    from bigslice_amd.frame import Frame
    from bigslice_amd.runtime.store import MemoryStore
    st = MemoryStore()
    k = torch.randint(0, 100, (50_000,), dtype=torch.int64,
                      device="cuda:0")
    st.put("t", 0, [Frame([k], prefix=1)], 50_000)
    stored = st._data[("t", 0)][0][0]
    assert stored.device == "cpu" and stored.columns[0].is_pinned()
    back = list(st.open("t", 0, device="cuda:0"))
    assert torch.equal(back[0].columns[0], k)
    # end-to-end under forced tiering
    import bigslice_amd as bs

    def build():
        keys = torch.randint(0, 997, (100_000,), dtype=torch.int64,
                             device="cuda:0")
        vals = torch.ones(100_000, dtype=torch.int64, device="cuda:0")
        return bs.Reduce(bs.Const(4, keys, vals), "sum")
    sess = bs.start(parallelism=4, device="cuda:0")
    got = dict(sess.run(bs.func(build)).scan())
    assert sum(got.values()) == 100_000


def test_external_sort_spill_path_gpu(kernels):
    # Forced tiny runs exercise the full spill machinery on device:
    # chunked run spill, D2H backpressure accounting, merge readback.
    from bigslice_amd.frame import Frame
    from bigslice_amd.sliceio import FrameReader
    from bigslice_amd.sortio import SortReader
    n = 3_000_000
    k = torch.randint(0, 1 << 40, (n,), dtype=torch.int64,
                      device="cuda:0")
    v = torch.arange(n, dtype=torch.int64, device="cuda:0")
    src = FrameReader(Frame([k, v], prefix=1), 400_000)
    sr = SortReader(src, run_bytes=4 << 20, device="cuda:0",
                    chunk=400_000)
    outs = [f for f in sr]
    ks = torch.cat([f.columns[0] for f in outs])
    vs = torch.cat([f.columns[1] for f in outs])
    ref_k, perm = torch.sort(k, stable=True)
    assert torch.equal(ks, ref_k)
    assert torch.equal(k[vs], ks)  # values travel with keys
