"""A/B: hash-insert group-by vs sort+segment-reduce, 125M rows."""
import torch, time, sys
sys.path.insert(0, ".")
from bigslice_amd import kernels

def timeit(fn, warm=2, iters=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

N = 125_000_000
for nkeys in (1_000, 1_000_000, 10_000_000):
    g = torch.Generator(device="cuda"); g.manual_seed(7)
    k = torch.randint(0, nkeys, (N,), dtype=torch.int64, device="cuda", generator=g)
    v = torch.ones(N, dtype=torch.int64, device="cuda")

    def hash_path():
        from bigslice_amd.kernels import GroupTable
        t = GroupTable([torch.int64], ["sum"], torch.device("cuda:0"))
        t.insert(k, [v])
        ks2, vs2 = t.finish()
        return ks2, vs2[0]

    def sort_path_k16():
        from bigslice_amd.kernels import _C
        ks, vs = kernels.radix_sort_kv(k, v)
        uq, sums, cnt = _C.segment_sum_sorted(ks, vs)
        m = int(cnt.item())
        return uq[:m], sums[:m]

    def sort_path():
        ks, vs = kernels.radix_sort_kv(k, v)
        mask = torch.empty(N, dtype=torch.bool, device="cuda")
        mask[0] = True
        torch.ne(ks[1:], ks[:-1], out=mask[1:])
        starts = mask.nonzero(as_tuple=True)[0]
        uk = ks[starts]
        cs = torch.cumsum(vs, 0)
        ends = torch.cat([starts[1:], torch.tensor([N], device="cuda")]) - 1
        tot = cs[ends]
        out = torch.empty_like(tot)
        out[0] = tot[0]
        torch.sub(tot[1:], tot[:-1], out=out[1:])
        return uk, out

    hk, hv = hash_path()
    sk, sv = sort_path()
    ho = torch.argsort(hk)
    assert torch.equal(hk[ho], sk), nkeys
    assert torch.equal(hv[ho], sv), nkeys
    k16k, k16v = sort_path_k16()
    o = torch.argsort(k16k)
    assert torch.equal(k16k[o], sk) and torch.equal(k16v[o], sv), nkeys
    th = timeit(hash_path)
    ts = timeit(sort_path)
    tk = timeit(sort_path_k16)
    print(f"nkeys={nkeys:>9,}: hash={th:7.2f} sort_torch={ts:7.2f} sort_k16={tk:7.2f} ms")
