"""A/B: one-level global insert vs 4-row software-pipelined variant."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from bigslice_amd import kernels
from bigslice_amd.kernels import _C, GB_SENTINEL, MAX_PROBES

def run(kind, k, v, cap):
    tkeys = torch.full((cap + 1,), GB_SENTINEL, dtype=torch.int64,
                       device="cuda")
    tab = torch.zeros(cap + 1, dtype=torch.int64, device="cuda")
    flags = torch.zeros(2, dtype=torch.int32, device="cuda")
    if kind == "mlp":
        _C.groupby_insert_mlp(k, v, tkeys, tab, flags, MAX_PROBES)
    else:
        _C.groupby_insert(k, [v], [0], tkeys, [tab], flags, MAX_PROBES)
    return tkeys, tab, flags

N = 125_000_000
for nkeys in (100_000, 1_000_000, 10_000_000):
    g = torch.Generator(device="cuda"); g.manual_seed(3)
    k = torch.randint(0, nkeys, (N,), dtype=torch.int64, device="cuda",
                      generator=g)
    v = torch.ones(N, dtype=torch.int64, device="cuda")
    cap = 1
    while cap < 2 * nkeys: cap <<= 1
    # correctness: same sums per slot arrangement
    def pairs(tk, tb):
        m = tk[:-1] != GB_SENTINEL
        ks_, vs_ = tk[:-1][m], tb[:-1][m]
        o = torch.argsort(ks_)
        return ks_[o], vs_[o]
    k1, v1 = pairs(*run("global", k, v, cap)[:2])
    k2, v2 = pairs(*run("mlp", k, v, cap)[:2])
    assert torch.equal(k1, k2) and torch.equal(v1, v2)
    for kind in ("global", "mlp"):
        for _ in range(2): run(kind, k, v, cap)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(5): run(kind, k, v, cap)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 5 * 1000
        print(f"nkeys={nkeys:>9,} {kind:>6}: {ms:7.2f} ms "
              f"({N/ms/1e6:.1f} G rows/s)")
