"""A/B: hand-written LSD radix sort (csrc/radix.hip) vs rocPRIM.

Correctness vs torch.sort(stable) on adversarial shapes, then timed
pairs/keys sorts across sizes and key distributions.  Run on a GPU box:

    python benchmarks/sort_ab.py [--quick]
"""

import argparse
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch


def _t(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def correctness():
    from bigslice_amd import kernels
    _C = kernels._C
    g = torch.Generator(device="cuda").manual_seed(7)
    cases = []
    for n in (1, 63, 64, 8191, 8192, 8193, 100_000, 1 << 22):
        cases.append(torch.randint(-(1 << 62), 1 << 62, (n,),
                                   dtype=torch.int64, device="cuda",
                                   generator=g))
    cases.append(torch.randint(0, 1000, (1 << 20,), dtype=torch.int64,
                               device="cuda", generator=g))
    cases.append(torch.zeros(1 << 20, dtype=torch.int64, device="cuda"))
    cases.append(torch.randint(-(1 << 31), 1 << 31, (1 << 20,),
                               dtype=torch.int32, device="cuda",
                               generator=g))
    for keys in cases:
        vals = torch.arange(keys.numel(), dtype=torch.int64,
                            device="cuda")
        sk, sv = _C.radix_sort_kv(keys, vals)
        ref_k, ref_i = torch.sort(keys, stable=True)
        assert torch.equal(sk, ref_k), (keys.dtype, keys.numel())
        assert torch.equal(sv, ref_i), \
            ("stability", keys.dtype, keys.numel())
        ko = _C.radix_sort_keys(keys)
        assert torch.equal(ko, ref_k)
    print("correctness OK:", len(cases), "cases (incl. stability)")


def bench(quick=False):
    from bigslice_amd import kernels
    _C = kernels._C
    g = torch.Generator(device="cuda").manual_seed(3)
    sizes = [125_000_000] if quick else [125_000_000, 500_000_000]
    for n in sizes:
        for name, hi in (("20bit", 1 << 20), ("32bit", 1 << 32),
                         ("full", 1 << 62)):
            keys = torch.randint(0, hi, (n,), dtype=torch.int64,
                                 device="cuda", generator=g)
            vals = torch.ones(n, dtype=torch.int64, device="cuda")
            t_pairs = _t(lambda: _C.radix_sort_kv(keys, vals))
            t_keys = _t(lambda: _C.radix_sort_keys(keys))
            gbs = n * 16 / t_pairs / 1e9
            print(f"n={n:>11,} {name:>6}: pairs {t_pairs*1e3:7.2f} ms "
                  f"({n/t_pairs/1e9:6.2f} Grows/s, {gbs:7.0f} GB/s "
                  f"pair-bytes) keys {t_keys*1e3:7.2f} ms")
            del keys, vals


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--impl", choices=["hand", "rocprim", "both"],
                    default="both")
    args = ap.parse_args()
    if args.impl == "both":
        for impl in ("hand", "rocprim"):
            env = dict(os.environ)
            env["BIGSLICE_SORT_ROCPRIM"] = \
                "1" if impl == "rocprim" else "0"
            env["BIGSLICE_SORT_HAND"] = \
                "0" if impl == "rocprim" else "1"
            print(f"==== {impl} ====", flush=True)
            r = subprocess.run(
                [sys.executable, __file__, "--impl", impl]
                + (["--quick"] if args.quick else []), env=env)
            if r.returncode:
                sys.exit(r.returncode)
        return
    if args.impl == "hand":
        correctness()
    bench(args.quick)


if __name__ == "__main__":
    main()
