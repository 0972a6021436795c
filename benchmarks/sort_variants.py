"""In-process A/B of hand radix-sort geometry variants vs rocPRIM.

All arms run in ONE process on ONE box back-to-back, so DVFS/box
variance cancels (guide §5.4: never compare across runs)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402

from bigslice_amd import kernels  # noqa: E402

_C = kernels._C


def timeit(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 125_000_000
    g = torch.Generator(device="cuda").manual_seed(3)
    dists = {"20bit": 1 << 20, "full": 1 << 62}
    iota = torch.arange(n, dtype=torch.int64, device="cuda")
    data = {name: (torch.randint(0, hi, (n,), dtype=torch.int64,
                                 device="cuda", generator=g), iota)
            for name, hi in dists.items()}
    # interior constant bytes: low 16 bits random, bytes 2-4 zero,
    # bytes 5-6 random — rocPRIM's contiguous bit range spans 7 bytes,
    # the hand path runs only the 4 non-constant ones
    gk = (torch.randint(0, 1 << 16, (n,), dtype=torch.int64,
                        device="cuda", generator=g)
          | (torch.randint(0, 1 << 16, (n,), dtype=torch.int64,
                           device="cuda", generator=g) << 40))
    data["gap"] = (gk, iota)
    # values are row ids, so a STABLE kv sort must reproduce the
    # stable argsort permutation exactly — validates keys AND values
    ref = {}
    for name, (k, v) in data.items():
        srt = torch.sort(k, stable=True)
        ref[name] = (srt.values, srt.indices.to(torch.int64))

    arms = [("auto", {"BIGSLICE_SORT_ROCPRIM": "0",
                      "BIGSLICE_SORT_HAND": "0",
                      "BIGSLICE_RADIX_VARIANT": "2"}),
            ("rocprim", {"BIGSLICE_SORT_ROCPRIM": "1",
                         "BIGSLICE_SORT_HAND": "0"})]
    vars_arg = sys.argv[2] if len(sys.argv) > 2 else "2"
    for var in (int(x) for x in vars_arg.split(",")):
        arms.append((f"hand v{var}",
                     {"BIGSLICE_SORT_ROCPRIM": "0",
                      "BIGSLICE_SORT_HAND": "1",
                      "BIGSLICE_RADIX_VARIANT": str(var)}))
    for rounds in range(2):  # two rounds to expose drift
        for name, env in arms:
            os.environ.update(env)
            line = f"[r{rounds}] {name:10s}"
            for dist, (k, v) in data.items():
                ms = timeit(lambda: _C.radix_sort_kv(k, v))
                sk, sv = _C.radix_sort_kv(k, v)
                assert torch.equal(sk, ref[dist][0]), (name, dist)
                assert torch.equal(sv, ref[dist][1]), (name, dist, "v")
                line += (f"  {dist} {ms:7.2f}ms"
                         f" ({n/ms/1e6:6.2f} G/s)")
            print(line, flush=True)


if __name__ == "__main__":
    main()
