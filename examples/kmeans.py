"""Iterative k-means on the slice engine: per-step assignment +
per-cluster Reduce run device-native; the driver updates centers
between invocations (the reference's iterative-computing pattern,
exec/session.go:40-43, with Func args carrying the model state).

  python examples/kmeans.py [--points 1000000] [--k 16] [--iters 10]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import bigslice_amd as bs

_POINTS = {}


def build_step(nshard, centers_x, centers_y):
    cx = torch.tensor(centers_x, dtype=torch.float32)
    cy = torch.tensor(centers_y, dtype=torch.float32)

    def gen(shard, ctx):
        yield _POINTS[shard]

    points = bs.ReaderFunc(nshard, gen,
                           bs.schema_of(torch.float32, torch.float32,
                                        prefix=1))

    def assign(x, y):
        # vectorized on device: nearest center per point
        cxd = cx.to(x.device)
        cyd = cy.to(x.device)
        d = (x[:, None] - cxd[None, :]) ** 2 + \
            (y[:, None] - cyd[None, :]) ** 2
        cid = d.argmin(dim=1).to(torch.int64)
        ones = torch.ones_like(x)
        return (cid, x, y, ones)

    assigned = bs.Map(points, assign,
                      out_schema=(torch.int64, torch.float32,
                                  torch.float32, torch.float32),
                      prefix=1)
    return bs.Reduce(assigned, "sum")  # per-cluster sum_x, sum_y, count


step = bs.func(build_step)


def run_kmeans(sess, nshard, k, iters, device, seed=3):
    g = torch.Generator().manual_seed(seed)
    cx = torch.rand(k, generator=g).tolist()
    cy = torch.rand(k, generator=g).tolist()
    for it in range(iters):
        res = sess.run(step, nshard, cx, cy)
        sums = {c: (sx, sy, n) for c, sx, sy, n in res.scan()}
        res.discard()
        shift = 0.0
        for c in range(k):
            if c in sums:
                sx, sy, n = sums[c]
                nx, ny = sx / n, sy / n
                shift = max(shift, abs(nx - cx[c]) + abs(ny - cy[c]))
                cx[c], cy[c] = nx, ny
        print(f"iter {it}: max center shift {shift:.5f}")
        if shift < 1e-5:
            break
    return cx, cy


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--points", type=int, default=1_000_000)
    ap.add_argument("--k", type=int, default=16)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--shards", type=int, default=8)
    ap.add_argument("--device", type=str, default=None)
    args = ap.parse_args()
    device = args.device or ("cuda:0" if torch.cuda.is_available()
                             else "cpu")
    per = args.points // args.shards
    g = torch.Generator().manual_seed(1)
    for s in range(args.shards):
        x = torch.rand(per, generator=g).to(device)
        y = torch.rand(per, generator=g).to(device)
        _POINTS[s] = (x, y)
    sess = bs.start(parallelism=args.shards, device=device)
    cx, cy = run_kmeans(sess, args.shards, args.k, args.iters, device)
    print("centers:", [(round(a, 3), round(b, 3))
                       for a, b in zip(cx, cy)][:8], "...")


if __name__ == "__main__":
    main()
