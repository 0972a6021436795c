"""Flagship benchmark: whole-node Reshuffle+Reduce group-by-sum.

BASELINE.json north-star config 3: group-by-sum over 1B int64 rows in 8
shards (1B at 8 GPUs; per-GPU rows fixed -> weak scaling).  Each step runs
the full engine pipeline on pre-generated synthetic device data: the
adaptive producer-side combine (LDS-tier hash insert / streaming global
insert / radix-sort+reduce-by-key, chosen from a sampled key
cardinality: K9/K10/K16), the fused hash+scatter partitioner (K4),
RCCL all-to-allv over xGMI at N>1, and the consumer-side final
aggregate (pre-combined passthrough at N=1).

Usage:
  python bench.py --gpus 1 --steps 5 --warmup 2
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 5 --warmup 2
"""

import argparse
import json
import os
import time

import torch

import bigslice_amd as bs

DEFAULT_ROWS_PER_GPU = 125_000_000
DEFAULT_NKEYS = 1_000_000
DEFAULT_SHARDS = 8

_DATA = {}


def gen_shard_data(shard, rows, nkeys, device):
    g = torch.Generator(device=device)
    g.manual_seed(0xB165 + shard)
    keys = torch.randint(0, nkeys, (rows,), dtype=torch.int64,
                         device=device, generator=g)
    vals = torch.ones(rows, dtype=torch.int64, device=device)
    return keys, vals


def build_groupby(nshard):
    def gen(shard, ctx):
        keys, vals = _DATA[shard]
        chunk = ctx.chunk
        for off in range(0, keys.shape[0], chunk):
            yield (keys[off:off + chunk], vals[off:off + chunk])
    src = bs.ReaderFunc(nshard, gen, bs.schema_of(int, int))
    return bs.Reduce(src, "sum")


groupby_func = bs.func(build_groupby)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows-per-gpu", type=int, default=DEFAULT_ROWS_PER_GPU)
    ap.add_argument("--nkeys", type=int, default=DEFAULT_NKEYS)
    ap.add_argument("--shards", type=int, default=DEFAULT_SHARDS)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--trace", type=str, default=None,
                    help="write a Chrome trace of the run to this path")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    distributed = world > 1
    if distributed:
        sess = bs.start(distributed=True, device=args.device,
                        trace_path=args.trace)
        device = sess.executor.device
        comm = sess.executor.comm
    else:
        device = args.device or (
            "cuda:0" if torch.cuda.is_available() else "cpu")
        sess = bs.start(parallelism=args.shards, device=device,
                        trace_path=args.trace)
        comm = None

    on_gpu = device.startswith("cuda")
    if on_gpu:
        from bigslice_amd import kernels
        assert kernels.have_extension(), \
            "HIP extension missing: build with setup.py build_ext --inplace"

    nshard = args.shards
    total_rows = args.rows_per_gpu * world
    rows_per_shard = total_rows // nshard

    # Pre-generate synthetic input for the shards this rank owns.
    for shard in range(nshard):
        if shard % world == rank:
            _DATA[shard] = gen_shard_data(shard, rows_per_shard,
                                          args.nkeys, device)

    def step():
        res = sess.run(groupby_func, nshard)
        res.discard()

    def sync():
        if on_gpu:
            torch.cuda.synchronize()
        if comm is not None:
            comm.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0

    # Max over ranks (whole-job time).
    if comm is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if comm.backend == "nccl":
            t = t.to(device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.cpu().item())

    ms_per_step = elapsed * 1000.0 / args.steps
    rows_per_sec = (rows_per_shard * nshard) / (ms_per_step / 1000.0)

    if args.trace:
        sess.shutdown()
    if rank == 0:
        out = {
            "metric": "rows/sec (whole node) Reshuffle+Reduce group-by",
            "value": rows_per_sec,
            "unit": "rows/sec",
            "n_gpus": world if distributed else 1,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "model": "Reshuffle+Reduce group-by-sum (BASELINE config 3)",
                "rows_total": rows_per_shard * nshard,
                "rows_per_gpu": args.rows_per_gpu,
                "distinct_keys": args.nkeys,
                "global_batch": rows_per_shard * nshard,
                "seq_len": None,
                "shards": nshard,
                "parallelism": f"dp{world}",
                "device": device,
            },
        }
        print(json.dumps(out), flush=True)
    if distributed:
        # Clean teardown: ranks exiting at different times with a live
        # process group can SIGABRT in the backend's helper threads.
        torch.distributed.barrier()
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    try:
        main()
    except BaseException:
        import traceback
        traceback.print_exc()
        raise
