"""Windowed-exchange overlap A/B (VERDICT task 1 "done" evidence).

World-N reshuffle where producer compute and shuffle volume are both
substantial; arm A forces ONE window (a full phase barrier: all
compute, then all exchange — round 1's behavior), arm B uses small
windows so transfers fly while later chunks compute.  On CPU/gloo the
identical code path RCCL drives on GPUs; only the transport differs.
Writes a Chrome trace of a windowed run (rank 0) next to the results.

    python benchmarks/overlap_ab.py [world] [rows_per_rank]
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402

ROWS = int(sys.argv[2]) if len(sys.argv) > 2 else 6_000_000
WORLD = int(sys.argv[1]) if len(sys.argv) > 1 else 4
CHUNKS = 12
# per-chunk producer compute (sleep releases the GIL like a HIP kernel
# launch stream would; gloo's transfer threads keep running)
COMPUTE_S = float(os.environ.get("OVERLAP_COMPUTE_S", "0.05"))


def worker(rank, world, port, arm, trace_path, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": str(rank),
        "BIGSLICE_EXCHANGE_WINDOW_BYTES":
            str((1 << 40) if arm == "single" else (ROWS * 16) // 8),
    })
    import torch.distributed as dist
    torch.set_num_threads(max(1, (os.cpu_count() or 8) // (2 * world)))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import bigslice_amd as bs

    # pre-generate outside the timed region so the producer's in-loop
    # cost is exactly the sleep (a stand-in for kernel-stream compute)
    g = torch.Generator().manual_seed(rank)
    KEYS = torch.randint(0, 1 << 30, (ROWS,), dtype=torch.int64,
                         generator=g)
    ONES = torch.ones(ROWS, dtype=torch.int64)

    def build(m):
        def gen(shard, ctx):
            per = ROWS // CHUNKS
            for i in range(CHUNKS):
                time.sleep(COMPUTE_S)  # the "map" compute
                yield (KEYS[i * per:(i + 1) * per],
                       ONES[i * per:(i + 1) * per])
        src = bs.ReaderFunc(m, gen, bs.schema_of(int, int))
        return bs.Reshuffle(src)

    fv = bs.func(build)
    sess = bs.start(distributed=True, device="cpu",
                    trace_path=trace_path if rank == 0 else None)
    dist.barrier()
    t0 = time.perf_counter()
    res = sess.run(fv, world)
    dist.barrier()
    dt = time.perf_counter() - t0
    n = sum(1 for _ in res.scan())
    if trace_path and rank == 0:
        sess.shutdown()
    q.put((rank, dt, n))
    dist.barrier()
    dist.destroy_process_group()


def run(arm, trace_path=None):
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=worker,
                         args=(r, WORLD, port, arm, trace_path, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    outs = [q.get() for _ in range(WORLD)]
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    total_rows = next(n for (r, _, n) in outs if r == 0)
    assert total_rows == ROWS * WORLD, total_rows
    return max(dt for (_, dt, _) in outs)


def main():
    compute_s = COMPUTE_S * CHUNKS
    print(f"world={WORLD} rows/rank={ROWS:,} chunks={CHUNKS} "
          f"producer-compute={compute_s:.2f}s/rank")
    t_single = min(run("single") for _ in range(2))
    t_win = min(run("windowed",
                    trace_path="gpurun_out/overlap_trace.json"
                    if os.path.isdir("gpurun_out") else
                    "/tmp/overlap_trace.json") for _ in range(2))
    comm_est = max(t_single - compute_s, 0.0)
    print(json.dumps({
        "single_window_s": round(t_single, 3),
        "windowed_s": round(t_win, 3),
        "producer_compute_s": round(compute_s, 3),
        "est_comm_s": round(comm_est, 3),
        "ideal_overlap_s": round(max(compute_s, comm_est), 3),
        "speedup": round(t_single / t_win, 3),
        "overlap_efficiency": round(
            (t_single - t_win) / min(compute_s, comm_est), 3)
        if min(compute_s, comm_est) > 0 else None,
    }))


if __name__ == "__main__":
    main()
