"""Evaluator/scheduler throughput microbenchmark (reference
exec/eval_test.go BenchmarkEval/BenchmarkEnqueue: multi-phase graphs of
10-5000 shards x 5 stages).

  python benchmarks/scheduler_bench.py [--shards 1000] [--stages 5]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bigslice_amd.runtime.eval import Executor, evaluate
from bigslice_amd.runtime.task import Task, TaskDep, TaskState


class NopExecutor(Executor):
    parallelism = 64

    def run(self, task):
        task.set_state(TaskState.RUNNING)
        task.set_state(TaskState.OK)


def build_graph(shards, stages):
    prev = None
    for _ in range(stages):
        group = []
        tasks = []
        for s in range(shards):
            deps = [TaskDep(prev, s)] if prev else []
            t = Task(name=f"t{id(group)}:{s}", invocation_index=1,
                     do=None, deps=deps, group=group, shard=s,
                     num_shards=shards)
            group.append(t)
            tasks.append(t)
        prev = tasks
    return prev


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--shards", type=int, default=None)
    ap.add_argument("--stages", type=int, default=5)
    args = ap.parse_args()
    shard_counts = [args.shards] if args.shards else [10, 100, 1000, 5000]
    for shards in shard_counts:
        roots = build_graph(shards, args.stages)
        t0 = time.perf_counter()
        evaluate(NopExecutor(), roots)
        dt = time.perf_counter() - t0
        n = shards * args.stages
        print(json.dumps({
            "metric": "tasks/sec scheduler throughput",
            "shards": shards, "stages": args.stages,
            "tasks": n, "seconds": dt, "tasks_per_sec": n / dt}))


if __name__ == "__main__":
    main()
