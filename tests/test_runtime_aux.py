"""Auxiliary runtime subsystems: metrics, tracing, chaos/fault injection,
file store, partial cache, slicetest harness."""

import json
import os
import random

import pytest
import torch

import bigslice_amd as bs
from bigslice_amd import slicetest
from bigslice_amd.runtime.local import LocalExecutor, TaskLost
from bigslice_amd.runtime.session import Session
from bigslice_amd.runtime.store import FileStore
from bigslice_amd.utils import metrics


def test_metrics_scope_merged():
    filtered = metrics.counter("filtered-rows")

    def build():
        def count_filter(x):
            keep = x % 2 == 0
            filtered.incr(int((~keep).sum()))
            return keep
        return bs.Filter(bs.Const(4, torch.arange(100, dtype=torch.int64)),
                         count_filter)

    res = slicetest.run(build)
    assert len(list(res.scan())) == 50
    snap = res.scope().snapshot()
    assert snap.get("filtered-rows") == 50


def test_tracer_writes_chrome_trace(tmp_path):
    path = str(tmp_path / "trace.json")
    fv = bs.func(lambda: bs.Const(2, torch.arange(4, dtype=torch.int64)))
    sess = bs.start(parallelism=2, device="cpu", trace_path=path)
    sess.run(fv)
    sess.shutdown()
    with open(path) as fp:
        data = json.load(fp)
    names = [e["name"] for e in data["traceEvents"]]
    assert any("const" in n for n in names)


def test_chaos_random_task_loss():
    # chaosmonkey_test.go analog: random transient losses; job completes
    # with the right answer.
    rng = random.Random(7)

    def build():
        keys = torch.arange(1000, dtype=torch.int64) % 13
        vals = torch.ones(1000, dtype=torch.int64)
        return bs.Reduce(bs.Const(6, keys, vals), "sum")

    ex = LocalExecutor(parallelism=4, device="cpu")

    def chaos(task):
        if task.consecutive_lost < 2 and rng.random() < 0.4:
            # also lose the task's stored output (machine-loss analog)
            ex.store.discard_task(task.name)
            raise TaskLost(task.name)
    ex.fault_hook = chaos

    sess = Session(ex)
    fv = bs.func(build)
    res = sess.run(fv)
    got = dict(res.scan())
    want = {}
    for k in (torch.arange(1000) % 13).tolist():
        want[k] = want.get(k, 0) + 1
    assert got == want


def test_chaos_interior_output_loss_recomputes():
    # A consumer fails because its dep's output vanished (machine loss
    # after OK): the evaluator must re-run the producer.
    calls = {"n": 0}

    def build():
        keys = torch.arange(100, dtype=torch.int64) % 5
        vals = torch.ones(100, dtype=torch.int64)
        return bs.Reduce(bs.Const(2, keys, vals), "sum")

    ex = LocalExecutor(parallelism=2, device="cpu")
    lost_once = {"done": False}

    def chaos(task):
        # when the first consumer (reduce) task starts, drop one
        # producer's output once
        if task.deps and not lost_once["done"]:
            lost_once["done"] = True
            prod = task.deps[0].head_tasks[0]
            ex.store.discard_task(prod.name)
            from bigslice_amd.runtime.task import TaskState
            prod.set_state(TaskState.LOST)
            raise TaskLost("dep output lost")
    ex.fault_hook = chaos
    res = Session(ex).run(bs.func(build))
    got = dict(res.scan())
    assert got == {k: 20 for k in range(5)}


def test_file_store_roundtrip(tmp_path):
    from bigslice_amd.frame import Frame
    st = FileStore(str(tmp_path / "store"))
    f = Frame([torch.arange(10, dtype=torch.int64)])
    st.put("taskA", 0, [f], 10)
    assert st.has("taskA", 0)
    size, rows = st.stat("taskA", 0)
    assert rows == 10
    frames = list(st.open("taskA", 0))
    assert len(frames) == 1 and frames[0].columns[0].tolist() == \
        list(range(10))
    st.discard_task("taskA")
    assert not st.has("taskA", 0)


def test_file_store_executor_end_to_end(tmp_path):
    # local executor with a file-backed store: every task output is a
    # durable checkpoint (reference fileStore semantics).
    ex = LocalExecutor(parallelism=2, device="cpu",
                       store=FileStore(str(tmp_path / "store")))
    fv = bs.func(lambda: bs.Reduce(
        bs.Const(2, torch.tensor([1, 2, 1], dtype=torch.int64),
                 torch.tensor([1, 1, 1], dtype=torch.int64)), "sum"))
    res = Session(ex).run(fv)
    assert sorted(res.scan()) == [(1, 2), (2, 1)]


def test_partial_cache_recomputes_missing(tmp_path):
    prefix = str(tmp_path / "pc")
    computed = []

    def gen(shard, ctx):
        computed.append(shard)
        yield (torch.arange(2, dtype=torch.int64) + shard * 10,)

    def build():
        return bs.Cache(bs.ReaderFunc(3, gen, bs.schema_of(int)), prefix,
                        partial=True)

    fv = bs.func(build)
    r1 = bs.start(parallelism=2, device="cpu").run(fv)
    assert len(computed) == 3
    # drop one shard's cache file: only that shard recomputes
    os.remove(prefix + "-0001-of-0003")
    computed.clear()
    r2 = bs.start(parallelism=2, device="cpu").run(fv)
    assert sorted(r2.scan()) == [0, 1, 10, 11, 20, 21]
    assert computed == [1]


def test_slicetest_harness():
    rows = slicetest.scan_all(
        lambda: bs.Const(2, torch.arange(5, dtype=torch.int64)))
    assert sorted(rows) == [0, 1, 2, 3, 4]
    err = slicetest.run_err(
        lambda: bs.Map(bs.Const(1, torch.arange(2, dtype=torch.int64)),
                       lambda x: 1 / 0, out_schema=(int,)))
    assert err is not None


def test_error_propagates_from_udf():
    def boom(x):
        raise ValueError("user boom")
    err = slicetest.run_err(
        lambda: bs.Map(bs.Const(2, torch.arange(4, dtype=torch.int64)),
                       boom, out_schema=(int,)))
    assert isinstance(err, ValueError)


def test_scan_after_output_loss_reevaluates():
    # reference session_test.go:188-246 analog: output vanishes between
    # run and scan; the scan re-evaluates the lost tasks.
    fv = bs.func(lambda: bs.Reduce(
        bs.Const(3, torch.arange(300, dtype=torch.int64) % 7,
                 torch.ones(300, dtype=torch.int64)), "sum"))
    sess = bs.start(parallelism=2, device="cpu")
    res = sess.run(fv)
    # lose every task's stored output
    for t in res.tasks:
        sess.executor.store.discard_task(t.name)
        for dep in t.deps:
            for h in dep.head_tasks:
                sess.executor.store.discard_task(h.name)
    got = dict(res.scan())
    want = {}
    for k in (torch.arange(300) % 7).tolist():
        want[k] = want.get(k, 0) + 1
    assert got == want


def test_repartition_custom_partitioner_colocates():
    # reference reshuffle_test.go lengthHashKey analog: partition by a
    # custom function of the key (here key % 3) and assert co-location.
    def part_fn(frame, nshard):
        return (frame.columns[0] % 3) % nshard

    keys = torch.arange(90, dtype=torch.int64)
    fv = bs.func(lambda: bs.Repartition(bs.Const(3, keys), part_fn))
    sess = bs.start(parallelism=2, device="cpu")
    res = sess.run(fv)
    shard_mods = []
    for t in res.tasks:
        mods = set()
        for f in sess.executor.reader(t, 0):
            mods.update((f.columns[0] % 3).tolist())
        shard_mods.append(mods)
    for i in range(len(shard_mods)):
        for j in range(i + 1, len(shard_mods)):
            assert not (shard_mods[i] & shard_mods[j])
    assert sorted(res.scan()) == list(range(90))


def test_status_rollup():
    from bigslice_amd.utils.status import format_status, rollup
    fv = bs.func(lambda: bs.Reduce(
        bs.Const(3, torch.arange(30, dtype=torch.int64) % 4,
                 torch.ones(30, dtype=torch.int64)), "sum"))
    sess = bs.start(parallelism=2, device="cpu")
    res = sess.run(fv)
    r = rollup(res.tasks)
    assert all(v == {"OK": 3} for v in r.values())
    assert "3/3" in format_status(res.tasks)


def test_machine_stats():
    from bigslice_amd.utils.machine import machine_stats
    s = machine_stats()
    assert s["rss_gb"] > 0


def test_eventlog_file(tmp_path):
    path = str(tmp_path / "events.jsonl")
    fv = bs.func(lambda: bs.Const(2, torch.arange(4, dtype=torch.int64)))
    sess = bs.start(parallelism=2, device="cpu", eventlog_path=path)
    sess.run(fv)
    sess.eventer.close()
    lines = [json.loads(l) for l in open(path)]
    assert lines[0]["event"] == "bigslice:sessionStart"
    assert any(l["event"] == "bigslice:taskComplete" and
               l["state"] == "OK" for l in lines)


def test_slicetrace_analyzer(tmp_path):
    trace = str(tmp_path / "t.json")
    fv = bs.func(lambda: bs.Reduce(
        bs.Const(2, torch.arange(10, dtype=torch.int64) % 3,
                 torch.ones(10, dtype=torch.int64)), "sum"))
    sess = bs.start(parallelism=2, device="cpu", trace_path=trace)
    sess.run(fv)
    sess.shutdown()
    import subprocess
    import sys as _sys
    p = subprocess.run(
        [_sys.executable, "-m", "bigslice_amd.tools.slicetrace", trace],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert "reduce" in p.stdout


def test_stats_map():
    from bigslice_amd.utils import stats
    m = stats.Map()
    m.add("x", 2)
    m.add("x")
    m.merge({"y": 5})
    assert m.values() == {"x": 3, "y": 5}
    assert "x=3" in str(m)


def test_concurrent_eval_and_discard_chaos():
    # TestDiscardChaos analog: interleave runs of the same Func with
    # discards of prior results; everything stays correct.
    import threading

    fv = bs.func(lambda: bs.Reduce(
        bs.Const(4, torch.arange(400, dtype=torch.int64) % 9,
                 torch.ones(400, dtype=torch.int64)), "sum"))
    sess = bs.start(parallelism=4, device="cpu")
    want = {}
    for k in (torch.arange(400) % 9).tolist():
        want[k] = want.get(k, 0) + 1
    errs = []

    def worker():
        try:
            for _ in range(5):
                res = sess.run(fv)
                got = dict(res.scan())
                assert got == want, got
                res.discard()
        except BaseException as e:
            errs.append(e)

    ts = [threading.Thread(target=worker) for _ in range(3)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(120)
    assert not errs, errs


def test_exclusive_func_serializes_tasks():
    import threading
    import time as _time
    active, peak = [], [0]
    lock = threading.Lock()

    def track(x):
        with lock:
            active.append(1)
            peak[0] = max(peak[0], len(active))
        _time.sleep(0.02)
        with lock:
            active.pop()
        return (x,)

    fv = bs.func(lambda: bs.Map(
        bs.Const(4, torch.arange(8, dtype=torch.int64)), track,
        out_schema=(int,)), exclusive=True)
    sess = bs.start(parallelism=4, device="cpu")
    res = sess.run(fv)
    assert len(list(res.scan())) == 8
    assert peak[0] == 1


def test_reader_helpers():
    from bigslice_amd.sliceio import (ErrReader, FuncReader, Scanner,
                                      read_all_or_empty, EmptyReader)
    from bigslice_amd.schema import Schema
    with pytest.raises(ValueError):
        ErrReader(ValueError("x")).read()
    calls = []

    def fn():
        if calls:
            return None
        calls.append(1)
        from bigslice_amd.frame import Frame
        return Frame([torch.arange(3, dtype=torch.int64)])
    fr = FuncReader(fn)
    frames = list(fr)
    assert len(frames) == 1
    empty = read_all_or_empty(EmptyReader(), Schema([torch.int64]))
    assert len(empty) == 0
    s = Scanner(EmptyReader())
    assert list(s.frames()) == []


def test_materialize_executes_as_checkpoint():
    computed = []

    def gen(shard, ctx):
        computed.append(shard)
        yield (torch.arange(3, dtype=torch.int64) + shard * 10,)

    def build():
        src = bs.ReaderFunc(2, gen, bs.schema_of(int))
        bs.materialize(src)
        a = bs.Map(src, lambda x: (x + 1,))
        b = bs.Map(src, lambda x: (x * 2,))
        return bs.Cogroup(bs.Map(a, lambda x: (x, x)),
                          bs.Map(b, lambda x: (x, x)))

    res = bs.slicetest.run(build)
    assert len(list(res.scan())) > 0
    # materialized source computed once per shard despite two consumers
    assert sorted(computed) == [0, 1]


def test_top_n():
    from bigslice_amd.utils.status import top_n
    counts = {"a": 5, "b": 9, "c": 1, "d": 9}
    assert top_n(counts, 2) == [("d", 9), ("b", 9)]
    assert top_n(counts, 10)[-1] == ("c", 1)


def test_store_disk_tier_roundtrip(monkeypatch):
    # Force the disk tier and check exact roundtrip + cleanup.
    from bigslice_amd.frame import Frame
    from bigslice_amd.runtime.store import MemoryStore, _DiskEntry
    st = MemoryStore()
    monkeypatch.setattr(st, "_tier_mode", lambda frames: "disk")
    k = torch.arange(5000, dtype=torch.int64)
    v = torch.randint(-9, 9, (5000,), dtype=torch.int64)
    st.put("t", 2, [Frame([k, v], prefix=1)], 5000)
    entry = st._data[("t", 2)][0]
    assert isinstance(entry, _DiskEntry) and os.path.exists(entry.path)
    assert st.stat("t", 2)[1] == 5000
    back = list(st.open("t", 2))
    assert torch.equal(back[0].columns[0], k)
    assert torch.equal(back[0].columns[1], v)
    path = entry.path
    st.discard_task("t")
    assert not os.path.exists(path)
    assert not st.has("t", 2)


def test_session_context_manager_and_result_iter():
    def build():
        return bs.Const(2, torch.tensor([1, 2, 1], dtype=torch.int64),
                        torch.tensor([5, 6, 7], dtype=torch.int64))
    with bs.start(parallelism=2, device="cpu") as sess:
        rows = sorted(sess.run(bs.func(build)))
    assert rows == [(1, 5), (1, 7), (2, 6)]


def test_machine_combiners_off_matches_on():
    def build():
        k = torch.arange(5000, dtype=torch.int64) % 37
        v = torch.ones(5000, dtype=torch.int64)
        return bs.Reduce(bs.Const(6, k, v, prefix=1), "sum")
    outs = []
    for mc in (True, False):
        sess = bs.start(parallelism=4, device="cpu",
                        machine_combiners=mc)
        outs.append(sorted(sess.run(bs.func(build)).scan()))
    assert outs[0] == outs[1]
    assert len(outs[0]) == 37
