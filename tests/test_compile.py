"""Compiler golden tests (reference exec/compile_test.go:23-137): the
task graph string for fixture pipelines is compared against goldens."""

import textwrap

import torch

import bigslice_amd as bs
from bigslice_amd.runtime.compile import Compiler, pipeline_slices
from bigslice_amd.runtime.task import graph_string


def compile_graph(slice_):
    c = Compiler(1)
    tasks = c.compile(slice_)
    return graph_string(tasks)


def test_trivial_pipeline_fuses():
    s = bs.Map(bs.Filter(bs.Const(2, torch.arange(4, dtype=torch.int64)),
                         lambda x: x > 0), lambda x: (x,))
    chain = pipeline_slices(s)
    assert [c.name.op for c in chain] == ["map", "filter", "const"]
    g = compile_graph(s)
    assert g == textwrap.dedent("""\
        inv1_const_filter_map@2:0
        inv1_const_filter_map@2:1""")


def test_shuffle_breaks_pipeline():
    s = bs.Reduce(bs.Map(bs.Const(2, torch.arange(4, dtype=torch.int64),
                                  torch.ones(4, dtype=torch.int64)),
                         lambda k, v: (k, v)), "sum")
    g = compile_graph(s)
    assert g == textwrap.dedent("""\
        inv1_reduce@2:0
          inv1_const_map@2:0
          inv1_const_map@2:1
        inv1_reduce@2:1
          inv1_const_map@2:0
          inv1_const_map@2:1""")


def test_branch_memoized_same_combiner():
    # two consumers with IDENTICAL combine specs share producer tasks
    base = bs.Map(bs.Const(2, torch.arange(4, dtype=torch.int64),
                           torch.ones(4, dtype=torch.int64)),
                  lambda k, v: (k, v))
    r1 = bs.Reduce(base, "sum")
    r2 = bs.Reduce(base, "sum")
    c = Compiler(1)
    t1 = c.compile(r1)
    t2 = c.compile(r2)
    assert t1[0].deps[0].head_tasks[0] is t2[0].deps[0].head_tasks[0]


def test_branch_not_memoized_across_combiners():
    # different combine specs must NOT share producers (pre-combine
    # would apply the wrong aggregation)
    base = bs.Map(bs.Const(2, torch.arange(4, dtype=torch.int64),
                           torch.ones(4, dtype=torch.int64)),
                  lambda k, v: (k, v))
    r1 = bs.Reduce(base, "sum")
    r2 = bs.Reduce(base, "max")
    c = Compiler(1)
    t1 = c.compile(r1)
    t2 = c.compile(r2)
    assert t1[0].deps[0].head_tasks[0] is not t2[0].deps[0].head_tasks[0]


def test_materialize_pragma_breaks_pipeline():
    from bigslice_amd.ops.slice_base import Pragma
    base = bs.Const(2, torch.arange(4, dtype=torch.int64))
    base.pragma = Pragma(materialize=True)
    s = bs.Map(base, lambda x: (x,))
    chain = pipeline_slices(s)
    assert [c.name.op for c in chain] == ["map"]
    g = compile_graph(s)
    assert "inv1_const@2" in g and "inv1_map@2" in g


def test_shuffle_task_flags():
    s = bs.Reduce(bs.Const(2, torch.arange(4, dtype=torch.int64),
                           torch.ones(4, dtype=torch.int64)), "sum")
    c = Compiler(1)
    tasks = c.compile(s)
    prod = tasks[0].deps[0].head_tasks[0]
    assert prod.shuffle_out
    assert prod.num_partitions == 2
    assert prod.combiner is not None
    assert not tasks[0].shuffle_out


def test_cache_decision_cuts_deps(tmp_path):
    prefix = str(tmp_path / "c")
    computed = []

    def gen(shard, ctx):
        computed.append(shard)
        yield (torch.arange(3, dtype=torch.int64) + shard * 10,)

    def build():
        src = bs.ReaderFunc(2, gen, bs.schema_of(int))
        return bs.Cache(src, prefix)

    fv = bs.func(build)
    sess = bs.start(parallelism=2, device="cpu")
    r1 = sess.run(fv)
    assert sorted(r1.scan()) == [0, 1, 2, 10, 11, 12]
    n_first = len(computed)
    assert n_first == 2
    # second run: cached -> no recompute (the panics-if-computed
    # invariant, cache_test.go:28-46)
    sess2 = bs.start(parallelism=2, device="cpu")
    r2 = sess2.run(fv)
    assert sorted(r2.scan()) == [0, 1, 2, 10, 11, 12]
    assert len(computed) == n_first
    # and the cached tasks have no deps
    assert all(not t.deps for t in r2.tasks)


def test_result_reuse_distinct_combiners():
    # Two consumers with different combiners over ONE prior Result must
    # not share pass-through shuffle tasks (max previously returned
    # sum's values).
    def build_src():
        k = torch.tensor([1, 1, 2, 2], dtype=torch.int64)
        v = torch.tensor([5, 9, 3, 4], dtype=torch.int64)
        return bs.Const(2, k, v, prefix=1)

    def build_two(res):
        return bs.Cogroup(bs.Reduce(res, "sum"), bs.Reduce(res, "max"))

    sess = bs.start(parallelism=2, device="cpu")
    res = sess.run(bs.func(build_src))
    out = sorted(sess.run(bs.func(build_two), res).scan())
    assert out == [(1, [14], [9]), (2, [7], [4])], out
