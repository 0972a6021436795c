"""Committed golden-graph compiler tests.

Role-parity: exec/compile_test.go:23-137 + exec/testdata/*.graph — the
six reference fixtures (trivial / shuffle / branch / branch-materialize
/ branch-shuffle / branch-different-partitions) compiled and diffed
against goldens committed under tests/testdata/.  Golden diffs catch
naming, fusion, partition-count and combine-spec regressions that
property assertions miss.

Regenerate after an intentional compiler change with:
    python -m pytest tests/test_compile_golden.py --update-goldens
(implemented via the BIGSLICE_UPDATE_GOLDENS env var in conftest).
"""

import difflib
import os

import pytest
import torch

import bigslice_amd as bs
from bigslice_amd.ops.slice_base import Pragma
from bigslice_amd.runtime.compile import Compiler
from bigslice_amd.runtime.task import graph_string

TESTDATA = os.path.join(os.path.dirname(__file__), "testdata")


def _const(n=2):
    return bs.Const(n, torch.arange(6, dtype=torch.int64),
                    torch.ones(6, dtype=torch.int64))


def fixture_trivial():
    return [bs.Map(bs.Filter(_const(), lambda k, v: k > 0),
                   lambda k, v: (k, v))]


def fixture_shuffle():
    return [bs.Reduce(bs.Map(_const(), lambda k, v: (k, v)), "sum")]


def fixture_branch():
    # two pipelines over one base: without materialize, fusion clones
    # the base into each consumer's pipeline (same as the reference)
    base = bs.Map(_const(), lambda k, v: (k, v + 1))
    return [bs.Map(base, lambda k, v: (k, v * 2)),
            bs.Filter(base, lambda k, v: v > 0)]


def fixture_branch_materialize():
    base = bs.Map(_const(), lambda k, v: (k, v + 1))
    base.pragma = Pragma(materialize=True)
    return [bs.Map(base, lambda k, v: (k, v * 2)),
            bs.Filter(base, lambda k, v: v > 0)]


def fixture_branch_shuffle():
    # two reduce consumers with identical combine specs share producer
    # tasks; a third with a different spec gets its own producers
    base = bs.Map(_const(), lambda k, v: (k, v))
    return [bs.Reduce(base, "sum"), bs.Reduce(base, "sum"),
            bs.Reduce(base, "max")]


def fixture_branch_different_partitions():
    # consumers at different shard counts must compile separate
    # producer task sets (partition fan differs)
    base = bs.Map(_const(4), lambda k, v: (k, v))
    return [bs.Reshard(bs.Reshuffle(base), 2),
            bs.Reshard(bs.Reshuffle(base), 3)]


FIXTURES = {
    "trivial": fixture_trivial,
    "shuffle": fixture_shuffle,
    "branch": fixture_branch,
    "branch_materialize": fixture_branch_materialize,
    "branch_shuffle": fixture_branch_shuffle,
    "branch_different_partitions": fixture_branch_different_partitions,
}


@pytest.mark.parametrize("name", sorted(FIXTURES))
def test_golden_graph(name):
    roots = FIXTURES[name]()
    c = Compiler(1)
    tasks = []
    for r in roots:
        tasks.extend(c.compile(r))
    got = graph_string(tasks, detail=True) + "\n"
    path = os.path.join(TESTDATA, name + ".graph")
    if os.environ.get("BIGSLICE_UPDATE_GOLDENS"):
        os.makedirs(TESTDATA, exist_ok=True)
        with open(path, "w") as fp:
            fp.write(got)
        return
    with open(path) as fp:
        want = fp.read()
    if got != want:
        diff = "".join(difflib.unified_diff(
            want.splitlines(keepends=True), got.splitlines(keepends=True),
            fromfile=f"testdata/{name}.graph", tofile="compiled"))
        raise AssertionError(f"task graph diverged from golden:\n{diff}")
