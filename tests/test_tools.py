"""Tools, sliceconfig, tarslice, debug HTTP and pragma tests."""

import io
import json
import os
import subprocess
import sys
import tarfile
import urllib.request

import pytest
import torch

import bigslice_amd as bs


def test_slicer_reduce_cpu():
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.slicer", "reduce",
         "--nshard", "4", "--nkey", "1000", "--device", "cpu"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert "reduce OK" in p.stdout


def test_slicer_cogroup_cpu():
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.slicer", "cogroup",
         "--nshard", "3", "--nkey", "200", "--device", "cpu"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert "cogroup OK" in p.stdout


def test_slicer_memiter_cpu():
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.slicer", "memiter",
         "--iters", "3", "--device", "cpu"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr


def test_slicer_oom_reported_cpu():
    """OOM surfaces as a clean task error (cmd/slicer/oom.go analog);
    the allocator rejects the absurd size fail-fast, so this is safe."""
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.slicer", "oom",
         "--device", "cpu"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert "oom OK" in p.stdout


def test_badfuncs_late_registration():
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.badfuncs", "late"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert "caught expected" in p.stdout


def test_wordcount_tool(tmp_path):
    f = tmp_path / "in.txt"
    f.write_text("a b a\nc a\n")
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.wordcount", str(f),
         "--shards", "2", "--local"],
        capture_output=True, text=True)
    assert p.returncode == 0, p.stderr
    assert p.stdout.splitlines()[0].split() == ["3", "a"]


def test_sliceconfig_profile(tmp_path):
    prof = tmp_path / "config"
    prof.write_text("parallelism = 3\ndevice = cpu\n")
    sess, rest = bs.sliceconfig.parse(
        ["--extra", "1"], profile_path=str(prof))
    assert sess.executor.parallelism == 3
    assert rest == ["--extra", "1"]


def test_tar_reader(tmp_path):
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as tf:
        for i in range(5):
            data = f"content{i}".encode()
            info = tarfile.TarInfo(name=f"f{i}.txt")
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
    raw = buf.getvalue()

    from bigslice_amd.ops.archive import TarReader
    fv = bs.func(lambda: TarReader(2, lambda: io.BytesIO(raw)))
    res = bs.start(parallelism=2, device="cpu").run(fv)
    rows = sorted(res.scan())
    assert [r[0] for r in rows] == [f"f{i}.txt" for i in range(5)]
    assert rows[0][1] == b"content0"


def test_debug_http_endpoints():
    from bigslice_amd.utils.debug_http import serve_session
    sess = bs.start(parallelism=2, device="cpu")
    server = serve_session(sess, 0)
    port = server.server_address[1]
    fv = bs.func(lambda: bs.Const(2, torch.arange(4, dtype=torch.int64)))
    sess.run(fv)
    with urllib.request.urlopen(
            f"http://127.0.0.1:{port}/debug/tasks") as r:
        tasks = json.loads(r.read())
    assert any("const" in t["task"] for t in tasks)
    with urllib.request.urlopen(
            f"http://127.0.0.1:{port}/debug/tasks/graph") as r:
        g = json.loads(r.read())
    assert g["nodes"]
    server.shutdown()


def test_exclusive_pragma_serializes():
    import threading
    active = []
    lock = threading.Lock()
    peak = [0]

    def track(x):
        with lock:
            active.append(1)
            peak[0] = max(peak[0], len(active))
        import time
        time.sleep(0.02)
        with lock:
            active.pop()
        return (x,)

    def build():
        s = bs.Map(bs.Const(4, torch.arange(8, dtype=torch.int64)),
                   track, out_schema=(int,))
        return bs.exclusive(s)

    res = bs.slicetest.run(build, parallelism=4)
    assert len(list(res.scan())) == 8
    assert peak[0] == 1  # exclusive tasks never overlap


def test_typecheck_tool(tmp_path):
    bad = tmp_path / "bad.py"
    bad.write_text(
        "import bigslice_amd as bs\n"
        "import torch\n"
        "def build(nshard, path):\n"
        "    return bs.Const(nshard, torch.arange(3))\n"
        "fv = bs.func(build)\n"
        "sess = bs.start()\n"
        "sess.run(fv, 4)\n"          # missing `path` arg
        "sess.run(fv, 4, 'x')\n")    # correct
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.typecheck", str(bad)],
        capture_output=True, text=True)
    assert p.returncode == 1
    assert "takes 2 argument(s)" in p.stdout
    assert p.stdout.count("\n") == 1  # only the bad call flagged


def test_typecheck_tool_literal_types(tmp_path):
    # annotated builder params vs literal call args (reference
    # analysis/typecheck checks type compatibility, not just arity)
    f = tmp_path / "types.py"
    f.write_text(
        "import bigslice_amd as bs\n"
        "def build(nshard: int, path: str, scale: float):\n"
        "    return None\n"
        "fv = bs.func(build)\n"
        "sess = bs.start()\n"
        "sess.run(fv, 4, 'x', 0.5)\n"     # ok
        "sess.run(fv, 4, 'x', 2)\n"       # ok: int satisfies float
        "sess.run(fv, 'four', 'x', 1.0)\n"  # bad: str for int
        "sess.run(fv, 4, 9, 1.0)\n"       # bad: int for str
        "sess.run(fv, n, 'x', 1.0)\n")    # non-literal: skipped
    p = subprocess.run(
        [sys.executable, "-m", "bigslice_amd.tools.typecheck", str(f)],
        capture_output=True, text=True)
    assert p.returncode == 1
    assert "argument 1 is str, builder annotates int" in p.stdout
    assert "argument 2 is int, builder annotates str" in p.stdout
    assert p.stdout.count("\n") == 2


def test_kmeans_example_converges():
    sys.path.insert(0, "examples")
    import importlib
    km = importlib.import_module("kmeans")
    torch.manual_seed(0)
    g = torch.Generator().manual_seed(1)
    for s in range(4):
        km._POINTS[s] = (torch.rand(2000, generator=g),
                         torch.rand(2000, generator=g))
    sess = bs.start(parallelism=4, device="cpu")
    cx, cy = km.run_kmeans(sess, 4, 4, 8, "cpu")
    assert len(cx) == 4
    assert all(0.0 <= c <= 1.0 for c in cx + cy)


def test_gpu_wordcount_recipe_cpu():
    from collections import Counter
    from bigslice_amd import recipes
    lines = [f"a b c d{i % 7} e{i % 3}" for i in range(500)]
    ref = Counter(w for ln in lines for w in ln.split())
    sess = bs.start(parallelism=4, device="cpu")
    got = recipes.gpu_wordcount(sess, 4, lines, "cpu")
    assert got == dict(ref)


def test_bench_contract_cpu():
    # The round-end driver depends on bench.py's JSON line; guard it.
    import json
    import subprocess
    import sys as _sys
    out = subprocess.run(
        [_sys.executable, "bench.py", "--rows-per-gpu", "80000",
         "--nkeys", "1000", "--steps", "2", "--warmup", "1",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(
            __file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling",
              "vs_baseline", "dtype", "data", "config"):
        assert k in d, k
    assert d["n_gpus"] == 1 and d["scaling"] == "weak"
    assert d["higher_is_better"] is True and d["data"] == "synthetic"
    assert d["config"]["rows_per_gpu"] == 80000
    assert d["value"] > 0


def test_urls_example():
    sys.path.insert(0, "examples")
    import importlib
    urls = importlib.import_module("urls")
    text = urls.synthesize_csv(3000)
    sess = bs.start(parallelism=4)
    counts = dict(sess.run(urls.domain_counts, 4, text).scan())
    assert sum(counts.values()) == 3000
    assert set(counts) <= set(urls._DOMAINS)


def test_benchdiff(tmp_path):
    from bigslice_amd.tools import benchdiff
    rec = {"metric": "m", "value": 100.0, "unit": "u",
           "higher_is_better": True,
           "config": {"model": "M", "rows_total": 10}}
    base = tmp_path / "base"
    base.mkdir()
    (base / "base.json").write_text(json.dumps(rec) + "\n")
    good = dict(rec, value=98.0)
    bad = dict(rec, value=50.0)
    (tmp_path / "new.json").write_text(json.dumps(good) + "\n")
    assert benchdiff.main([str(tmp_path / "new.json"),
                           "--baseline-dir", str(base)]) == 0
    (tmp_path / "new2.json").write_text(json.dumps(bad) + "\n")
    assert benchdiff.main([str(tmp_path / "new2.json"),
                           "--baseline-dir", str(base)]) == 1


def test_wordcount_example_cli(tmp_path):
    import subprocess
    import sys as _sys
    path = tmp_path / "in.txt"
    path.write_text("a b a\nc a\n")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [_sys.executable, os.path.join(repo, "examples", "wordcount.py"),
         str(path)], capture_output=True, text=True, timeout=300,
        cwd=repo)
    assert out.returncode == 0, out.stderr[-1000:]
    assert out.stdout.splitlines()[0].split() == ["3", "a"]
