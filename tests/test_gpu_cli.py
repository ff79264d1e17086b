"""
End-to-end CLI on the GPU engine (pytest -m gpu): scan/build/query via
`dn` with DRAGNET_ENGINE=gpu must produce byte-identical output to the
CPU engine.
"""

import os

import pytest

from scan_cases import SCAN_CASES

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def require_gpu():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def run_both(dn, monkeypatch, argv):
    monkeypatch.setenv("DRAGNET_ENGINE", "cpu")
    c = dn(*argv)
    monkeypatch.setenv("DRAGNET_ENGINE", "gpu")
    g = dn(*argv)
    return c, g


def test_cli_scan_gpu_vs_cpu(dn, fixture_tree, monkeypatch):
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "src", "--path=" + one)
    assert r.code == 0, r.err
    for case in SCAN_CASES:
        for extra in ((), ("--points",)):
            argv = ["scan", *extra, *case, "src"]
            c, g = run_both(dn, monkeypatch, argv)
            assert c.code == 0 and g.code == 0, (argv, g.err)
            assert g.out == c.out, argv


def test_cli_build_query_gpu(dn, fixture_tree, tmp_path, monkeypatch):
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idx")
    monkeypatch.setenv("DRAGNET_ENGINE", "gpu")
    r = dn("datasource-add", "input", "--path=" + one,
           "--index-path=" + idx, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "input", "m", "-b",
           "host,operation,req.caller,req.method,latency[aggr=quantize]")
    assert r.code == 0, r.err
    r = dn("build", "input")
    assert r.code == 0, r.err
    for case in SCAN_CASES[:-1]:
        s = dn("scan", *case, "input")
        q = dn("query", *case, "input")
        assert s.code == 0 and q.code == 0, (case, q.err)
        assert q.out == s.out, case


def test_cli_query_index_gpu_path(dn, fixture_tree, tmp_path,
                                  monkeypatch):
    """K7: DRAGNET_INDEX_GPU=1 answers index queries by streaming the
    stored rows through the fused kernel as skinner points; results
    must equal the SQLite path."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idxg")
    monkeypatch.setenv("DRAGNET_ENGINE", "gpu")
    r = dn("datasource-add", "input", "--path=" + one,
           "--index-path=" + idx, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "input", "m", "-b",
           "host,operation,req.method,latency[aggr=quantize]")
    assert r.code == 0, r.err
    r = dn("build", "input")
    assert r.code == 0, r.err
    cases = [(), ("-b", "operation"),
             ("-f", '{ "eq": [ "req.method", "GET" ] }',
              "-b", "host,operation"),
             ("-b", "host,latency[aggr=quantize]")]
    for case in cases:
        monkeypatch.delenv("DRAGNET_INDEX_GPU", raising=False)
        sql_res = dn("query", *case, "input")
        monkeypatch.setenv("DRAGNET_INDEX_GPU", "1")
        gpu_res = dn("query", *case, "input")
        assert sql_res.code == 0 and gpu_res.code == 0, \
            (case, gpu_res.err)
        assert gpu_res.out == sql_res.out, case


def test_cli_counters_gpu_vs_cpu(dn, fixture_tree, monkeypatch):
    r = dn("datasource-add", "tree", "--path=" + fixture_tree,
           "--time-format=%Y/%m-%d", "--time-field=time")
    assert r.code == 0, r.err
    argv = ["scan", "--counters",
            "-b", "timestamp[date,field=time,aggr=lquantize,step=86400]",
            "tree"]
    c, g = run_both(dn, monkeypatch, argv)
    assert c.code == 0 and g.code == 0, g.err
    assert g.out == c.out
    # the pipeline counter dumps agree stage by stage (FindX stages and
    # json parser/Datetime/Aggregator taxonomies are engine-agnostic)
    assert g.err == c.err


def test_cli_sharded_backend_gpu(dn, fixture_tree, monkeypatch):
    """Sharded backend, world=1, GPU engine: equals the file backend."""
    monkeypatch.setenv("DRAGNET_ENGINE", "gpu")
    r = dn("datasource-add", "sh", "--backend=sharded",
           "--path=" + fixture_tree)
    assert r.code == 0, r.err
    r = dn("datasource-add", "fl", "--path=" + fixture_tree)
    assert r.code == 0, r.err
    a = dn("scan", "-b", "req.method", "sh")
    b = dn("scan", "-b", "req.method", "fl")
    assert a.code == 0 and b.code == 0, (a.err, b.err)
    assert a.out == b.out


def test_cli_query_index_gpu_columnar_bounds(dn, fixture_tree,
                                             tmp_path, monkeypatch):
    """K7 columnar with time-bounded queries over an interval tree
    (__dn_ts date column in the WHERE) and with weighted rows: GPU
    columnar == SQLite."""
    idx = str(tmp_path / "idxb")
    monkeypatch.setenv("DRAGNET_ENGINE", "gpu")
    r = dn("datasource-add", "tree", "--path=" + fixture_tree,
           "--index-path=" + idx, "--time-field=time",
           "--time-format=%Y/%m-%d")
    assert r.code == 0, r.err
    # time-bounded index queries need an explicit date breakdown in
    # the metric (findMetric requires a date param; reference
    # lib/index-query.js:190-203)
    r = dn("metric-add", "tree", "m", "-b",
           "ts[date,field=time,aggr=lquantize,step=3600],"
           "host,req.method,latency[aggr=quantize]")
    assert r.code == 0, r.err
    r = dn("build", "tree")
    assert r.code == 0, r.err
    cases = [("-b", "host", "--after", "2014-05-01",
              "--before", "2014-05-03"),
             ("-b", "req.method,latency[aggr=quantize]",
              "--after", "2014-05-02", "--before", "2014-05-05"),
             ("-b", "ts[date,field=time,aggr=lquantize,step=3600],host",
              "--after", "2014-05-02", "--before", "2014-05-03"),
             ("-b", "host,req.method")]
    for case in cases:
        monkeypatch.setenv("DRAGNET_INDEX_GPU", "0")
        sql_res = dn("query", *case, "tree")
        monkeypatch.setenv("DRAGNET_INDEX_GPU", "1")
        gpu_res = dn("query", *case, "tree")
        assert sql_res.code == 0 and gpu_res.code == 0, \
            (case, gpu_res.err)
        assert gpu_res.out == sql_res.out, case


def test_gpu_build_tree_equals_cpu(dn, tmp_path, monkeypatch):
    """Multi-metric index BUILD on the GPU engine (dense partials +
    MFMA reduce + native sink) produces a logically identical index
    tree to the CPU oracle build."""
    import sqlite3

    from dragnet_amd.tools.mktestdata import generate_lines
    data = tmp_path / "data"
    data.mkdir()
    lines = list(generate_lines(30000, seed=321))
    (data / "a.log").write_bytes(b"".join(lines[:17000]))
    (data / "b.log").write_bytes(b"".join(lines[17000:]))

    def build(engine, idx):
        monkeypatch.setenv("DRAGNET_ENGINE", engine)
        r = dn("datasource-add", "src_" + engine,
               "--path=" + str(data), "--index-path=" + idx,
               "--time-field=time")
        assert r.code == 0, r.err
        r = dn("metric-add", "src_" + engine, "requests", "-b",
               "host,req.method,res.statusCode,operation,"
               "latency[aggr=quantize]")
        assert r.code == 0, r.err
        r = dn("metric-add", "src_" + engine, "errors",
               "--filter", '{ "ge": [ "res.statusCode", 500 ] }',
               "-b", "operation")
        assert r.code == 0, r.err
        r = dn("build", "src_" + engine)
        assert r.code == 0, r.err

    idx_g = str(tmp_path / "idx_gpu")
    idx_c = str(tmp_path / "idx_cpu")
    build("gpu", idx_g)
    build("cpu", idx_c)

    def dump_tree(root):
        out = {}
        byday = os.path.join(root, "by_day")
        for name in sorted(os.listdir(byday)):
            db = sqlite3.connect(os.path.join(byday, name))
            tdump = {}
            for (tbl,) in db.execute(
                    "SELECT name FROM sqlite_master WHERE "
                    "type='table' ORDER BY name"):
                rows = db.execute("SELECT * FROM %s" % tbl).fetchall()
                tdump[tbl] = sorted(map(tuple, rows))
            db.close()
            out[name] = tdump
        return out

    g = dump_tree(idx_g)
    c = dump_tree(idx_c)
    assert list(g.keys()) == list(c.keys())  # same day files
    assert g == c


def test_cli_stdin_scan_gpu(tmp_path):
    """Char-device datasources (/dev/stdin) through the GPU engine:
    the sequential-read path must produce byte-identical output to the
    CPU engine (the reference uses stdin datasources as fixtures
    throughout its suite).  Runs dn as a SUBPROCESS so the real
    /dev/stdin carries the bytes."""
    import subprocess
    import sys as _sys

    from dragnet_amd.tools.mktestdata import generate_lines
    pool = b"".join(generate_lines(20000, seed=55))
    cfg = str(tmp_path / "rc.json")
    env = dict(os.environ, DRAGNET_CONFIG=cfg)
    subprocess.run([_sys.executable, "-m", "dragnet_amd.cli",
                    "datasource-add", "stdin", "--path=/dev/stdin"],
                   env=env, check=True)
    filt = "{ \"eq\": [ \"req.method\", \"GET\" ] }"
    argv = [_sys.executable, "-m", "dragnet_amd.cli", "scan",
            "-f", filt,
            "-b", "req.method,res.statusCode", "stdin"]
    # redirect stdin FROM A FILE (the reference's stdin fixtures do
    # the same: under a pipe /dev/stdin is a FIFO, which fs-find
    # skips — regular files and char devices only)
    pf = tmp_path / "pool.ndjson"
    pf.write_bytes(pool)
    out = {}
    for engine in ("cpu", "gpu"):
        e = dict(env, DRAGNET_ENGINE=engine)
        with open(pf, "rb") as fh:
            r = subprocess.run(argv, stdin=fh, capture_output=True,
                               env=e)
        assert r.returncode == 0, (engine, r.stderr[-1500:])
        out[engine] = r.stdout
    assert out["gpu"] == out["cpu"]
    assert b"GET" in out["gpu"]


def test_cli_index_pipe_gpu(dn, fixture_tree, tmp_path, monkeypatch):
    """index-scan | index-read with the GPU engine on the map side ==
    the CPU pipe (map scans run the fused kernel; the reduce runs the
    native point codec)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    outs = {}
    for eng in ("cpu", "gpu"):
        monkeypatch.setenv("DRAGNET_ENGINE", eng)
        src = "psrc_" + eng
        dst = "pdst_" + eng
        assert dn("datasource-add", src, "--path=" + one,
                  "--index-path=" + str(tmp_path / (eng + "0")),
                  "--time-field=time").code == 0
        assert dn("metric-add", src, "m", "-b",
                  "operation,req.method").code == 0
        r = dn("index-scan", src)
        assert r.code == 0, r.err
        assert dn("datasource-add", dst, "--path=/dev/null",
                  "--index-path=" + str(tmp_path / ("idx_" + eng)),
                  "--time-field=time").code == 0
        assert dn("metric-add", dst, "m", "-b",
                  "operation,req.method").code == 0
        assert dn("index-read", dst, stdin=r.out.encode()).code == 0
        q = dn("query", "-b", "operation", dst)
        assert q.code == 0, q.err
        outs[eng] = (r.out, q.out)
    assert outs["gpu"] == outs["cpu"]
    assert outs["gpu"][1] != ""
