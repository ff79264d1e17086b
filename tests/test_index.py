"""
Index subsystem tests: the core invariant is EQUIVALENCE — an
index-backed `dn query` must reproduce raw-scan answers (the reference
runs its shared scan cases against both paths, tst.index_file.sh:14-37).
"""

import os
import sqlite3

from scan_cases import INDEX_EQUIV_CASES

BIG_METRIC = "host,operation,req.caller,req.method,latency[aggr=quantize]"


def setup_indexed(dn, fixture_tree, tmp_path, interval="day",
                  metric_breakdowns=BIG_METRIC):
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idx")
    r = dn("datasource-add", "input", "--path=" + one,
           "--index-path=" + idx, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "input", "big_metric", "-b", metric_breakdowns)
    assert r.code == 0, r.err
    r = dn("build", "--interval=" + interval, "input")
    assert r.code == 0, r.err
    return idx


def test_query_equals_scan(dn, fixture_tree, tmp_path):
    setup_indexed(dn, fixture_tree, tmp_path)
    for case in INDEX_EQUIV_CASES:
        s = dn("scan", *case, "input")
        q = dn("query", *case, "input")
        assert s.code == 0, (case, s.err)
        assert q.code == 0, (case, q.err)
        assert q.out == s.out, case
        sp = dn("scan", "--points", *case, "input")
        qp = dn("query", "--points", *case, "input")
        assert sorted(qp.out.splitlines()) == sorted(sp.out.splitlines())


def test_query_interval_all(dn, fixture_tree, tmp_path):
    idx = setup_indexed(dn, fixture_tree, tmp_path, interval="all")
    assert os.path.exists(os.path.join(idx, "all"))
    s = dn("scan", "-b", "operation", "input")
    q = dn("query", "--interval=all", "-b", "operation", "input")
    assert q.out == s.out


def test_index_file_layout(dn, fixture_tree, tmp_path):
    """Index tree layout + schema (reference lib/dragnet-impl.js:194-236,
    lib/index-sink.js:116-163)."""
    idx = setup_indexed(dn, fixture_tree, tmp_path, interval="day")
    f = os.path.join(idx, "by_day", "2014-05-01.sqlite")
    assert os.path.exists(f)
    db = sqlite3.connect(f)
    cfg = dict(db.execute("SELECT key, value FROM dragnet_config"))
    assert cfg["version"] == "2.0.0"
    assert int(cfg["dn_start"]) == 1398902400
    mets = db.execute(
        "SELECT id, label, filter, params FROM dragnet_metrics"
    ).fetchall()
    assert len(mets) == 1 and mets[0][1] == "big_metric"
    cols = [r[1] for r in db.execute(
        "PRAGMA table_info(dragnet_index_0)")]
    assert cols == ["host", "operation", "req_caller", "req_method",
                    "latency", "value"]
    total = db.execute(
        "SELECT SUM(value) FROM dragnet_index_0").fetchone()[0]
    assert total == 250
    db.close()


def test_filtered_metric(dn, fixture_tree, tmp_path):
    """A metric with a filter serves only exact-filter-match queries
    (reference tst.index_file.sh:33-39, findMetric semantics)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idx2")
    r = dn("datasource-add", "input", "--path=" + one,
           "--index-path=" + idx, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "input", "filtered_metric",
           "-f", '{ "eq": [ "req.method", "GET" ] }')
    assert r.code == 0, r.err
    r = dn("build", "input")
    assert r.code == 0, r.err

    q = dn("query", "-f", '{ "eq": [ "req.method", "GET" ] }', "input")
    s = dn("scan", "-f", '{ "eq": [ "req.method", "GET" ] }', "input")
    assert q.out == s.out
    # a non-matching filter cannot be served
    q = dn("query", "-f", '{ "eq": [ "req.method", "PUT" ] }', "input")
    assert q.code != 0
    assert "no metrics available" in q.err


def test_ds_filter_baked_into_index(dn, fixture_tree, tmp_path):
    """Datasource filter applies at build; queries need no filter
    (reference tst.index_file.sh:42-52)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idx3")
    r = dn("datasource-add", "input", "--path=" + one,
           "--index-path=" + idx, "--time-field=time",
           "--filter", '{ "eq": [ "req.method", "GET" ] }')
    assert r.code == 0, r.err
    r = dn("metric-add", "input", "bycode", "-b", "res.statusCode")
    assert r.code == 0, r.err
    r = dn("build", "input")
    assert r.code == 0, r.err

    q = dn("query", "input")
    s = dn("scan", "input")
    assert q.out == s.out
    q = dn("query", "-f", '{ "eq": [ "res.statusCode", 200 ] }', "input")
    s = dn("scan", "-f", '{ "eq": [ "res.statusCode", 200 ] }', "input")
    assert q.out == s.out


def test_before_after_query(dn, fixture_tree, tmp_path):
    """Hourly indexes + before/after pruning across the fileset."""
    idx = str(tmp_path / "idx4")
    r = dn("datasource-add", "tree", "--path=" + fixture_tree,
           "--index-path=" + idx, "--time-field=time",
           "--time-format=%Y/%m-%d")
    assert r.code == 0, r.err
    # time-bounded queries need a date breakdown in the metric
    # (findMetric requires a date param; reference
    # lib/index-query.js:190-203)
    r = dn("metric-add", "tree", "m", "-b",
           "timestamp[date,field=time,aggr=lquantize,step=60],"
           "operation,req.method")
    assert r.code == 0, r.err
    r = dn("build", "tree")
    assert r.code == 0, r.err
    assert os.path.exists(
        os.path.join(idx, "by_day", "2014-05-03.sqlite"))

    case = ("--after", "2014-05-02", "--before", "2014-05-04",
            "-b", "operation")
    s = dn("scan", *case, "tree")
    q = dn("query", *case, "tree")
    assert s.code == 0, s.err
    assert q.code == 0, q.err
    assert q.out == s.out


def test_index_scan_index_read_pipeline(dn, fixture_tree, tmp_path):
    """index-scan | index-read == build (the distributed build's
    map/reduce decomposition, reference lib/datasource-manta.js:45-78)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx_a = str(tmp_path / "idxa")
    idx_b = str(tmp_path / "idxb")

    r = dn("datasource-add", "a", "--path=" + one,
           "--index-path=" + idx_a, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "a", "m", "-b", "operation,req.method")
    assert r.code == 0, r.err
    r = dn("build", "a")
    assert r.code == 0, r.err

    r = dn("index-scan", "a")
    assert r.code == 0, r.err
    points = r.out

    r = dn("datasource-add", "b", "--path=/dev/null",
           "--index-path=" + idx_b, "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "b", "m", "-b", "operation,req.method")
    assert r.code == 0, r.err
    r = dn("index-read", "b", stdin=points.encode())
    assert r.code == 0, r.err

    qa = dn("query", "-b", "operation", "a")
    qb = dn("query", "-b", "operation", "b")
    assert qa.out == qb.out
    assert qa.out != ""


def test_index_read_codec_vs_python_cli(dn, fixture_tree, tmp_path,
                                        monkeypatch):
    """The native point codec and the pure-Python loop produce
    identical indexes through the real `dn index-scan | dn
    index-read` pipe (DRAGNET_PY_POINTS=1 forces Python end-to-end:
    both the reduce parse and the emit serializer)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "src", "--path=" + one,
           "--index-path=" + str(tmp_path / "i0"),
           "--time-field=time")
    assert r.code == 0, r.err
    r = dn("metric-add", "src", "m", "-b",
           "operation,req.method,latency[aggr=quantize]")
    assert r.code == 0, r.err

    outs = {}
    for mode in ("native", "python"):
        if mode == "python":
            monkeypatch.setenv("DRAGNET_PY_POINTS", "1")
        else:
            monkeypatch.delenv("DRAGNET_PY_POINTS", raising=False)
        r = dn("index-scan", "src")
        assert r.code == 0, r.err
        points = r.out
        name = "dst_" + mode
        idx = str(tmp_path / ("idx_" + mode))
        assert dn("datasource-add", name, "--path=/dev/null",
                  "--index-path=" + idx,
                  "--time-field=time").code == 0
        assert dn("metric-add", name, "m", "-b",
                  "operation,req.method,latency[aggr=quantize]"
                  ).code == 0
        assert dn("index-read", name, stdin=points.encode()).code == 0
        q = dn("query", "-b", "operation,latency[aggr=quantize]", name)
        assert q.code == 0, q.err
        outs[mode] = (points, q.out)
    assert outs["native"] == outs["python"]
    assert outs["native"][1] != ""
