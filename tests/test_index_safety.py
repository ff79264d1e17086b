"""Index crash-safety: tmp+rename atomicity (reference
lib/index-sink.js:64,288-297) and abort cleanup."""

import os

from dragnet_amd.index import IndexQuerier, IndexSink

METRICS = [{"name": "m", "filter": None,
            "breakdowns": [{"name": "host", "field": "host"}]}]


def test_sink_atomic_rename(tmp_path):
    f = str(tmp_path / "idx.sqlite")
    sink = IndexSink(f, METRICS)
    sink.write_point({"fields": {"__dn_metric": 0, "host": "a"},
                      "value": 3})
    # before flush: only the tmp file exists
    assert not os.path.exists(f)
    assert os.path.exists(sink.tmpfilename)
    sink.flush()
    assert os.path.exists(f)
    assert not os.path.exists(sink.tmpfilename)

    iq = IndexQuerier(f)
    assert iq.config["version"] == "2.0.0"
    assert iq.metrics[0]["label"] == "m"
    iq.close()


def test_sink_abort_leaves_nothing(tmp_path):
    f = str(tmp_path / "idx.sqlite")
    sink = IndexSink(f, METRICS)
    sink.write_point({"fields": {"__dn_metric": 0, "host": "a"},
                      "value": 1})
    sink.abort()
    assert not os.path.exists(f)
    assert not os.path.exists(sink.tmpfilename)


def test_rebuild_clobbers(tmp_path):
    """Rebuilding replaces the index atomically; readers of the old
    file are unaffected (rename semantics)."""
    f = str(tmp_path / "idx.sqlite")
    for v in (1, 2):
        sink = IndexSink(f, METRICS)
        sink.write_point({"fields": {"__dn_metric": 0, "host": "a"},
                          "value": v})
        sink.flush()
    iq = IndexQuerier(f)
    rows = list(iq.db.execute("SELECT host, value FROM dragnet_index_0"))
    iq.close()
    assert [tuple(r) for r in rows] == [("a", 2)]


def test_native_vs_python_sink_identical(tmp_path):
    """The native C-API sink and the pure-Python sqlite3 sink must
    produce byte-equivalent logical content (schema + rows)."""
    import os as _os
    import sqlite3

    from dragnet_amd.index import sink as sinkmod
    metrics = [
        {"name": "m", "filter": {"eq": ["a", "x"]},
         "breakdowns": [{"name": "host", "field": "host"},
                        {"name": "lat", "field": "lat",
                         "aggr": "quantize"},
                        {"name": "t", "field": "t", "date": "",
                         "aggr": "lquantize", "step": 86400}]},
        {"name": "m2", "filter": None,
         "breakdowns": [{"name": "op", "field": "op"}]},
    ]
    points = []
    for i in range(500):
        points.append({"fields": {"__dn_metric": 0,
                                  "host": "h%d" % (i % 7),
                                  "lat": 2 ** (i % 10),
                                  "t": 86400 * (i % 3)},
                       "value": i + 1})
        points.append({"fields": {"__dn_metric": 1,
                                  "op": "op%d" % (i % 5)},
                       "value": 2})

    def build(path, force_py):
        if force_py:
            _os.environ["DRAGNET_PY_SINK"] = "1"
        else:
            _os.environ.pop("DRAGNET_PY_SINK", None)
        try:
            s = sinkmod.IndexSink(path, metrics,
                                  config={"dn_start": 123})
            assert (s._cs is None) == force_py
            for p in points:
                s.write_point(p)
            s.flush()
        finally:
            _os.environ.pop("DRAGNET_PY_SINK", None)

    f_native = str(tmp_path / "native.sqlite")
    f_py = str(tmp_path / "py.sqlite")
    build(f_native, False)
    build(f_py, True)

    def dump(path):
        db = sqlite3.connect(path)
        out = {}
        for (tbl,) in db.execute(
                "SELECT name FROM sqlite_master WHERE type='table' "
                "ORDER BY name"):
            rows = db.execute("SELECT * FROM %s" % tbl).fetchall()
            out[tbl] = sorted(map(tuple, rows))
        db.close()
        return out

    assert dump(f_native) == dump(f_py)


def test_colliding_column_names_clean_error(dn, fixture_tree,
                                            tmp_path):
    """Breakdown names that collide after [.-]->_ escaping (or hit an
    SQL keyword) fail the build with a clean `dn:` error, matching
    the reference's unquoted CREATE TABLE failure mode — never a
    traceback, and no stray tmp file."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "cidx")
    assert dn("datasource-add", "c", "--path=" + one,
              "--index-path=" + idx, "--time-field=time").code == 0
    assert dn("metric-add", "c", "m", "-b", "a-b,a.b").code == 0
    r = dn("build", "c")
    assert r.code == 1
    assert r.err.startswith("dn: cannot materialize index:")
    assert "duplicate column" in r.err
    # tmp files cleaned up (abort path)
    leftovers = []
    for root, _d, names in os.walk(idx):
        leftovers += [n for n in names if not n.endswith(".sqlite")]
    assert leftovers == [], leftovers
