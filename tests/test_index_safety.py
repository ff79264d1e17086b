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
