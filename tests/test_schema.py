"""Schema validation + misc tool analogs (reference
tools/validate-schema, tools/json_streamer, schema/*.js)."""

import json
import subprocess
import sys

from dragnet_amd.schema import (SCHEMAS, SchemaError, validate)


def ok(name, data):
    validate(SCHEMAS[name], data)


def bad(name, data):
    import pytest
    with pytest.raises(SchemaError):
        validate(SCHEMAS[name], data)


def test_user_index_schema():
    ok("user-index", {"name": "idx", "format": "json",
                      "columns": ["a", "b.c"]})
    ok("user-index", {"name": "idx", "format": "json",
                      "columns": [{"name": "lat", "field": "latency",
                                   "aggr": "quantize"}],
                      "filter": {"eq": ["a", 1]},
                      "fsroot": "/data"})
    bad("user-index", {"format": "json", "columns": []})   # no name
    bad("user-index", {"name": "x", "columns": []})        # no format
    bad("user-index", {"name": "x", "format": "csv",
                       "columns": []})                     # bad enum
    bad("user-index", {"name": "x", "format": "json"})     # no columns
    bad("user-index", {"name": "x", "format": "json",
                       "columns": [42]})                   # bad item
    bad("user-index", {"name": "x", "format": "json",
                       "columns": [{"field": "f"}]})       # no col name


def test_user_query_schema():
    ok("user-query", {"index": "i"})
    ok("user-query", {"index": "i", "timeStart": "2014-05-01",
                      "timeResolution": 60,
                      "breakdowns": ["a", "b"]})
    bad("user-query", {})                                  # no index
    bad("user-query", {"index": 5})
    bad("user-query", {"index": "i", "breakdowns": [1]})
    bad("user-query", {"index": "i", "timeResolution": "x"})


def test_validate_schema_cli(tmp_path):
    f = tmp_path / "idx.json"
    f.write_text(json.dumps({"name": "x", "format": "json",
                             "columns": ["a"]}))
    r = subprocess.run(
        [sys.executable, "-m", "dragnet_amd.tools.validate_schema",
         "user-index", str(f)], capture_output=True, text=True)
    assert r.returncode == 0
    assert r.stdout.strip() == "%s okay" % f

    f2 = tmp_path / "bad.json"
    f2.write_text(json.dumps({"format": "json", "columns": []}))
    r = subprocess.run(
        [sys.executable, "-m", "dragnet_amd.tools.validate_schema",
         "user-index", str(f2)], capture_output=True, text=True)
    assert r.returncode == 1
    assert "missing and required" in r.stderr

    r = subprocess.run(
        [sys.executable, "-m", "dragnet_amd.tools.validate_schema",
         "nope", str(f)], capture_output=True, text=True)
    assert r.returncode == 2
    assert "available schemas" in r.stderr


def test_json_streamer(tmp_path):
    data = (b'{"a": 1}\n{"b": 2}\nnot json\n{"c": 3}\n')
    r = subprocess.run(
        [sys.executable, "-m", "dragnet_amd.tools.json_streamer"],
        input=data, capture_output=True)
    assert r.returncode == 0
    assert r.stdout.strip() == b"3"
    assert b"warn:" in r.stderr


def test_memwatch(tmp_path):
    import subprocess
    import sys as _sys
    p = subprocess.Popen([_sys.executable, "-c",
                          "import time; x='a'*(40<<20); time.sleep(1.2)"])
    r = subprocess.run(
        [_sys.executable, "-m", "dragnet_amd.tools.memwatch",
         str(p.pid), "0.2"], capture_output=True, text=True)
    p.wait()
    assert r.returncode == 0
    rss, vsz = map(int, r.stdout.split())
    assert rss > 40 * 1024  # saw the 40 MB allocation (KB units)
    assert vsz >= rss
