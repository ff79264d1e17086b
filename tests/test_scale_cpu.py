"""
Resource/scale test (the reference's tst.scan_250k analog, SURVEY.md
§4 item 3): a large generated scan must produce the exact record count
and bounded aggregate state.  Kept to 100k records on the CPU oracle
so the suite stays fast; the GPU suite covers 200k+.
"""

import os
import resource

from dragnet_amd.engine.cpu import CpuEngine
from dragnet_amd.query import query_load
from dragnet_amd.tools.mktestdata import generate_lines

N = 100_000


def test_scan_100k_counts(tmp_path):
    path = tmp_path / "bulk.ndjson"
    with open(path, "wb") as f:
        for line in generate_lines(N, seed=42):
            f.write(line)

    res = CpuEngine().scan([str(path)], [query_load()])
    agg = res.aggregators[0]
    assert agg.points() == [{"fields": {}, "value": N}]

    # memory stays bounded by unique-tuple count, not input size
    # (README.md:666-674 scaling law): the aggregate table for a
    # count-all is a single entry
    assert len(agg.table) == 1

    res = CpuEngine().scan(
        [str(path)],
        [query_load(breakdown_specs="req.method,res.statusCode")])
    agg = res.aggregators[0]
    assert sum(agg.table.values()) == N
    assert len(agg.table) == 4 * 7  # methods x status codes

    # sanity cap on peak RSS (generous; catches O(records) blowups)
    max_rss_kb = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
    assert max_rss_kb < 4 * 1024 * 1024, max_rss_kb


def test_index_config_example(dn, tmp_path, fixture_tree):
    """The shipped example index config drives a build
    (examples/index-muskie-local.json)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    idx = str(tmp_path / "idx")
    r = dn("datasource-add", "src", "--path=" + one,
           "--index-path=" + idx, "--time-field=time")
    assert r.code == 0, r.err
    example = os.path.join(os.path.dirname(__file__), "..",
                           "examples", "index-muskie-local.json")
    r = dn("build", "--index-config=" + example, "src")
    assert r.code == 0, r.err
    r = dn("query", "-b", "req.method", "src")
    assert r.code == 0, r.err
    assert "GET" in r.out
