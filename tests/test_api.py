"""Library facade + count-conservation invariant tests."""

import os

from dragnet_amd import api, config as mod_config


def test_api_scan_build_query(fixture_tree, tmp_path, monkeypatch):
    cfgfile = str(tmp_path / "rc.json")
    monkeypatch.setenv("DRAGNET_CONFIG", cfgfile)
    monkeypatch.setenv("DRAGNET_ENGINE", "cpu")
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    cfg = mod_config.DragnetConfig()
    cfg.datasource_add(mod_config.Datasource(
        name="src", path=one, index_path=str(tmp_path / "idx"),
        time_field="time"))
    cfg.metric_add(mod_config.Metric(
        name="m", datasource="src",
        breakdowns=[{"name": "operation", "field": "operation"},
                    {"name": "req.method", "field": "req.method"}]))
    mod_config.save_config(cfg, cfgfile)

    pts = api.scan("src", breakdowns="req.method")
    assert sum(p["value"] for p in pts) == 250

    written = api.build("src")
    assert written and all(os.path.exists(w) for w in written)

    qpts = api.query("src", breakdowns="req.method")
    assert qpts == pts

    ic = api.index_config("src")
    assert ic["metrics"][0]["name"] == "m"


def test_count_conservation(fixture_tree):
    """Sigma(drops) + outputs == inputs at every pipeline stage
    (SURVEY.md §5 integrity invariant) over the full fixture tree,
    which contains every drop kind."""
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load
    files = []
    for root, _dirs, names in os.walk(fixture_tree):
        for n in sorted(names):
            files.append(os.path.join(root, n))
    files.sort()
    q = query_load(
        filter={"eq": ["req.method", "GET"]},
        breakdown_specs="ts[date,field=time,aggr=lquantize,step=3600],"
                        "operation")
    res = CpuEngine().scan(files, [q], time_field="time")
    assert api.check_conservation(res.stages) == []
    # and the parser accounted for every line
    parser = dict(res.stages)["json parser"]
    assert parser["ninputs"] == 2254
    assert parser["noutputs"] + parser["invalid json"] == 2254


def test_examples_build_and_query(dn, fixture_tree, tmp_path):
    """The committed examples/ configs are valid and drive a real
    build + query end-to-end (BASELINE config #4 uses
    examples/index-muskie-local.json)."""
    import json
    import os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exdir = os.path.join(root, "examples")
    with open(os.path.join(exdir, "index-muskie-local.json")) as f:
        excfg = json.load(f)
    assert excfg["metrics"][0]["breakdowns"]
    with open(os.path.join(exdir, "query-muskie-requests.json")) as f:
        json.load(f)

    idx = str(tmp_path / "exidx")
    r = dn("datasource-add", "exsrc", "--path=" + fixture_tree,
           "--time-field=time", "--time-format=%Y/%m-%d",
           "--index-path=" + idx)
    assert r.code == 0, r.err
    r = dn("build", "--index-config=" +
           os.path.join(exdir, "index-muskie-local.json"), "exsrc")
    assert r.code == 0, r.err
    r = dn("query", "-b", "req.method,res.statusCode", "exsrc")
    assert r.code == 0, r.err
    assert "VALUE" in r.out and len(r.out.splitlines()) > 2


def test_short_option_aliases(dn, fixture_tree):
    """Reference short aliases -A/-B/-n/-i/-f/-b (bin/dn:146-215)."""
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    assert dn("datasource-add", "al", "--path=" + fixture_tree,
              "--time-field=time",
              "--time-format=%Y/%m-%d").code == 0
    long = dn("scan", "--filter", '{"eq":["req.method","GET"]}',
              "--breakdowns", "operation",
              "--after", "2014-05-01", "--before", "2014-05-02", "al")
    short = dn("scan", "-f", '{"eq":["req.method","GET"]}',
               "-b", "operation",
               "-A", "2014-05-01", "-B", "2014-05-02", "al")
    assert long.code == 0 and short.code == 0, short.err
    assert short.out == long.out and long.out != ""
    dry = dn("scan", "-n", "al")
    assert dry.code == 0
    assert "would scan" in dry.err + dry.out


def test_api_index_config(fixture_tree, tmp_path, monkeypatch):
    """api.index_config returns the metric set the CLI's
    `dn index-config` prints."""
    monkeypatch.setenv("DRAGNET_CONFIG", str(tmp_path / "rc.json"))
    monkeypatch.setenv("DRAGNET_ENGINE", "cpu")
    from dragnet_amd import api, config as mod_config
    cfg = mod_config.DragnetConfig()
    cfg.datasource_add(mod_config.Datasource(
        name="s", backend="file", path=fixture_tree))
    cfg.metric_add(mod_config.Metric(
        name="m", datasource="s",
        breakdowns=[{"name": "operation"}]))
    mod_config.save_config(cfg, str(tmp_path / "rc.json"))
    out = api.index_config("s")
    assert out["metrics"][0]["name"] == "m"
    assert out["metrics"][0]["breakdowns"][0]["name"] == "operation"
