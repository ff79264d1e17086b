"""Diagnose the empty result for a date-breakdown columnar K7 query."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    import tempfile

    from dragnet_amd.config import Datasource
    from dragnet_amd.datasource.file import FileDatasource
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.engine.gpu import GpuEngine
    from dragnet_amd.index import IndexQuerier
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import make_fixture_tree

    root = tempfile.mkdtemp()
    make_fixture_tree(root)
    idx = tempfile.mkdtemp()
    ds = Datasource(name="t", backend="file", path=root,
                    index_path=idx, time_field="time",
                    time_format="%Y/%m-%d")
    gpu = GpuEngine()
    fd = FileDatasource(ds, engine=gpu)
    metrics = [{"name": "m", "filter": None, "breakdowns": [
        {"name": "ts", "field": "time", "date": "",
         "aggr": "lquantize", "step": 3600},
        {"name": "host", "field": "host"},
        {"name": "req.method", "field": "req.method"},
        {"name": "latency", "field": "latency", "aggr": "quantize"},
    ]}]
    fd.build(metrics, interval="day")

    q = query_load(
        breakdowns=[{"name": "ts", "field": "time", "date": "",
                     "aggr": "lquantize", "step": 3600},
                    {"name": "host", "field": "host"}],
        time_after="2014-05-02", time_before="2014-05-03",
        allow_reserved=True)

    from dragnet_amd import krill
    from dragnet_amd.datasource.file import _ms_to_iso
    from dragnet_amd.query import QueryConfig
    eff = QueryConfig(
        filter=None,
        breakdowns=[dict(b) for b in q.breakdowns],
        time_after=_ms_to_iso(q.after_ms),
        time_before=_ms_to_iso(q.before_ms),
        allow_reserved=True)

    files = list(fd._find_index_files(q, "day"))
    print("index files:", [p for p, _ in files])
    for path, _st in files:
        iq = IndexQuerier(path)
        table = iq.find_metric(eff)
        print("== file", path)
        print("  table:", table["table"], "datefield:",
              table["datefield"])
        when = eff.time_bounds_filter(table["datefield"])
        filt = krill.filter_and(None, when)
        print("  filt:", filt)
        sql_part = iq.run(eff)
        print("  sqlite partial:", len(sql_part.table), "groups,",
              sum(sql_part.table.values()), "total")
        from dragnet_amd.index import _csink
        from dragnet_amd.index.sink import sqlite3_escape
        params = table["params"]
        kinds = "".join("n" if ("aggr" in p or "date" in p) else "s"
                        for p in params)
        sql = "SELECT %s, value from %s" % (
            ", ".join(sqlite3_escape(p["name"]) for p in params),
            table["table"])
        cols, vals = _csink.read_columns(iq.filename, sql, kinds)
        print("  kinds:", kinds, "nrows:", len(vals))
        if len(vals):
            for ci, p in enumerate(params):
                c = cols[ci]
                if kinds[ci] == "n":
                    print("   col %-12s num head:" % p["name"],
                          c[:3])
                else:
                    print("   col %-12s str head:" % p["name"],
                          c[0][:40])
        gp = gpu.columnar_query(eff, filt, params, kinds, cols, vals)
        print("  gpu partial:", len(gp.table), "groups,",
              sum(gp.table.values()) if gp.table else 0, "total,",
              "ninputs", gp.ninputs)
        if gp.table != sql_part.table:
            sk = set(sql_part.table) - set(gp.table)
            gk = set(gp.table) - set(sql_part.table)
            print("   only-sql keys:", list(sk)[:4])
            print("   only-gpu keys:", list(gk)[:4])
        iq.close()


if __name__ == "__main__":
    main()
