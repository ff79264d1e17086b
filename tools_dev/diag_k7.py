"""A/B the K7 index-query paths: SQLite (CPU) vs GPU columnar, at
several table sizes — sets DRAGNET_INDEX_GPU_ROWS."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def build_index(path, nrows):
    from dragnet_amd.index.sink import IndexSink
    metrics = [{"name": "m", "filter": None,
                "breakdowns": [
                    {"name": "__dn_ts", "field": "time", "date": "",
                     "aggr": "lquantize", "step": 86400},
                    {"name": "host", "field": "host"},
                    {"name": "req.method", "field": "req.method"},
                    {"name": "latency", "field": "latency",
                     "aggr": "quantize"}]}]
    s = IndexSink(path, metrics)
    t0 = 1399000000 // 86400
    for i in range(nrows):
        s.write_point({"fields": {
            "__dn_metric": 0,
            "__dn_ts": (t0 + (i % 30)) * 86400,
            "host": "host%03d" % (i % 300),
            "req.method": ("GET", "PUT", "DELETE", "HEAD")[i % 4],
            "latency": 1 << (i % 18),
        }, "value": 1 + (i % 5)})
    s.flush()
    return metrics


def main():
    import tempfile

    from dragnet_amd.config import Datasource
    from dragnet_amd.datasource.file import FileDatasource
    from dragnet_amd.engine.gpu import GpuEngine
    from dragnet_amd.index import IndexQuerier
    from dragnet_amd.query import query_load

    eng = GpuEngine()
    for nrows in (50_000, 500_000, 2_000_000):
        d = tempfile.mkdtemp()
        p = os.path.join(d, "idx.sqlite")
        t0 = time.time()
        build_index(p, nrows)
        t_build = time.time() - t0
        ds = Datasource(name="t", backend="file", path=d,
                        index_path=d, time_field="time")
        fd = FileDatasource(ds, engine=eng)
        q = query_load(
            filter={"eq": ["req.method", "GET"]},
            breakdown_specs="host,latency[aggr=quantize]")

        results = {}
        for mode in ("0", "1"):
            iq = IndexQuerier(p)
            # warm
            r = fd._index_query(iq, q, mode)
            t0 = time.time()
            reps = 3
            for _ in range(reps):
                r = fd._index_query(iq, q, mode)
            dt = (time.time() - t0) / reps
            results[mode] = (dt, sorted(
                (tuple(pp["fields"].items()), pp["value"])
                for pp in r.points()))
            iq.close()
        same = results["0"][1] == results["1"][1]
        print("rows=%8d build=%5.2fs sqlite=%7.1fms gpu=%7.1fms "
              "speedup=%.2fx identical=%s"
              % (nrows, t_build, results["0"][0] * 1e3,
                 results["1"][0] * 1e3,
                 results["0"][0] / results["1"][0], same))
        assert same, "GPU columnar != SQLite!"


if __name__ == "__main__":
    main()
