"""
Local file datasource: scan / build / query / index-scan / index-read.

Orchestration mirrors the reference file backend
(reference lib/datasource-file.js): enumerate files (time-format pruned),
run the scan engine over their bytes, fan one parse pass into one
aggregation per metric for index builds, route aggregated points into
per-interval SQLite index files, and answer queries by merging per-file
index partials through a final aggregator.

The scan engine is pluggable: the CPU oracle (engine/cpu.py) or the HIP
GPU engine (engine/gpu.py).
"""

import math
import os
import sys

from .. import jsdate
from .. import krill
from ..fsfind import FindCounters, find_data_files
from ..index import IndexQuerier, IndexSink
from ..index.query import IndexError_
from ..points import Aggregator
from ..query import QueryConfig

# interval -> (iso prefix length, suffix completing a full timestamp,
#              subdir, strftime file pattern);
# reference lib/datasource-file.js:466-476, lib/dragnet-impl.js:194-236
INTERVALS = {
    "hour": (len("2014-07-02T00"), ":00:00Z", "by_hour",
             "%Y-%m-%d-%H.sqlite", 3600),
    "day": (len("2014-07-02"), "T00:00:00Z", "by_day",
            "%Y-%m-%d.sqlite", 86400),
}


def metric_query(metric, interval, time_field, after_ms=None,
                 before_ms=None):
    """Build the QueryConfig for one metric's index aggregation
    (reference lib/dragnet-impl.js:290-323): for chunked intervals,
    prepend a reserved __dn_ts lquantize breakdown over the time field.
    """
    breakdowns = [dict(b) for b in metric.get("breakdowns", [])]
    if interval != "all":
        step = INTERVALS[interval][4]
        breakdowns.insert(0, {
            "name": "__dn_ts", "aggr": "lquantize", "step": step,
            "field": time_field, "date": "",
        })
    return QueryConfig(
        filter=metric.get("filter"),
        breakdowns=breakdowns,
        time_after=_ms_to_iso(after_ms),
        time_before=_ms_to_iso(before_ms),
        allow_reserved=True)


def _ms_to_iso(ms):
    return None if ms is None else jsdate.to_iso(ms / 1000.0)


class ScanResult(object):
    """A scan's output: per-query aggregators + counter stages.

    nonroot marks an empty result on a non-rank-0 process of a
    distributed scan (the CLI suppresses output for these)."""

    def __init__(self, aggregators, stages, files=None, nonroot=False):
        self.aggregators = aggregators
        self.stages = stages
        self.files = files or []
        self.nonroot = nonroot


class FileDatasource(object):
    def __init__(self, ds, engine=None):
        self.ds = ds
        self._engine = engine

    def close(self):
        pass

    def engine(self):
        if self._engine is None:
            from ..engine import get_engine
            self._engine = get_engine()
        return self._engine

    # ---- raw data scanning ----

    def _find_data(self, query, counters=None, warn=None):
        if (query.before_ms is not None
                and self.ds.time_field is None):
            raise ValueError(
                'datasource is missing "timefield" for "before" and '
                '"after" constraints')
        if (query.before_ms is not None
                and self.ds.time_format is None):
            sys.stderr.write(
                'warn: datasource is missing "timeformat" for '
                '"before" and "after" constraints\n')
        if warn is None:
            # per-file stat/read errors are warn-and-skip, never fatal
            # (reference lib/fs-find.js:274-279)
            def warn(path, msg):
                sys.stderr.write("warn: %s: %s\n" % (path, msg))
        return find_data_files(
            self.ds.path, self.ds.time_format,
            query.after_ms, query.before_ms,
            counters=counters, warn=warn)

    def scan(self, query, dry_run=False, out=None):
        """Run one query over the raw data.  Returns a ScanResult with a
        single aggregator (or prints the file list for dry runs)."""
        from ..config import VALID_FORMATS
        if self.ds.data_format not in VALID_FORMATS:
            # validated at use time, like the reference's parserFor
            # (lib/dragnet-impl.js:131-140; tst.badargs.sh)
            raise ValueError(
                'unsupported format: "%s"' % self.ds.data_format)
        counters = FindCounters()
        files = list(self._find_data(query, counters=counters))
        if dry_run:
            out = out or sys.stderr
            out.write("would scan files:\n")
            for path, _ in files:
                out.write("    %s\n" % path)
            return None
        result = self.engine().scan(
            files=[p for p, _ in files],
            queries=[query],
            ds_filter=self.ds.filter,
            time_field=self.ds.time_field,
            data_format=self.ds.data_format)
        result.stages = counters.stages() + result.stages
        result.files = [p for p, _ in files]
        return result

    def scan_multi(self, queries, after_ms, before_ms, dry_run=False,
                   out=None):
        """One parse pass, N metric aggregations (index builds;
        reference lib/datasource-file.js:386-432)."""
        pseudo = QueryConfig(time_after=_ms_to_iso(after_ms),
                             time_before=_ms_to_iso(before_ms))
        counters = FindCounters()
        files = list(self._find_data(pseudo, counters=counters))
        if dry_run:
            out = out or sys.stderr
            out.write("would scan files:\n")
            for path, _ in files:
                out.write("    %s\n" % path)
            return None
        result = self.engine().scan(
            files=[p for p, _ in files],
            queries=queries,
            ds_filter=self.ds.filter,
            time_field=self.ds.time_field,
            data_format=self.ds.data_format)
        result.stages = counters.stages() + result.stages
        result.files = [p for p, _ in files]
        return result

    # ---- index building ----

    def metric_queries(self, metrics, interval, after_ms, before_ms):
        if interval not in ("all", "hour", "day"):
            raise ValueError('unsupported interval: "%s"' % interval)
        if interval != "all" and self.ds.time_field is None:
            raise ValueError(
                'datasource is missing "timefield" needed for '
                'interval "%s"' % interval)
        return [metric_query(m, interval, self.ds.time_field,
                             after_ms, before_ms)
                for m in metrics]

    def index_scan_points(self, metrics, interval, after_ms=None,
                          before_ms=None, dry_run=False):
        """The distributed-build map phase: scan raw data, emit points
        tagged with __dn_metric (and __dn_ts for chunked intervals).
        Returns (points, stages) or None for dry runs."""
        queries = self.metric_queries(metrics, interval, after_ms,
                                      before_ms)
        result = self.scan_multi(queries, after_ms, before_ms,
                                 dry_run=dry_run)
        if result is None:
            return None
        points = []
        for qi, agg in enumerate(result.aggregators):
            for p in agg.points():
                p["fields"]["__dn_metric"] = qi
                points.append(p)
        return points, result.stages

    def build(self, metrics, interval="day", after_ms=None,
              before_ms=None, dry_run=False):
        """Materialize indexes (reference lib/datasource-file.js:307-432).
        Returns the list of index files written."""
        if not self.ds.index_path:
            raise ValueError(
                'datasource is missing "indexPath" for index operations')
        rv = self.index_scan_points(metrics, interval, after_ms,
                                    before_ms, dry_run=dry_run)
        if rv is None:
            return None
        points, _stages = rv
        return write_index(self.ds.index_path, metrics, interval, points)

    # ---- index querying ----

    def _find_index_files(self, query, interval, counters=None):
        if interval == "all":
            root = os.path.join(self.ds.index_path, "all")
            return find_files([root], counters=counters)
        _, _, subdir, pattern, _ = INTERVALS[interval]
        root = os.path.join(self.ds.index_path, subdir)
        return find_data_files(root, pattern, query.after_ms,
                               query.before_ms, counters=counters)

    def query(self, query, interval="day", dry_run=False, out=None):
        """Answer a query from the index tree: per-file partials merged
        through a final aggregator (reference
        lib/datasource-file.js:573-691)."""
        if not self.ds.index_path:
            raise ValueError(
                'datasource is missing "indexPath" for index operations')
        fcounters = FindCounters()
        files = list(self._find_index_files(query, interval,
                                            counters=fcounters))
        if dry_run:
            out = out or sys.stderr
            out.write("would scan files:\n")
            for path, _ in files:
                out.write("    %s\n" % path)
            return None

        # The datasource filter is NOT applied at query time: it was
        # already applied when the index was built (observed in
        # tst.index_file.sh: `dn query` on a GET-filtered datasource
        # returns the GET subset with no filter in the query).
        eff = QueryConfig(
            filter=query.filter,
            breakdowns=[dict(b) for b in query.breakdowns],
            time_after=_ms_to_iso(query.after_ms),
            time_before=_ms_to_iso(query.before_ms),
            allow_reserved=True)

        final = Aggregator(query)
        nerrors = []
        # K7 policy: "1" forces the GPU columnar path, "0" forces
        # SQLite; default "auto" uses the GPU only on a GPU engine and
        # only above a measured row-count threshold (small tables are
        # dominated by per-file setup; see profiles/r02_k7.md)
        mode = os.environ.get("DRAGNET_INDEX_GPU", "auto")
        for path, _st in files:
            try:
                iq = IndexQuerier(path)
            except IndexError_ as e:
                nerrors.append((path, str(e)))
                continue
            try:
                partial = self._index_query(iq, eff, mode)
                for p in partial.points():
                    final.write(p)
            except IndexError_ as e:
                nerrors.append((path, str(e)))
            finally:
                iq.close()
        # Counter stages mirror the reference's query pipeline
        # (lib/datasource-file.js:608-641): the find pipeline over the
        # index tree, then the "Index List" passthrough and the result
        # aggregator — BOTH carry the partial result rows streamed out
        # of the per-file index queriers (pinned across four goldens:
        # 2 rows/1 file, 24 rows/24 files, 120 rows/120 files, 1
        # row/empty index).
        stages = fcounters.stages() + [
            ("Index List", {"ninputs": final.ninputs,
                            "noutputs": final.ninputs}),
            ("Index Result Aggregator",
             {"ninputs": final.ninputs,
              "noutputs": final.noutputs()}),
        ]
        result = ScanResult([final], stages, [p for p, _ in files])
        result.errors = nerrors
        return result

    def _index_query(self, iq, query, mode="auto"):
        """K7: per-file index query, GPU-columnar when it pays.

        The GPU path reads the metric table's typed
        columns through the native SQLite reader (_csink.read_columns)
        and evaluates the query on-device over the raw columns
        (columnar_query_kernel) — no NDJSON serialization or re-parse.
        Reference semantics: lib/index-query.js:303-338 (SELECT ...
        WHERE pred GROUP BY + re-aggregation).
        """
        if mode == "0" or self.engine().name != "gpu":
            return iq.run(query)
        table = iq.find_metric(query)
        if mode != "1":
            n = iq.db.execute(
                "SELECT COUNT(*) FROM %s" % table["table"]
            ).fetchone()[0]
            if n < _env_rows_threshold():
                return iq.run(query)

        from ..index import _csink
        from ..index.sink import sqlite3_escape
        when = query.time_bounds_filter(table["datefield"]) \
            if table["datefield"] else None
        qfilter = None if table["ignore_filter"] else query.filter
        filt = krill.filter_and(qfilter, when)

        params = table["params"]
        kinds = "".join(
            "n" if ("aggr" in p or "date" in p) else "s"
            for p in params)
        sql = "SELECT %s from %s" % (
            ", ".join([sqlite3_escape(p["name"]) for p in params]
                      + ["value"]), table["table"])
        cols, vals = _csink.read_columns(iq.filename, sql, kinds)
        # stored strings are DECODED text; the kernel's string compare
        # unescapes record spans, so literal backslashes would be
        # misread — rare corner, served by SQLite instead
        for k, c in zip(kinds, cols):
            if k == "s" and b"\\" in c[0]:
                return iq.run(query)
        # two breakdowns reading the same source column can't share a
        # kernel-plan name — rare; served by SQLite
        srcs = [b.get("field", b["name"]) for b in query.breakdowns]
        if len(set(srcs)) != len(srcs):
            return iq.run(query)
        return self.engine().columnar_query(query, filt, params,
                                            kinds, cols, vals)

    def index_read_points(self, metrics, interval="day"):
        """Emit every stored row of every index file as tagged points
        (the distributed-build reduce input; also `dn index-read`)."""
        pseudo = QueryConfig()
        files = list(self._find_index_files(pseudo, interval))
        for path, _st in files:
            iq = IndexQuerier(path)
            try:
                for met in iq.metrics:
                    # tag with the STORED id (== write-time position in
                    # the metrics list), never the enumeration index:
                    # the two must agree with the table name even if
                    # SQLite returns dragnet_metrics rows out of
                    # insertion order
                    tbl = "dragnet_index_%d" % met["id"]
                    cols = [b["name"] for b in met["params"]]
                    sql = "SELECT %s from %s" % (
                        ", ".join(["%s" % _esc(c) for c in cols]
                                  + ["value"]), tbl)
                    for row in iq.db.execute(sql):
                        fields = {}
                        for c, v in zip(cols, row[:-1]):
                            fields[c] = v
                        fields["__dn_metric"] = met["id"]
                        yield {"fields": fields, "value": row[-1]}
            finally:
                iq.close()


def _esc(c):
    from ..index.sink import sqlite3_escape
    return sqlite3_escape(c)


def _env_rows_threshold():
    """Row count above which the GPU columnar index query would pay.

    Measured (profiles/r02_k7.md): extracting columns out of the
    SQLite B-tree alone costs ~2.5x what SQLite's own WHERE+GROUP BY
    takes end-to-end, so the GPU path loses at EVERY size while
    indexes are SQLite files — auto mode therefore never selects it
    (threshold = infinity) and DRAGNET_INDEX_GPU=1 forces it."""
    v = os.environ.get("DRAGNET_INDEX_GPU_ROWS")
    if v is None:
        return float("inf")
    try:
        return int(v)
    except ValueError:
        return float("inf")


def find_files(roots, counters=None):
    from ..fsfind import find_files as ff
    return ff(roots, counters=counters)


def write_index(index_path, metrics, interval, points, partition=None):
    """Route aggregated points into per-interval IndexSinks; atomic
    rename on flush.  Returns ALL index file paths of the tree.

    partition=(rank, world): distributed builds — every rank holds the
    full merged point set (allreduce merge), so the interval buckets
    are round-robin assigned over the SORTED bucket list and each rank
    materializes only the files it owns (disjoint parallel writes,
    reference reduce semantics lib/datasource-manta.js:334-350 spread
    across ranks).  The returned list still names the whole tree.
    """
    from ..log import get_logger
    _log = get_logger().child("datasource-file")
    _log.info("writing index", index_path=index_path,
              interval=interval, nmetrics=len(metrics))
    if interval == "all":
        if partition is not None and partition[0] != 0:
            return [os.path.join(index_path, "all")]
        sink = IndexSink(os.path.join(index_path, "all"), metrics)
        for p in points:
            sink.write_point(p)
        sink.flush()
        return [sink.filename]

    prefixlen, suffix, subdir, _pat, _step = INTERVALS[interval]
    root = os.path.join(index_path, subdir)
    buckets = {}
    for p in points:
        dnts = p["fields"]["__dn_ts"]
        assert isinstance(dnts, (int, float)) and not math.isnan(dnts)
        datestr = jsdate.to_iso(dnts)
        buckets.setdefault(datestr[:prefixlen], []).append(p)
    written = []
    for i, name in enumerate(sorted(buckets)):
        label = name.replace("T", "-")
        filename = os.path.join(root, label + ".sqlite")
        written.append(filename)
        if partition is not None and i % partition[1] != partition[0]:
            continue
        start = jsdate.parse_ms(name + suffix) // 1000
        sink = IndexSink(filename, metrics,
                         config={"dn_start": start})
        for p in buckets[name]:
            sink.write_point(p)
        sink.flush()
    return written
