"""
Index writer: aggregated points -> SQLite index file.

Logical schema matches the reference index format version 2.0.0
(reference lib/index-sink.js:116-230):

    dragnet_config   (key varchar primary key, value varchar)
                     incl. version='2.0.0' and optional dn_start
    dragnet_metrics  (id, label, filter, params)  filter/params as JSON
    dragnet_index_<i> one table per metric: one column per breakdown
                     (integer if aggregated else varchar(128)) + value

The hot insert path runs through the native C++ sink (_csink: SQLite C
API, prepared statements, one transaction, columnar row batches — the
equivalent of the reference's native sqlite3 binding); set
DRAGNET_PY_SINK=1 to force the pure-Python sqlite3 path (differential
tests).

Crash safety: writes to <file>.<pid>, atomic rename on flush
(reference lib/index-sink.js:64, 288-297).  synchronous=off; the caller
owns durability.
"""

import json
import os

INDEX_VERSION = "2.0.0"


class SinkError(Exception):
    """Index materialization failed (e.g. a breakdown name that
    collides after [.-]->_ escaping, or an SQL-keyword column — the
    reference's unquoted CREATE TABLE fails identically; we surface
    it as a clean `dn:` error instead of a traceback)."""


def sqlite3_escape(name):
    """Column-name escaping: [.-] -> _ (reference lib/index-sink.js:232)."""
    return name.replace(".", "_").replace("-", "_")


def _load_csink():
    if os.environ.get("DRAGNET_PY_SINK") == "1":
        return None
    try:
        from . import _csink
        return _csink
    except ImportError:
        return None


class IndexSink(object):
    def __init__(self, filename, metrics, config=None):
        """metrics: list of metric dicts {name, filter, breakdowns}
        (breakdowns are normalized dicts with name/field/aggr/step/date).
        """
        self.filename = filename
        self.tmpfilename = filename + "." + str(os.getpid())
        self.metrics = metrics
        self.config = dict(config or {})
        self.nwritten = 0

        d = os.path.dirname(self.tmpfilename)
        if d:
            os.makedirs(d, exist_ok=True)
        if os.path.exists(self.tmpfilename):
            os.unlink(self.tmpfilename)

        self._cs = _load_csink()
        if self._cs is not None:
            self.db = self._cs.CSink(self.tmpfilename)
        else:
            import sqlite3
            self.db = sqlite3.connect(self.tmpfilename)
            self.db.execute("pragma synchronous = off;")
        # buffered columnar per metric: one list per breakdown + values
        self._cols = [[[] for _ in m.get("breakdowns", [])]
                      for m in metrics]
        self._vals = [[] for _ in metrics]
        self._init_db()

    # ---- schema ----

    def _schema_sql(self):
        """(create statements, per-metric insert statements)."""
        creates = [
            "CREATE TABLE dragnet_config ("
            "key varchar(128) primary key, value varchar(128))",
            "CREATE TABLE dragnet_metrics ("
            "id integer, label varchar(64), filter varchar(1024), "
            "params varchar(1024))",
        ]
        inserts = []
        for i, m in enumerate(self.metrics):
            tbl = "dragnet_index_%d" % i
            cols = []
            for b in m.get("breakdowns", []):
                ctype = "integer" if "aggr" in b else "varchar(128)"
                cols.append("%s %s" % (sqlite3_escape(b["name"]), ctype))
            cols.append("value integer")
            creates.append(
                "CREATE TABLE %s (%s)" % (tbl, ", ".join(cols)))
            nvals = len(m.get("breakdowns", [])) + 1
            inserts.append("INSERT INTO %s VALUES (%s)"
                           % (tbl, ", ".join("?" * nvals)))
        return creates, inserts

    def _config_rows(self):
        pairs = [("version", INDEX_VERSION)]
        for k, v in self.config.items():
            assert k != "version"
            pairs.append((k, v))
        return pairs

    def _metric_rows(self):
        return [
            (i, m["name"],
             json.dumps(m.get("filter"), separators=(",", ":")),
             json.dumps(m.get("breakdowns", []), separators=(",", ":")))
            for i, m in enumerate(self.metrics)]

    def _init_db(self):
        try:
            self._init_db_inner()
        except Exception as e:
            msg = str(e)
            try:
                self.abort()
            except Exception:
                pass
            if "sqlite" in msg or "column" in msg or "syntax" in msg \
                    or type(e).__module__ == "sqlite3":
                raise SinkError("cannot materialize index: %s" % msg)
            raise

    def _init_db_inner(self):
        creates, inserts = self._schema_sql()
        if self._cs is not None:
            for sql in creates:
                self.db.exec(sql)
            self.db.prepare_inserts(
                ["INSERT INTO dragnet_config VALUES (?, ?)",
                 "INSERT INTO dragnet_metrics VALUES (?, ?, ?, ?)"]
                + inserts)
            for row in self._config_rows():
                self.db.insert_row(0, row)
            for row in self._metric_rows():
                self.db.insert_row(1, row)
        else:
            for sql in creates:
                self.db.execute(sql)
            self.db.executemany("INSERT INTO dragnet_config VALUES "
                                "(?, ?)", self._config_rows())
            self.db.executemany("INSERT INTO dragnet_metrics VALUES "
                                "(?, ?, ?, ?)", self._metric_rows())
            self._inserts = inserts

    # ---- rows ----

    def write_point(self, point):
        """Write one aggregated point.  fields must carry __dn_metric
        (the metric index) and a value per breakdown of that metric
        (reference lib/index-sink.js:240-261)."""
        fields = point["fields"]
        mi = fields["__dn_metric"]
        cols = self._cols[mi]
        for j, b in enumerate(self.metrics[mi].get("breakdowns", [])):
            cols[j].append(fields[b["name"]])
        self._vals[mi].append(point["value"])
        self.nwritten += 1

    def _columns(self, mi):
        """Columnar batch for the native insert: int64 arrays for
        aggregated columns, str lists for varchar columns (non-str
        values are coerced in C++ the way TEXT affinity would)."""
        import numpy as np
        bds = self.metrics[mi].get("breakdowns", [])
        cols = []
        for j, b in enumerate(bds):
            if "aggr" in b:
                cols.append(np.array(self._cols[mi][j],
                                     dtype=np.int64))
            else:
                cols.append(self._cols[mi][j])
        values = np.array(self._vals[mi], dtype=np.float64)
        return cols, values

    def _iter_rows(self, mi):
        for row in zip(*(self._cols[mi] + [self._vals[mi]])):
            yield row

    def flush(self):
        """Insert buffered rows, commit, close, atomic rename."""
        if self._cs is not None:
            for mi in range(len(self.metrics)):
                if not self._vals[mi]:
                    continue
                try:
                    cols, values = self._columns(mi)
                    self.db.insert_columnar(2 + mi, cols, values)
                except (OverflowError, ValueError, TypeError):
                    # exotic scalar in an integer column: generic path
                    for row in self._iter_rows(mi):
                        self.db.insert_row(2 + mi, row)
            self.db.commit()
            self.db.close()
        else:
            for mi in range(len(self.metrics)):
                if self._vals[mi]:
                    self.db.executemany(self._inserts[mi],
                                        self._iter_rows(mi))
            self.db.commit()
            self.db.close()
        self._cols = [[[] for _ in m.get("breakdowns", [])]
                      for m in self.metrics]
        self._vals = [[] for _ in self.metrics]
        os.replace(self.tmpfilename, self.filename)

    def abort(self):
        try:
            self.db.close()
        finally:
            if os.path.exists(self.tmpfilename):
                os.unlink(self.tmpfilename)
