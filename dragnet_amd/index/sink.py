"""
Index writer: aggregated points -> SQLite index file.

Logical schema matches the reference index format version 2.0.0
(reference lib/index-sink.js:116-230):

    dragnet_config   (key varchar primary key, value varchar)
                     incl. version='2.0.0' and optional dn_start
    dragnet_metrics  (id, label, filter, params)  filter/params as JSON
    dragnet_index_<i> one table per metric: one column per breakdown
                     (integer if aggregated else varchar(128)) + value

Crash safety: writes to <file>.<pid>, atomic rename on flush
(reference lib/index-sink.js:64, 288-297).  synchronous=off; the caller
owns durability.
"""

import json
import os
import sqlite3

INDEX_VERSION = "2.0.0"


def sqlite3_escape(name):
    """Column-name escaping: [.-] -> _ (reference lib/index-sink.js:232)."""
    return name.replace(".", "_").replace("-", "_")


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
        self.db = sqlite3.connect(self.tmpfilename)
        self.db.execute("pragma synchronous = off;")
        self._init_db()

    def _init_db(self):
        db = self.db
        db.execute("CREATE TABLE dragnet_config ("
                   "key varchar(128) primary key, value varchar(128))")
        db.execute("CREATE TABLE dragnet_metrics ("
                   "id integer, label varchar(64), filter varchar(1024), "
                   "params varchar(1024))")

        pairs = [("version", INDEX_VERSION)]
        for k, v in self.config.items():
            assert k != "version"
            pairs.append((k, v))
        db.executemany("INSERT INTO dragnet_config VALUES (?, ?)", pairs)

        self._inserts = []
        for i, m in enumerate(self.metrics):
            db.execute("INSERT INTO dragnet_metrics VALUES (?, ?, ?, ?)", (
                i, m["name"],
                json.dumps(m.get("filter"), separators=(",", ":")),
                json.dumps(m.get("breakdowns", []), separators=(",", ":")),
            ))
            tbl = "dragnet_index_%d" % i
            cols = []
            for b in m.get("breakdowns", []):
                ctype = "integer" if "aggr" in b else "varchar(128)"
                cols.append("%s %s" % (sqlite3_escape(b["name"]), ctype))
            cols.append("value integer")
            db.execute("CREATE TABLE %s (%s)" % (tbl, ", ".join(cols)))
            nvals = len(m.get("breakdowns", [])) + 1
            self._inserts.append(
                "INSERT INTO %s VALUES (%s)"
                % (tbl, ", ".join("?" * nvals)))

    def write_point(self, point):
        """Write one aggregated point.  fields must carry __dn_metric
        (the metric index) and a value per breakdown of that metric
        (reference lib/index-sink.js:240-261)."""
        fields = point["fields"]
        mi = fields["__dn_metric"]
        m = self.metrics[mi]
        row = []
        for b in m.get("breakdowns", []):
            row.append(fields[b["name"]])
        row.append(point["value"])
        self.db.execute(self._inserts[mi], row)
        self.nwritten += 1

    def flush(self):
        """Commit, close, atomic rename into place."""
        self.db.commit()
        self.db.close()
        os.replace(self.tmpfilename, self.filename)

    def abort(self):
        try:
            self.db.close()
        finally:
            if os.path.exists(self.tmpfilename):
                os.unlink(self.tmpfilename)
