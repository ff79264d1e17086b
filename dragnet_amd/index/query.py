"""
Index reader: serve a query from a SQLite index file.

Opens the index read-only, validates version ~2 (reference
lib/index-query.js:22, 82-88), selects a metric by filter-exact-match +
field coverage (findMetric, reference lib/index-query.js:154-263), then
streams `SELECT cols, SUM(value) ... WHERE <pred> GROUP BY cols` rows back
through a fresh aggregator (reference lib/index-query.js:269-405).
"""

import json
import sqlite3

from .. import krill
from ..points import Aggregator
from .sink import sqlite3_escape


class IndexError_(Exception):
    pass


class IndexQuerier(object):
    def __init__(self, filename):
        self.filename = filename
        self.db = sqlite3.connect("file:%s?mode=ro" % filename, uri=True)
        self.db.row_factory = sqlite3.Row
        self.config = {}
        self.metrics = []
        self._load_config()

    def close(self):
        self.db.close()

    def _load_config(self):
        try:
            rows = self.db.execute("SELECT * FROM dragnet_config").fetchall()
        except sqlite3.Error as e:
            raise IndexError_("reading dragnet_config: %s" % e)
        for r in rows:
            self.config[r["key"]] = r["value"]
        version = self.config.get("version")
        if version is None:
            raise IndexError_('index missing dragnet "version"')
        if not str(version).startswith("2."):
            raise IndexError_('unsupported index version: "%s"' % version)

        rows = self.db.execute(
            "SELECT * FROM dragnet_metrics ORDER BY id").fetchall()
        for r in rows:
            filt = json.loads(r["filter"]) if r["filter"] is not None \
                else None
            params = json.loads(r["params"]) if r["params"] is not None \
                else []
            self.metrics.append({
                "id": r["id"],
                "label": r["label"],
                "filter": filt,
                "filter_raw": r["filter"],
                "params": params,
            })

    def find_metric(self, query):
        """Pick the first metric that can serve this query; returns
        {datefield, table, ignore_filter, params} or raises IndexError_.
        """
        filter_raw = None
        if query.filter is not None:
            filter_raw = json.dumps(query.filter, separators=(",", ":"))

        for met in self.metrics:
            datefield = None
            if met["filter"] is not None:
                if query.filter is None:
                    continue
                # exact-match only (a metric whose filter is a superset
                # is conservatively skipped; same as the reference)
                if _norm_json(met["filter_raw"]) != filter_raw:
                    continue

            if query.before_ms is not None or query.after_ms is not None:
                dates = [p for p in met["params"] if "date" in p]
                if not dates:
                    continue
                datefield = dates[0]["name"]

            fields_needed = set()
            if query.filter is not None and met["filter"] is None:
                fields_needed.update(
                    krill.create_predicate(query.filter).fields())
            for b in query.breakdowns:
                fields_needed.add(b["name"])
            fields_have = set(p["name"] for p in met["params"])

            if fields_needed <= fields_have:
                return {
                    "datefield": datefield,
                    "table": "dragnet_index_%d" % met["id"],
                    "ignore_filter": met["filter"] is not None,
                    "params": met["params"],
                }
        raise IndexError_("no metrics available to serve query")

    def run(self, query):
        """Execute the query against this index; returns an Aggregator
        holding the per-file partial result."""
        table = self.find_metric(query)

        when = query.time_bounds_filter(table["datefield"]) \
            if table["datefield"] else None
        qfilter = None if table["ignore_filter"] else query.filter
        filt = krill.filter_and(qfilter, when)

        groupby = [sqlite3_escape(b["name"]) for b in query.breakdowns
                   if "date" not in b or b["field"] == b["name"]]
        columns = list(groupby)
        columns.append("SUM(value) as value")

        sql = "SELECT %s from %s " % (", ".join(columns), table["table"])
        if filt is not None:
            name_map = {}
            pred = krill.create_predicate(filt)
            for f in pred.fields():
                name_map[f] = sqlite3_escape(f)
            sql += "WHERE " + pred.to_sql(name_map) + " "
        if groupby:
            sql += "GROUP BY " + ", ".join(groupby)

        agg = Aggregator(query)
        for row in self.db.execute(sql):
            agg.write(self._deserialize_row(query, row))
        return agg

    def _deserialize_row(self, query, row):
        value = row["value"]
        if value is None:
            value = 0
        fields = {}
        keys = row.keys()
        for b in query.breakdowns:
            col = sqlite3_escape(b["name"])
            if col in keys:
                fields[b["name"]] = row[col]
        return {"fields": fields, "value": value}


def _norm_json(raw):
    """Normalize a stored filter JSON string for exact-match compare."""
    if raw is None:
        return None
    try:
        return json.dumps(json.loads(raw), separators=(",", ":"))
    except ValueError:
        return raw
