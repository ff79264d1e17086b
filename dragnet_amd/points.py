"""
Skinner-style streaming aggregation: points, the Aggregator, the Flattener.

A "point" is {'fields': {...}, 'value': n}.  Aggregation of points is
associative and commutative (value-weighted re-aggregation), which is the
load-bearing semantic that makes both the index subsystem and the multi-GPU
RCCL merge correct (reference: the x3 idempotence test
tests/dn/local/tst.format_skinner.sh:25-37 and the Manta reduce phase
lib/datasource-manta.js:212-219).

Canonical grouping values per breakdown column (derived from reference
golden outputs, e.g. tests/dn/local/tst.scan_file.sh.out:136-147 where
null/missing req.caller group as the strings "null"/"undefined"):

  * aggregated (quantize/lquantize) columns -> ordinal bucket index (int);
    emitted in points as the bucket minimum (a number)
  * date (synthetic, non-aggregated) columns -> unix seconds (int)
  * all other columns -> strings; numbers/booleans/null/missing coerce via
    JavaScript string conversion ("200", "true", "null", "undefined")
"""

from . import krill
from .krill import MISSING, pluck

_INF = float("inf")


def js_num_str(v):
    """JavaScript String(number) for the values we encounter."""
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, int):
        return str(v)
    if isinstance(v, float):
        if v != v:
            return "NaN"
        if v == float("inf"):
            return "Infinity"
        if v == float("-inf"):
            return "-Infinity"
        if v == int(v) and abs(v) < 2 ** 53:
            return str(int(v))
        return repr(v)
    return str(v)


def lookup(fields, name):
    """Field lookup for aggregation: literal key first, then dotted path.

    The literal-first order is what lets skinner-format points (whose
    fields are literally keyed "req.method") re-aggregate identically to
    raw nested records.
    """
    if isinstance(fields, dict) and name in fields:
        return fields[name]
    return pluck(fields, name)


def canonical(val, has_date=False):
    """Canonical grouping value for a NON-aggregated column."""
    if val is MISSING:
        return "undefined"
    if val is None:
        return "null"
    if isinstance(val, bool):
        return "true" if val else "false"
    if has_date:
        # synthetic date fields carry unix seconds; keep numeric.
        # Non-numeric values (possible only on the re-aggregation
        # path — scan-side synthetics drop them first) coerce like
        # any JS object key below, mirroring skinner: a raw
        # passthrough here crashed on list values (found by the
        # 300-seed codec soak).
        if isinstance(val, (int, float)):
            return int(val)
    if isinstance(val, (int, float)):
        return js_num_str(val)
    if isinstance(val, str):
        return val
    # objects/arrays as group-by values coerce like JS object keys
    if isinstance(val, dict):
        return "[object Object]"
    if isinstance(val, list):
        return js_array_str(val)
    return str(val)


def js_array_str(lst):
    """JS Array.prototype.toString: join(','), null/undefined -> ''."""
    parts = []
    for x in lst:
        if x is None:
            parts.append("")
        elif isinstance(x, list):
            parts.append(js_array_str(x))
        elif isinstance(x, dict):
            parts.append("[object Object]")
        elif isinstance(x, bool):
            parts.append("true" if x else "false")
        elif isinstance(x, (int, float)):
            parts.append(js_num_str(x))
        else:
            parts.append(str(x))
    return ",".join(parts)


class Aggregator(object):
    """Streaming group-by/count over points.

    Internal keys hold ordinal bucket indices for aggregated columns.
    Records whose aggregated column is non-numeric are dropped
    ('nonnumeric' counter; reference README.md:718-722).
    """

    def __init__(self, query):
        self.query = query
        self.breakdowns = query.breakdowns
        self.bucketizers = query.bucketizers
        self.table = {}
        self.ninputs = 0
        self.ndropped_nonnumeric = 0

    def write(self, point):
        self.ninputs += 1
        fields = point["fields"]
        key = []
        for b in self.breakdowns:
            name = b["name"]
            val = lookup(fields, name)
            bk = self.bucketizers.get(name)
            if bk is not None:
                if isinstance(val, str):
                    # numeric STRINGS coerce (JS arithmetic in the
                    # bucketizer): the reference's own golden counts
                    # {"latency": "26"} into the p2 histogram
                    # (tests/data/2014/05-05/more.log:1 vs
                    # tst.scan_fileset.sh.out bucket 16), despite
                    # README.md:718-722 claiming otherwise
                    val = krill.to_number(val)
                if isinstance(val, bool) or \
                        not isinstance(val, (int, float)) or \
                        val != val or val in (_INF, -_INF):
                    self.ndropped_nonnumeric += 1
                    return False
                key.append(bk.bucket(val))
            else:
                key.append(canonical(val, "date" in b))
        k = tuple(key)
        self.table[k] = self.table.get(k, 0) + point["value"]
        return True

    def noutputs(self):
        # A zero-breakdown aggregation always emits exactly one point,
        # even over empty input (observed tst.empty.sh.out:1-24).
        if not self.breakdowns:
            return 1
        return len(self.table)

    def points(self):
        """Emit aggregated results as points (sorted for determinism).

        Aggregated fields carry bucket minimums (numbers), matching the
        reference's resultsAsPoints output
        (tests/dn/local/tst.scan_file.sh.out:306-314).
        """
        if not self.breakdowns:
            total = sum(self.table.values()) if self.table else 0
            return [{"fields": {}, "value": total}]
        out = []
        for key in sorted(self.table.keys(), key=_sort_key):
            fields = {}
            for b, kv in zip(self.breakdowns, key):
                bk = self.bucketizers.get(b["name"])
                fields[b["name"]] = bk.bucket_min(kv) if bk else kv
            out.append({"fields": fields, "value": self.table[key]})
        return out

    def rows(self):
        """Flattened rows: tuples + value, aggregated columns as ORDINAL
        bucket indices (expanded at print time).  Zero breakdowns ->
        [total] (a bare number), mirroring the reference flattener."""
        if not self.breakdowns:
            total = sum(self.table.values()) if self.table else 0
            return [total]
        return [list(k) + [v] for k, v in self.table.items()]


def _sort_key(key_tuple):
    return tuple((0, v) if isinstance(v, (int, float)) else (1, v)
                 for v in key_tuple)


class Flattener(object):
    """Re-aggregates a point stream into one flattened `rows` payload
    (reference lib/skinner-flattener.js:20-35)."""

    def __init__(self, query):
        self.agg = Aggregator(query)

    def write(self, point):
        return self.agg.write(point)

    def rows(self):
        return self.agg.rows()


def fast_reduce_spec(query):
    """Column spec for the C++ tagged-point reducer (_points): one
    (name, kind, step) per breakdown, kinds 0=canonical 1=date
    2=quantize 3=lquantize — mirrors Aggregator.write exactly."""
    cols = []
    for b in query.breakdowns:
        name = b["name"]
        bk = query.bucketizers.get(name)
        if bk is None:
            cols.append((name, 1 if "date" in b else 0, 0.0))
        elif bk.aggr == "quantize":
            cols.append((name, 2, 0.0))
        else:
            cols.append((name, 3, float(bk.step)))
    return cols


def reduce_tagged_stream(stream, aggs, queries):
    """Drain a tagged-point NDJSON stream through the C++ reducer,
    merging fast-path groups into each Aggregator and returning the
    punted lines (anything outside the flat-scalar fast shape) for
    the caller's per-line Python path.  Raises ImportError when the
    native extension is unavailable (caller falls back wholesale)."""
    import os
    if os.environ.get("DRAGNET_PY_POINTS") == "1":
        raise ImportError("DRAGNET_PY_POINTS=1")
    from .index import _points

    specs = [fast_reduce_spec(q) for q in queries]
    punted_all = []

    def run(data):
        tables, nin, nonn, punted = _points.reduce_tagged(data, specs)
        for agg, t, n, d in zip(aggs, tables, nin, nonn):
            agg.ninputs += n
            agg.ndropped_nonnumeric += d
            tab = agg.table
            for k, v in t.items():
                tab[k] = tab.get(k, 0) + v
        punted_all.extend(punted)

    rem = b""
    while True:
        chunk = stream.read(64 << 20)
        if not chunk:
            break
        chunk = rem + chunk if rem else chunk
        cut = chunk.rfind(b"\n")
        if cut < 0:
            rem = chunk
            continue
        rem = chunk[cut + 1:]
        run(chunk[:cut + 1])
    if rem:
        run(rem)
    return punted_all
