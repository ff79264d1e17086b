"""
CPU reference scan pipeline — the semantic oracle.

Implements exactly the record pipeline of the reference hot path
(reference lib/stream-scan.js:40-94 composition):

    bytes -> line split -> JSON parse -> [datasource filter]
          -> [user filter] -> synthetic date fields -> [time filter]
          -> aggregator

Every HIP kernel is differential-tested against this pipeline.  Per-record
failures are warn-and-drop with attributed counters, matching the reference
drop taxonomy ('invalid json', 'nfilteredout', 'nfailedeval', 'undef',
'baddate'; reference lib/krill-skinner-stream.js:29-52,
lib/stream-synthetic.js:37-85).
"""

import json

from . import jsdate
from . import krill
from .points import Aggregator


def _reject_const(x):
    raise ValueError("invalid JSON constant: %s" % x)


def parse_json_line(line):
    """JSON.parse semantics: no NaN/Infinity literals."""
    return json.loads(line, parse_constant=_reject_const)


class FilterStage(object):
    def __init__(self, name, predicate, warn=None):
        self.name = name
        self.pred = predicate
        self.warn = warn
        self.counters = {"ninputs": 0, "noutputs": 0,
                         "nfilteredout": 0, "nfailedeval": 0}

    def accept(self, fields):
        c = self.counters
        c["ninputs"] += 1
        try:
            ok = self.pred.eval(fields)
        except Exception as e:
            # vsWarn('nfailedeval') — reference
            # lib/krill-skinner-stream.js:43-45
            c["nfailedeval"] += 1
            if self.warn is not None:
                self.warn(str(e), self.name, c["ninputs"])
            return False
        if not ok:
            c["nfilteredout"] += 1
            return False
        c["noutputs"] += 1
        return True


class SyntheticStage(object):
    """Materialize synthetic date fields (reference
    lib/stream-synthetic.js:37-85): pluck source field, numbers pass
    through, strings Date.parse -> floor(ms/1000); first failure per
    record is counted ('undef' or 'baddate') and the record dropped."""

    def __init__(self, synthetic, warn=None):
        self.name = "Datetime parser"
        self.synthetic = synthetic
        self.warn = warn
        self.counters = {"ninputs": 0, "noutputs": 0,
                         "undef": 0, "baddate": 0}

    def _warn(self, msg):
        if self.warn is not None:
            self.warn(msg, self.name, self.counters["ninputs"])

    def accept(self, fields):
        c = self.counters
        c["ninputs"] += 1
        nerrors = 0
        for fc in self.synthetic:
            val = krill.pluck(fields, fc["field"])
            if val is krill.MISSING:
                if nerrors == 0:
                    c["undef"] += 1
                    # reference lib/stream-synthetic.js:50-52
                    self._warn('field "%s" is undefined' % fc["field"])
                nerrors += 1
                continue
            if isinstance(val, bool):
                # typeof bool != 'number' in JS -> Date.parse(bool) -> NaN
                if nerrors == 0:
                    c["baddate"] += 1
                    self._warn('field "%s" is not a valid date'
                               % fc["field"])
                nerrors += 1
                continue
            if isinstance(val, (int, float)):
                fields[fc["name"]] = val
                continue
            ms = jsdate.parse_ms(val)
            if ms is None:
                if nerrors == 0:
                    c["baddate"] += 1
                    # reference lib/stream-synthetic.js:70-72
                    self._warn('field "%s" is not a valid date'
                               % fc["field"])
                nerrors += 1
                continue
            fields[fc["name"]] = ms // 1000
        if nerrors:
            return False
        c["noutputs"] += 1
        return True


class ScanPipeline(object):
    """One query's scan pipeline over byte/record input.

    Arguments:
        query        QueryConfig
        ds_filter    datasource-level predicate JSON (applied first)
        time_field   datasource time field (required for before/after)
        data_format  'json' | 'json-skinner'
    """

    WARN_CAP = 1000  # per-record warning entries retained

    def __init__(self, query, ds_filter=None, time_field=None,
                 data_format="json", collect_warnings=False):
        self.query = query
        self.data_format = data_format
        self.parser_counters = {"ninputs": 0, "noutputs": 0,
                                "invalid json": 0}
        self.stages = []
        # per-record vstream-style warnings: (message, context label)
        # pairs, emitted by each stage as records drop (reference
        # bin/dn warn(): 'warn: <msg>\n    at <context.label()>')
        self.warnings = []
        self.context_file = None  # set by the engine per input file
        warn = self._warn_record if collect_warnings else None

        if ds_filter is not None:
            self.stages.append(FilterStage(
                "Datasource filter", krill.create_predicate(ds_filter),
                warn=warn))

        if query.filter is not None:
            self.stages.append(FilterStage(
                "User filter", krill.create_predicate(query.filter),
                warn=warn))

        synthetic = list(query.synthetic)
        if query.before_ms is not None or query.after_ms is not None:
            if not time_field:
                raise ValueError(
                    'datasource is missing "timefield" for "before" '
                    'and "after" constraints')
            synthetic.append(
                {"name": "dn_ts", "field": time_field, "date": ""})
        if synthetic:
            self.stages.append(SyntheticStage(synthetic, warn=warn))

        tbf = query.time_bounds_filter("dn_ts")
        if tbf is not None:
            self.stages.append(FilterStage(
                "Time filter", krill.create_predicate(tbf)))

        self.aggr = Aggregator(query)
        self._collect = collect_warnings
        self._partial = b""
        self.last_point = None  # most recent successfully parsed point

    def _warn_record(self, message, stage, n):
        """Record one vstream-style warning with its context chain."""
        if len(self.warnings) >= self.WARN_CAP:
            return
        label = "%s input %d" % (stage, n)
        if self.context_file:
            label += " (%s)" % self.context_file
        self.warnings.append((message, label))

    # ---- byte-stream input ----

    def write_bytes(self, data):
        data = self._partial + data
        lines = data.split(b"\n")
        self._partial = lines.pop()
        for line in lines:
            self.write_line(line)

    def finish(self):
        if self._partial:
            self.write_line(self._partial)
            self._partial = b""

    def write_line(self, line):
        if isinstance(line, bytes):
            try:
                line = line.decode("utf-8")
            except UnicodeDecodeError:
                self.parser_counters["ninputs"] += 1
                self.parser_counters["invalid json"] += 1
                return
        self.parser_counters["ninputs"] += 1
        try:
            obj = parse_json_line(line)
        except ValueError as e:
            self.parser_counters["invalid json"] += 1
            if self._collect:
                self._warn_record("invalid json: %s" % e,
                                  "json parser",
                                  self.parser_counters["ninputs"])
            return
        self.parser_counters["noutputs"] += 1
        if self.data_format == "json-skinner":
            if (not isinstance(obj, dict) or "fields" not in obj
                    or not isinstance(obj.get("value"), (int, float))
                    or isinstance(obj.get("value"), bool)):
                self.parser_counters["noutputs"] -= 1
                self.parser_counters["invalid json"] += 1
                return
            self.last_point = obj
            self.write_point(obj)
        else:
            self.last_point = {"fields": obj, "value": 1}
            self.write_point(self.last_point)

    # ---- record input ----

    def write_point(self, point):
        fields = point["fields"]
        for stage in self.stages:
            if not stage.accept(fields):
                return
        self.aggr.write(point)

    # ---- results ----

    def points(self):
        self.finish()
        return self.aggr.points()

    def rows(self):
        self.finish()
        return self.aggr.rows()

    def counter_stages(self):
        """[(stage name, counters dict)] in pipeline order, for
        --counters output."""
        out = [("json parser", self.parser_counters)]
        if self.data_format == "json":
            n = self.parser_counters["noutputs"]
            out.append(("SkinnerAdapterStream",
                        {"ninputs": n, "noutputs": n}))
        for s in self.stages:
            out.append((s.name, s.counters))
        out.append(("Aggregator", {
            "ninputs": self.aggr.ninputs,
            "noutputs": self.aggr.noutputs(),
            "nonnumeric": self.aggr.ndropped_nonnumeric,
        }))
        return out
