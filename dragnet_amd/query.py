"""
Query model: bucketizers, field/breakdown validation, QueryConfig.

Mirrors the reference's QueryConfig semantics (reference lib/dragnet.js:28-244):
breakdown fields may carry `aggr` (quantize | lquantize+step), `date` (parse
the source field as a date into unix seconds) and `field` (source field name,
defaulting to the breakdown name); `__dn`-prefixed names are reserved;
`before`/`after` must be given together.
"""

import math

from . import jsdate
from . import krill


class QueryError(Exception):
    pass


class P2Bucketizer(object):
    """Power-of-two bucketizer.

    Ordinal 0 holds values < 1; ordinal i (i>=1) holds [2^(i-1), 2^i).
    bucket_min(0) == 0, bucket_min(i) == 2^(i-1) — matches the
    DTrace-style quantize output in the reference goldens
    (tests/dn/local/tst.scan_file.sh.out:288-315).
    """

    aggr = "quantize"

    def bucket(self, v):
        if v < 1:
            return 0
        _, e = math.frexp(float(v))  # v = m * 2^e with 0.5 <= m < 1
        return e

    def bucket_min(self, i):
        if i <= 0:
            return 0
        return 2 ** (i - 1)


class LinearBucketizer(object):
    """Fixed-step bucketizer: ordinal floor(v/step), min i*step."""

    aggr = "lquantize"

    def __init__(self, step):
        self.step = step

    def bucket(self, v):
        return int(math.floor(v / self.step))

    def bucket_min(self, i):
        return i * self.step


def parse_field(b, allow_reserved=False):
    """Validate/normalize one breakdown dict (from attrs_parse).

    Mutates and returns b; raises QueryError on invalid specs.
    (reference lib/dragnet.js:210-244)
    """
    if not isinstance(b.get("name"), str):
        raise QueryError("field has no name")
    if "aggr" in b:
        if b["aggr"] not in ("quantize", "lquantize"):
            raise QueryError('unsupported aggr: "%s"' % b["aggr"])
        if b["aggr"] == "lquantize":
            if "step" not in b:
                raise QueryError('aggr "lquantize" requires "step"')
            try:
                b["step"] = int(str(b["step"]), 10)
            except ValueError:
                raise QueryError(
                    'aggr "lquantize": invalid value for "step": "%s"'
                    % b["step"])
    if not allow_reserved and b["name"].startswith("__dn"):
        raise QueryError(
            'field names starting with "__dn" are reserved')
    if "field" not in b:
        b["field"] = b["name"]
    return b


def parse_fields(breakdowns, allow_reserved=False):
    out = []
    for i, b in enumerate(breakdowns):
        try:
            out.append(parse_field(dict(b), allow_reserved))
        except QueryError as e:
            raise QueryError(
                'field %d ("%s") is invalid: %s'
                % (i, b.get("name", "?"), e))
    return out


def parse_time_bounds(time_after, time_before):
    """Both-or-neither before/after parsing (reference lib/dragnet.js:151-186).

    Returns (after_ms, before_ms) or (None, None).
    """
    if time_after:
        if not time_before:
            raise QueryError('"after" requires specifying "before" too')
        after_ms = jsdate.parse_ms(time_after)
        if after_ms is None:
            raise QueryError(
                '"after": not a valid date: "%s"' % time_after)
        before_ms = jsdate.parse_ms(time_before)
        if before_ms is None:
            raise QueryError(
                '"before": not a valid date: "%s"' % time_before)
        if after_ms > before_ms:
            raise QueryError(
                '"after" timestamp may not come after "before"')
        return (after_ms, before_ms)
    if time_before:
        raise QueryError('"before" requires specifying "after" too')
    return (None, None)


class QueryConfig(object):
    """Immutable parameters of one query.

    Attributes (reference lib/dragnet.js:28-77):
        filter       predicate JSON or None
        breakdowns   list of normalized breakdown dicts
        before_ms / after_ms   time bounds in epoch ms (None or both set)
        bucketizers  {breakdown name: bucketizer}
        synthetic    list of {name, field, ...} date fields to materialize
        time_field   the datasource time field used for before/after
    """

    def __init__(self, filter=None, breakdowns=(), time_after=None,
                 time_before=None, time_field=None, allow_reserved=False):
        if filter is not None:
            try:
                krill.create_predicate(filter)
            except krill.KrillError as e:
                raise QueryError("invalid filter: %s" % e)
        self.filter = filter
        self.breakdowns = parse_fields(breakdowns, allow_reserved)
        self.after_ms, self.before_ms = parse_time_bounds(
            time_after, time_before)
        self.time_field = time_field

        self.fields_by_name = {}
        self.bucketizers = {}
        self.synthetic = []

        if time_field:
            self.synthetic.append(
                {"name": time_field, "field": time_field, "date": ""})

        for b in self.breakdowns:
            self.fields_by_name[b["name"]] = b
            if "date" in b:
                self.synthetic.append(b)
            if "aggr" not in b:
                continue
            if b["aggr"] == "quantize":
                self.bucketizers[b["name"]] = P2Bucketizer()
            else:
                self.bucketizers[b["name"]] = LinearBucketizer(b["step"])

    def decomps(self):
        return [b["name"] for b in self.breakdowns]

    def time_bounds_filter(self, timefield):
        """Krill filter for the query's time bounds in unix seconds
        (reference lib/dragnet-impl.js:94-125: ceil to whole seconds,
        ge/lt)."""
        if self.before_ms is None:
            return None
        return {"and": [
            {"ge": [timefield, int(math.ceil(self.after_ms / 1000.0))]},
            {"lt": [timefield, int(math.ceil(self.before_ms / 1000.0))]},
        ]}


def query_load(filter=None, breakdown_specs=None, breakdowns=None,
               time_after=None, time_before=None, time_field=None,
               allow_reserved=False):
    """Build a QueryConfig from CLI-ish inputs.

    breakdown_specs: a raw comma string parsed with attrs_parse;
    breakdowns: an already-parsed list of dicts.
    """
    from .attrs import attrs_parse, AttrsError
    if breakdown_specs is not None:
        parsed = attrs_parse(breakdown_specs)
        if isinstance(parsed, AttrsError):
            raise QueryError("invalid breakdowns: %s" % parsed)
        breakdowns = parsed
    return QueryConfig(
        filter=filter, breakdowns=breakdowns or [],
        time_after=time_after, time_before=time_before,
        time_field=time_field, allow_reserved=allow_reserved)
