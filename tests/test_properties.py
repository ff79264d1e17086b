"""Property-based invariants of the CPU oracle (hypothesis): for ANY
byte input and any well-formed query, every pipeline stage conserves
records (ninputs == noutputs + attributed drops), stage chaining is
exact, and the aggregate total equals the accepted record count.
These are the same invariants the premature-exit guard enforces at
runtime and the GPU engine is differential-tested against."""

import json

from hypothesis import given, settings
from hypothesis import strategies as st

from dragnet_amd.query import query_load
from dragnet_amd.scan_cpu import ScanPipeline

scalars = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-10**12, max_value=10**12),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=12))

values = st.recursive(
    scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=3),
        st.dictionaries(st.text(max_size=6), children, max_size=4)),
    max_leaves=8)

records = st.dictionaries(
    st.sampled_from(["a", "b", "req", "time", "lat", "x.y"]),
    values, max_size=5)

lines = st.lists(st.one_of(
    records.map(lambda r: json.dumps(r).encode()),
    st.binary(max_size=30).map(lambda b: b.replace(b"\n", b"x")),
    st.just(b""),
), max_size=40)

queries = st.sampled_from([
    {},
    {"breakdown_specs": "a"},
    {"breakdown_specs": "req.m,b"},
    {"breakdown_specs": "lat[aggr=quantize]"},
    {"filter": {"eq": ["a", 1]}},
    {"filter": {"and": [{"gt": ["lat", 0]}, {"ne": ["b", "x"]}]},
     "breakdown_specs": "b"},
    {"breakdown_specs": "t[date,field=time,aggr=lquantize,step=60]"},
])

DROPS = ("invalid json", "nfilteredout", "nfailedeval", "undef",
         "baddate", "nonnumeric")


@settings(max_examples=120, deadline=None)
@given(lines=lines, qkw=queries)
def test_pipeline_conservation(lines, qkw):
    q = query_load(**qkw)
    p = ScanPipeline(q)
    p.write_bytes(b"".join(ln + b"\n" for ln in lines))
    p.finish()
    stages = p.counter_stages()

    prev_out = None
    for name, c in stages:
        if name == "Aggregator":
            continue
        drops = sum(c.get(k, 0) for k in DROPS)
        assert c["ninputs"] == c["noutputs"] + drops, (name, c)
        if prev_out is not None:
            assert c["ninputs"] == prev_out, (name, c)
        prev_out = c["noutputs"]

    agg = dict(stages)["Aggregator"]
    assert agg["ninputs"] == prev_out
    accepted = agg["ninputs"] - agg.get("nonnumeric", 0)
    total = sum(p.aggr.table.values()) if p.aggr.table else 0
    assert total == accepted


@settings(max_examples=60, deadline=None)
@given(lines=lines)
def test_point_reaggregation_idempotent(lines):
    """points() -> re-aggregate x3 triples values exactly (the
    load-bearing associativity the distributed merge rests on)."""
    q = query_load(breakdown_specs="a,b")
    p = ScanPipeline(q)
    p.write_bytes(b"".join(ln + b"\n" for ln in lines))
    p.finish()
    pts = p.aggr.points()

    from dragnet_amd.points import Aggregator
    agg3 = Aggregator(q)
    for _ in range(3):
        for pt in pts:
            agg3.write({"fields": dict(pt["fields"]),
                        "value": pt["value"]})
    assert agg3.points() == [
        {"fields": dict(pt["fields"]), "value": 3 * pt["value"]}
        for pt in pts]


@settings(max_examples=100, deadline=None)
@given(items=st.lists(
    st.tuples(
        st.tuples(
            st.one_of(st.text(max_size=8),
                      st.integers(min_value=-2**40, max_value=2**40)),
            st.one_of(st.text(max_size=4),
                      st.integers(min_value=-100, max_value=100))),
        st.one_of(st.integers(min_value=1, max_value=10**9),
                  st.floats(min_value=0.001, max_value=1e9,
                            allow_nan=False))),
    max_size=30, unique_by=lambda kv: kv[0]))
def test_merge_wire_format_roundtrip(items):
    """The dense-merge encoder/decoder round-trips ANY table exactly
    (str/int key elements, int/float values) — the single-rank slice
    of the C1 wire format."""
    from dragnet_amd.distributed import _encode_table, _rebuild_table
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    q = query_load(breakdown_specs="a,b")
    agg = Aggregator(q)
    for k, v in items:
        agg.table[k] = v
    codes, tags, vals, strings = _encode_table(agg, 2)
    out = _rebuild_table(q, codes, tags, vals, strings)
    want = {k: (int(v) if float(v).is_integer() else float(v))
            for k, v in agg.table.items()}
    assert out.table == want


@settings(max_examples=150, deadline=None)
@given(y=st.integers(min_value=1970, max_value=2199),
       mo=st.integers(min_value=1, max_value=12),
       d=st.integers(min_value=1, max_value=31),
       hh=st.integers(min_value=0, max_value=23),
       mm=st.integers(min_value=0, max_value=59),
       ss=st.integers(min_value=0, max_value=59),
       ms=st.integers(min_value=0, max_value=999))
def test_jsdate_matches_datetime(y, mo, d, hh, mm, ss, ms):
    """parse_ms agrees with Python's datetime for every valid
    Z-suffixed timestamp; invalid day-of-month yields None."""
    import datetime as dt

    from dragnet_amd import jsdate
    s = "%04d-%02d-%02dT%02d:%02d:%02d.%03dZ" % (y, mo, d, hh, mm,
                                                 ss, ms)
    try:
        t = dt.datetime(y, mo, d, hh, mm, ss,
                        tzinfo=dt.timezone.utc)
        # integer epoch math: float timestamp()*1000 truncates the
        # millisecond (this very test caught that in its first form)
        want = int(t.timestamp()) * 1000 + ms
    except ValueError:
        want = None
    assert jsdate.parse_ms(s) == want
    if want is not None:
        # round trip through to_iso
        assert jsdate.to_iso(want / 1000.0) == s


@settings(max_examples=150, deadline=None)
@given(parts=st.lists(
    st.tuples(
        st.text(alphabet="abcxyz_.", min_size=1, max_size=8)
          .filter(lambda t: "," not in t and "[" not in t),
        st.sampled_from([None, "date", "quantize"])),
    min_size=1, max_size=4))
def test_attrs_roundtrip(parts):
    """attrs_parse on a spec rendered from structured parts recovers
    the same names/attrs (modulo the documented reference off-by-one
    we deliberately FIX)."""
    from dragnet_amd.attrs import attrs_parse
    frags = []
    for name, attr in parts:
        if attr == "date":
            frags.append("%s[date]" % name)
        elif attr == "quantize":
            frags.append("%s[aggr=quantize]" % name)
        else:
            frags.append(name)
    spec = ",".join(frags)
    out = attrs_parse(spec)
    assert not isinstance(out, Exception), (spec, out)
    assert [o["name"] for o in out] == [p[0] for p in parts]
    for o, (name, attr) in zip(out, parts):
        if attr == "date":
            assert "date" in o
        elif attr == "quantize":
            assert o.get("aggr") == "quantize"
