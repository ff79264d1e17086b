"""Aggregator / bucketizer / date semantics."""

from dragnet_amd import jsdate
from dragnet_amd.points import Aggregator, canonical, js_num_str
from dragnet_amd.query import (LinearBucketizer, P2Bucketizer, QueryConfig,
                               query_load)


def test_p2_bucketizer():
    b = P2Bucketizer()
    assert b.bucket(0) == 0
    assert b.bucket(0.5) == 0
    assert b.bucket(1) == 1
    assert b.bucket(2) == 2
    assert b.bucket(3) == 2
    assert b.bucket(4) == 3
    assert b.bucket(1024) == 11
    assert b.bucket(2047) == 11
    assert b.bucket(2048) == 12
    assert b.bucket_min(0) == 0
    assert b.bucket_min(1) == 1
    assert b.bucket_min(2) == 2
    assert b.bucket_min(3) == 4
    assert b.bucket_min(12) == 2048
    for v in [1, 2, 3, 7, 100, 4096]:
        assert b.bucket(b.bucket_min(b.bucket(v))) == b.bucket(v)


def test_linear_bucketizer():
    b = LinearBucketizer(100)
    assert b.bucket(0) == 0
    assert b.bucket(99) == 0
    assert b.bucket(100) == 1
    assert b.bucket_min(3) == 300
    assert b.bucket(b.bucket_min(b.bucket(250))) == 2


def test_canonical():
    from dragnet_amd.krill import MISSING
    assert canonical(MISSING) == "undefined"
    assert canonical(None) == "null"
    assert canonical(True) == "true"
    assert canonical(200) == "200"
    assert canonical(200.0) == "200"
    assert canonical("GET") == "GET"


def test_js_num_str():
    assert js_num_str(200) == "200"
    assert js_num_str(200.5) == "200.5"
    assert js_num_str(0) == "0"
    assert js_num_str(-3) == "-3"


def test_aggregator_basic():
    q = query_load(breakdown_specs="m")
    a = Aggregator(q)
    a.write({"fields": {"m": "GET"}, "value": 1})
    a.write({"fields": {"m": "GET"}, "value": 2})
    a.write({"fields": {"m": "PUT"}, "value": 1})
    assert a.points() == [
        {"fields": {"m": "GET"}, "value": 3},
        {"fields": {"m": "PUT"}, "value": 1},
    ]


def test_aggregator_nested_and_literal():
    q = query_load(breakdown_specs="req.method")
    a = Aggregator(q)
    a.write({"fields": {"req": {"method": "GET"}}, "value": 1})
    a.write({"fields": {"req.method": "GET"}, "value": 5})
    assert a.points() == [{"fields": {"req.method": "GET"}, "value": 6}]


def test_aggregator_null_undefined():
    q = query_load(breakdown_specs="req.caller")
    a = Aggregator(q)
    a.write({"fields": {"req": {"caller": None}}, "value": 1})
    a.write({"fields": {"req": {}}, "value": 1})
    a.write({"fields": {}, "value": 1})
    a.write({"fields": {"req": {"caller": "admin"}}, "value": 1})
    pts = a.points()
    assert pts == [
        {"fields": {"req.caller": "admin"}, "value": 1},
        {"fields": {"req.caller": "null"}, "value": 1},
        {"fields": {"req.caller": "undefined"}, "value": 2},
    ]


def test_aggregator_quantize_drop_nonnumeric():
    q = query_load(breakdown_specs="latency[aggr=quantize]")
    a = Aggregator(q)
    a.write({"fields": {"latency": 5}, "value": 1})
    # numeric STRINGS coerce (JS arithmetic in the reference's
    # bucketizer; proved by its own fileset golden counting
    # {"latency": "26"} — despite README.md:718-722's claim)
    a.write({"fields": {"latency": "5"}, "value": 1})
    a.write({"fields": {"latency": "abc"}, "value": 1})  # NaN: dropped
    a.write({"fields": {"latency": "Infinity"}, "value": 1})  # dropped
    a.write({"fields": {"latency": True}, "value": 1})  # bool: dropped
    a.write({"fields": {}, "value": 1})                 # missing: dropped
    assert a.ndropped_nonnumeric == 4
    assert a.points() == [{"fields": {"latency": 4}, "value": 2}]


def test_aggregator_zero_breakdowns():
    q = query_load()
    a = Aggregator(q)
    assert a.points() == [{"fields": {}, "value": 0}]
    assert a.rows() == [0]
    a.write({"fields": {"x": 1}, "value": 7})
    assert a.points() == [{"fields": {}, "value": 7}]
    assert a.noutputs() == 1


def test_points_reaggregate_idempotent():
    """points -> re-aggregate x3 triples the values (the reference's
    format_skinner invariant, tst.format_skinner.sh:25-37)."""
    q = query_load(breakdown_specs="m,latency[aggr=quantize]")
    a = Aggregator(q)
    for v, lat in [(1, 3), (2, 100), (5, 3)]:
        a.write({"fields": {"m": "GET", "latency": lat}, "value": v})
    pts = a.points()
    b = Aggregator(q)
    for _ in range(3):
        for p in pts:
            b.write(p)
    assert b.points() == [
        {"fields": dict(p["fields"]), "value": 3 * p["value"]}
        for p in pts]


def test_jsdate_roundtrip():
    for s, ms in [
        ("2014-05-01T00:00:00.000Z", 1398902400000),
        ("2014-05-01", 1398902400000),
        ("2014-05-01T00:05:45.600Z", 1398902745600),
        ("2014-05-02T04:05:06.123", 1399003506123),
        ("1970-01-01T00:00:00Z", 0),
        ("2014-05-02T00:00:00+02:00", 1398988800000 - 7200000),
    ]:
        assert jsdate.parse_ms(s) == ms, s
    assert jsdate.parse_ms("not-a-date") is None
    assert jsdate.parse_ms("2014-13-01") is None
    # V8 accepts hour 24 only as exactly 24:00:00.000
    assert jsdate.parse_ms("2014-05-01T24:00:00.000Z") == \
        1398902400000 + 86400000
    assert jsdate.parse_ms("2014-05-01T24:30:00Z") is None
    assert jsdate.parse_ms("2014-05-01T24:00:01Z") is None
    assert jsdate.parse_ms("2014-05-01T24:00:00.500Z") is None
    assert jsdate.to_iso(1398902745) == "2014-05-01T00:05:45.000Z"
    assert jsdate.to_iso(1398902745.6) == "2014-05-01T00:05:45.600Z"


def test_query_validation():
    import pytest
    from dragnet_amd.query import QueryError
    with pytest.raises(QueryError):
        query_load(breakdown_specs="__dn_ts")
    q = query_load(breakdown_specs="__dn_ts[aggr=lquantize,step=60]",
                   allow_reserved=True)
    assert q.breakdowns[0]["step"] == 60
    with pytest.raises(QueryError):
        query_load(breakdown_specs="x[aggr=bogus]")
    with pytest.raises(QueryError):
        query_load(breakdown_specs="x[aggr=lquantize]")
    with pytest.raises(QueryError):
        query_load(time_after="2014-01-01")
    with pytest.raises(QueryError):
        QueryConfig(time_after="2014-01-02", time_before="2014-01-01")
    q = query_load(breakdown_specs="ts[field=time,date]")
    assert q.synthetic == [{"name": "ts", "field": "time", "date": ""}]


def test_date_column_nonscalar_reaggregation():
    """A list/dict value reaching a DATE column on the
    re-aggregation path (scan-side synthetics can't filter it there)
    coerces like any JS object key instead of crashing — found by
    the codec soak (points.canonical raw-passthrough bug)."""
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    q = query_load(
        breakdown_specs="t[date,field=time,aggr=lquantize,step=60],a")
    agg = Aggregator(q)
    # lquantize'd date column: non-numeric -> nonnumeric drop
    assert agg.write({"fields": {"t": [1], "a": "x"}, "value": 1}) \
        is False
    assert agg.ndropped_nonnumeric == 1
    # plain (non-aggregated) date column: JS string coercion
    q2 = query_load(breakdown_specs="t[date,field=time],a")
    agg2 = Aggregator(q2)
    assert agg2.write({"fields": {"t": [1, None], "a": "x"},
                       "value": 2}) is True
    assert agg2.write({"fields": {"t": {"z": 1}, "a": "x"},
                       "value": 3}) is True
    assert agg2.table == {("1,", "x"): 2, ("[object Object]", "x"): 3}
