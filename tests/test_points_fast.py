"""Differential test for the C++ tagged-point reducer (index/_points)
against the pure-Python cmd_index_read loop (json.loads +
points.Aggregator.write): identical tables and counters over an
adversarial line mix, with every doubtful shape punted back to Python.

Float values in the mix are dyadic (k/2^n) so sums are exact and
order-independent — the fast path may reorder additions within a
group (documented last-ulp corner; the reference's Manta reduce order
is equally arbitrary, lib/datasource-manta.js:212-219)."""

import io
import json
import random

import pytest

from dragnet_amd.points import Aggregator, reduce_tagged_stream
from dragnet_amd.query import query_load

pytest.importorskip("dragnet_amd.index._points")


@pytest.fixture(autouse=True)
def _force_native(monkeypatch):
    """These tests exercise the codec itself — clear the env gate
    that forces the pure-Python path."""
    monkeypatch.delenv("DRAGNET_PY_POINTS", raising=False)


QUERIES = [
    query_load(breakdown_specs="a,t[date,field=time],"
                               "lat[aggr=quantize],"
                               "ts[aggr=lquantize,step=60]"),
    query_load(breakdown_specs="req.method,res.statusCode"),
    query_load(),  # zero breakdowns
]


def py_reduce(data, queries):
    """The exact cmd_index_read per-line Python body."""
    aggs = [Aggregator(q) for q in queries]
    for line in data.split(b"\n"):
        line = line.strip()
        if not line:
            continue
        try:
            p = json.loads(line)
        except ValueError:
            continue
        mi = p.get("fields", {}).get("__dn_metric")
        if not isinstance(mi, int) or not (0 <= mi < len(aggs)):
            continue
        aggs[mi].write(p)
    return aggs


def fast_reduce(data, queries):
    aggs = [Aggregator(q) for q in queries]
    punted = reduce_tagged_stream(io.BytesIO(data), aggs, queries)
    for line in punted:
        line = line.strip()
        if not line:
            continue
        try:
            p = json.loads(line)
        except ValueError:
            continue
        mi = p.get("fields", {}).get("__dn_metric")
        if not isinstance(mi, int) or not (0 <= mi < len(aggs)):
            continue
        aggs[mi].write(p)
    return aggs


def check(lines):
    data = b"\n".join(lines) + b"\n"
    a = py_reduce(data, QUERIES)
    b = fast_reduce(data, QUERIES)
    for i, (pa, pb) in enumerate(zip(a, b)):
        assert pa.ninputs == pb.ninputs, i
        assert pa.ndropped_nonnumeric == pb.ndropped_nonnumeric, i
        assert pa.table == pb.table, (i, pa.table, pb.table)
        for (ka, va), (kb, vb) in zip(sorted(pa.table.items(),
                                             key=repr),
                                      sorted(pb.table.items(),
                                             key=repr)):
            assert type(va) is type(vb), (ka, va, vb)


def tag(mi, fields, value=1):
    f = {"__dn_metric": mi}
    f.update(fields)
    return json.dumps({"fields": f, "value": value}).encode()


def test_flat_fast_shapes():
    check([
        tag(0, {"a": "x", "t": 1400000000, "lat": 26, "ts": 61}),
        tag(0, {"a": "x", "t": 1400000000, "lat": 26, "ts": 61}, 5),
        tag(0, {"a": "y", "t": 1400000003, "lat": 0, "ts": 0}),
        tag(0, {"a": None, "t": True, "lat": "26", "ts": "100"}),
        tag(0, {"t": 1400000000.9, "lat": 3.5, "ts": -61.5}),
        tag(1, {"req.method": "GET", "res.statusCode": 200}),
        tag(1, {"req.method": "GET", "res.statusCode": "200"}),
        tag(1, {"req.method": False, "res.statusCode": -17}),
        tag(2, {}),
        tag(2, {"unrelated": "zzz"}, 3),
    ])


def test_nonnumeric_drops_and_coercions():
    check([
        tag(0, {"a": "k", "t": 1, "lat": None, "ts": 1}),       # drop
        tag(0, {"a": "k", "t": 1, "lat": True, "ts": 1}),       # drop
        tag(0, {"a": "k", "t": 1, "lat": "zzz", "ts": 1}),      # drop
        tag(0, {"a": "k", "t": 1, "lat": "", "ts": 1}),         # ""->0
        tag(0, {"a": "k", "t": 1, "lat": " 12 ", "ts": 1}),
        tag(0, {"a": "k", "t": 1, "lat": "1e999", "ts": 1}),    # inf drop
        tag(0, {"a": "k", "t": 1, "lat": "-3.5e2", "ts": 1}),
        tag(0, {"a": "k", "t": 1, "ts": 1}),                    # missing
        tag(0, {"a": "k", "t": 1, "lat": "0x10", "ts": 1}),     # punt
        tag(0, {"a": "k", "t": 1, "lat": "Infinity", "ts": 1}),  # punt
        tag(0, {"a": "k", "t": 1, "lat": "1_0", "ts": 1}),      # punt
    ])


def test_punted_shapes_match():
    check([
        # escapes, unicode, nesting, big ints, non-integral floats
        tag(0, {"a": "x\ny", "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": "café", "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": {"nested": 1}, "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": [1, "b"], "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": 10**20, "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": 2.5, "t": 1, "lat": 1, "ts": 1}),
        tag(0, {"a": 1e300, "t": 1, "lat": 1, "ts": 1}),
        tag(1, {"req.method": "G", "res.statusCode": 1}, 2.25),
        # bad metric routing: skipped by both paths
        tag(5, {"a": "x"}),
        tag(-1, {"a": "x"}),
        json.dumps({"fields": {"__dn_metric": 1.0, "a": "x"},
                    "value": 1}).encode(),
        json.dumps({"fields": {"__dn_metric": True,
                               "req.method": "B",
                               "res.statusCode": 2},
                    "value": 1}).encode(),
        # garbage / blank / invalid json
        b"", b"   ", b"not json", b"{truncated",
        b'{"fields":{"__dn_metric":0,"a":"dup","a":"dup2",'
        b'"t":1,"lat":1,"ts":1},"value":1}',
    ])


def test_randomized_mix():
    rng = random.Random(7)
    vals = ["s", "x", "", None, True, False, 0, 7, -3, 2**40,
            0.5, -1.25, "26", " 7 ", "NaN", "café", [1], {"z": 1}]
    lines = []
    for _ in range(800):
        mi = rng.choice([0, 1, 2, 3, -2])
        fields = {}
        for name in ("a", "t", "lat", "ts", "req.method",
                     "res.statusCode", "junk"):
            if rng.random() < 0.6:
                fields[name] = rng.choice(vals)
        # dyadic float values keep sums order-independent
        value = rng.choice([1, 2, 5, 0.5, 0.25, 3])
        lines.append(tag(mi, fields, value))
        if rng.random() < 0.1:
            lines.append(rng.choice(
                [b"", b"garbage", b'{"value":1}']))
    check(lines)


def test_streaming_slab_boundary():
    """Slab splitting in reduce_tagged_stream can't lose or split
    lines (feed through a 1-byte-read stream wrapper)."""
    lines = [tag(0, {"a": "q%d" % i, "t": 1, "lat": i, "ts": i})
             for i in range(50)]
    data = b"\n".join(lines) + b"\n"

    class Dribble(io.RawIOBase):
        def __init__(self, b):
            self.b = b
            self.i = 0

        def read(self, n=-1):
            if self.i >= len(self.b):
                return b""
            chunk = self.b[self.i:self.i + 7]
            self.i += 7
            return chunk

    aggs = [Aggregator(q) for q in QUERIES]
    punted = reduce_tagged_stream(Dribble(data), aggs, QUERIES)
    assert punted == []
    ref = py_reduce(data, QUERIES)
    assert aggs[0].table == ref[0].table
    assert aggs[0].ninputs == ref[0].ninputs


def test_serializer_byte_exact():
    """serialize_points output is byte-identical to per-point
    json.dumps across escapes, unicode, numbers, and fallback
    shapes (nested values)."""
    from dragnet_amd.index import _points
    from dragnet_amd.output import point_json
    pts = [
        {"fields": {"a": "x", "n": 200, "f": 2.5}, "value": 1},
        {"fields": {"a": 'q"\\\n\t\x07\x7f', "b": "café",
                    "c": "\U0001F600"}, "value": 3},
        {"fields": {"neg": -17, "big": 10**25, "z": -0.0,
                    "e": 1e-5, "E": 1e16}, "value": 0.25},
        {"fields": {"t": True, "f2": False, "n2": None}, "value": 7},
        {"fields": {}, "value": 2},
        {"fields": {"nested": {"x": 1}, "l": [1, "a", None]},
         "value": 1},  # falls back to point_json
        {"fields": {"inf": float("inf"), "nan": float("nan")},
         "value": 1},
    ]
    got = _points.serialize_points(pts, point_json)
    want = b"".join(point_json(p).encode() + b"\n" for p in pts)
    assert got == want


def test_serializer_reducer_roundtrip():
    """points -> serialize -> reduce recovers the exact table (the
    map/reduce pipe identity the index build rests on)."""
    from dragnet_amd.index import _points
    from dragnet_amd.output import point_json
    q = QUERIES[0]
    src = Aggregator(q)
    rng = random.Random(11)
    for i in range(500):
        src.write({"fields": {"a": "k%d" % (i % 17),
                              "t": 1400000000 + i,
                              "lat": rng.choice([1, 5, 80]),
                              "ts": rng.randrange(0, 600)},
                   "value": rng.randint(1, 9)})
    pts = src.points()
    for p in pts:
        p["fields"]["__dn_metric"] = 0
    data = _points.serialize_points(pts, point_json)
    aggs = [Aggregator(qq) for qq in QUERIES]
    punted = reduce_tagged_stream(io.BytesIO(data), aggs, QUERIES)
    assert punted == []
    assert aggs[0].table == src.table
    assert sum(aggs[0].table.values()) == sum(src.table.values())


from hypothesis import given, settings  # noqa: E402
from hypothesis import strategies as st  # noqa: E402

_scalar = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-2**70, max_value=2**70),
    st.floats(allow_nan=False, allow_infinity=False),
    st.text(max_size=10),
    st.lists(st.integers(min_value=0, max_value=3), max_size=2),
)
_fieldname = st.sampled_from(
    ["a", "t", "lat", "ts", "req.method", "res.statusCode",
     "__dn_metric", "junk", "ünïcode"])
_point_line = st.builds(
    lambda f, v: json.dumps({"fields": f, "value": v}).encode(),
    st.dictionaries(_fieldname, _scalar, max_size=6),
    st.one_of(st.integers(min_value=-10**6, max_value=10**6),
              st.sampled_from([0.5, 0.25, 1.75, 3.0])))
_raw_line = st.binary(max_size=24).map(lambda b: b.replace(b"\n", b"x"))


@settings(max_examples=150, deadline=None)
@given(lines=st.lists(st.one_of(_point_line, _raw_line, st.just(b"")),
                      max_size=25))
def test_property_reducer_differential(lines):
    """Any mix of point-shaped and arbitrary lines reduces
    identically through the C++ fast path + punts and the pure
    Python loop (values restricted to exactly-representable sums so
    addition order cannot matter)."""
    # lines whose json is a non-dict would crash BOTH paths
    # identically (AttributeError in the CLI); keep the differential
    # on the non-crashing subset
    keep = []
    for ln in lines:
        try:
            v = json.loads(ln.strip() or b"{}")
            if not isinstance(v, dict):
                continue
            if not isinstance(v.get("fields", {}), dict):
                continue
            mi = v.get("fields", {}).get("__dn_metric")
            if isinstance(mi, int) and 0 <= mi < len(QUERIES) \
                    and "value" not in v:
                continue  # KeyError in both paths
        except ValueError:
            pass
        keep.append(ln)
    check(keep) if keep else None
