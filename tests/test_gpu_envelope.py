"""
Divergence-envelope pinning (pytest -m gpu): the documented
CPU-vs-GPU divergence corners (COMPONENTS.md "Known divergence
corners") must stay DROPS or sentinels — a regression from
"documented drop" to "silent wrong answer" fails here.

Each test asserts the exact documented behavior deterministically,
plus the conservation invariant (every record is aggregated or
attributed to a drop counter) for a fuzz mode that deliberately
CROSSES the envelope.
"""

import json
import random

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engines():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.engine.gpu import GpuEngine
    return CpuEngine(), GpuEngine()


def q(**kw):
    from dragnet_amd.query import query_load
    return query_load(**kw)


def gpu_scan(gpu, tmp_path, lines, query):
    f = tmp_path / "env.log"
    f.write_bytes(b"".join(ln + b"\n" for ln in lines))
    return gpu.scan([str(f)], [query])


def stage(res, name):
    return dict(res.stages)[name]


def total_value(res):
    return sum(p["value"] for p in res.aggregators[0].points())


def test_depth_beyond_64_drops_as_invalid(engines, tmp_path):
    """Nesting deeper than MAX_DEPTH=64 must DROP (invalid json
    counter) — never parse wrongly.  Depth exactly 64 must parse and
    match the oracle."""
    cpu, gpu = engines

    def nested(depth, val):
        return ("{\"k\":" * depth) + json.dumps(val) + ("}" * depth)

    ok_line = nested(63, {"a": "x"})  # 64 levels incl. the leaf obj
    deep_line = nested(80, {"a": "x"})
    plain = json.dumps({"a": "y"})
    lines = [ok_line.encode(), deep_line.encode(), plain.encode()]

    query = q(breakdown_specs="a")
    g = gpu_scan(gpu, tmp_path, lines, query)
    p = stage(g, "json parser")
    # the deep record is attributed, not silently mangled
    assert p["ninputs"] == 3
    assert p["invalid json"] == 1
    assert p["noutputs"] == 2
    # the accepted records aggregate exactly like the oracle's view
    # of the same two lines
    f2 = tmp_path / "ok.log"
    f2.write_bytes(ok_line.encode() + b"\n" + plain.encode() + b"\n")
    c = cpu.scan([str(f2)], [query])
    assert g.aggregators[0].points() == c.aggregators[0].points()


def test_escaped_key_stays_sentinel(engines, tmp_path):
    """A backslash-u-escaped KEY hashes its raw bytes, so lookups for
    the unescaped name MISS: the documented behavior is the
    'undefined' sentinel group — never a match against the unescaped
    key, never a crash."""
    _cpu, gpu = engines
    lines = [
        b'{"\\u0061bc": "v1"}',   # key "abc" spelled escaped
        b'{"abc": "v2"}',
    ]
    g = gpu_scan(gpu, tmp_path, lines, q(breakdown_specs="abc"))
    pts = g.aggregators[0].points()
    by_key = {p["fields"]["abc"]: p["value"] for p in pts}
    # escaped-key record lands in the sentinel group, plain one matches
    assert by_key == {"undefined": 1, "v2": 1}
    # conservation: both records accounted for
    assert total_value(g) == 2


def test_huge_digit_numbers_conserve(engines, tmp_path):
    """>19-significant-digit numbers may round differently than strtod
    in the last ulp (dict identity only): the records must still
    AGGREGATE (never drop, never crash) and group counts conserve."""
    cpu, gpu = engines
    lines = [
        b'{"n": 12345678901234567890123}',
        b'{"n": 12345678901234567890123}',
        b'{"n": 9.87654321098765432109876543e40}',
        b'{"n": 1}',
    ]
    query = q(breakdown_specs="n")
    g = gpu_scan(gpu, tmp_path, lines, query)
    p = stage(g, "json parser")
    assert p["invalid json"] == 0
    assert total_value(g) == 4
    # identical spellings intern identically: the duplicate pair
    # groups together on both engines
    vals = sorted(pp["value"] for pp in g.aggregators[0].points())
    f2 = tmp_path / "cpu.log"
    f2.write_bytes(b"".join(ln + b"\n" for ln in lines))
    c = cpu.scan([str(f2)], [query])
    cvals = sorted(pp["value"] for pp in c.aggregators[0].points())
    assert vals == cvals  # same grouping structure (2,1,1)


def test_envelope_crossing_fuzz_classification(engines, tmp_path):
    """Fuzz ACROSS the envelope (deep nesting, >19-digit numbers,
    escaped keys, binary garbage): only drop-vs-accept classification
    and conservation are asserted — parsed + invalid == lines, and
    every accepted record lands in some group (total value ==
    aggregator ninputs)."""
    _cpu, gpu = engines
    rng = random.Random(99)
    lines = []
    for i in range(4000):
        r = rng.random()
        if r < 0.2:
            d = rng.randint(60, 90)
            lines.append((("{\"k\":" * d) + '"v"' + ("}" * d)).encode())
        elif r < 0.4:
            digits = "".join(rng.choice("123456789")
                             for _ in range(rng.randint(18, 30)))
            lines.append(('{"n": %s}' % digits).encode())
        elif r < 0.55:
            lines.append(b'{"\\u006b%d": "x"}' % (i % 7))
        elif r < 0.7:
            lines.append(bytes(rng.randrange(256)
                               for _ in range(rng.randint(1, 40)))
                         .replace(b"\n", b"x"))
        else:
            lines.append(json.dumps(
                {"k%d" % (i % 7): rng.choice(["a", "b", 3, None]),
                 "n": rng.randint(0, 99)}).encode())

    query = q(breakdown_specs="n")
    g = gpu_scan(gpu, tmp_path, lines, query)
    p = stage(g, "json parser")
    assert p["ninputs"] == len(lines)
    assert p["ninputs"] == p["noutputs"] + p["invalid json"]
    agg = stage(g, "Aggregator")
    assert agg["ninputs"] == p["noutputs"]
    # every accepted record is in some group (value weight 1 each)
    assert total_value(g) == agg["ninputs"] - agg.get("nonnumeric", 0)
