"""
GPU engine differential tests: the HIP scan kernel must reproduce the
CPU oracle's aggregates and drop counters exactly, over the fixture
tree and generated synthetic data (numerics strategy per the build
contract: HIP kernel vs plain reference of the same op).

All tests here require an MI355X (pytest -m gpu).
"""

import os

import pytest

from scan_cases import SCAN_CASES

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engines():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.engine.gpu import GpuEngine
    return CpuEngine(), GpuEngine()


def queries_from_case(case):
    from dragnet_amd.query import query_load
    import json
    filt = None
    bds = None
    it = iter(case)
    for a in it:
        if a == "-f":
            filt = json.loads(next(it))
        elif a == "-b":
            bds = next(it)
    return query_load(filter=filt, breakdown_specs=bds)


def assert_same(cpu_res, gpu_res):
    for ca, ga in zip(cpu_res.aggregators, gpu_res.aggregators):
        assert ga.points() == ca.points()
        assert ga.ninputs == ca.ninputs
        assert ga.ndropped_nonnumeric == ca.ndropped_nonnumeric


def test_scan_cases_single_file(engines, fixture_tree):
    cpu, gpu = engines
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    for case in SCAN_CASES:
        q = queries_from_case(case)
        c = cpu.scan([one], [q])
        g = gpu.scan([one], [q])
        assert_same(c, g)


def test_scan_cases_fileset(engines, fixture_tree):
    """Whole tree: includes invalid JSON, bad dates, missing time."""
    cpu, gpu = engines
    files = []
    for root, _dirs, names in os.walk(fixture_tree):
        for n in sorted(names):
            files.append(os.path.join(root, n))
    files.sort()
    for case in SCAN_CASES[:6]:
        q = queries_from_case(case)
        c = cpu.scan(files, [q])
        g = gpu.scan(files, [q])
        assert_same(c, g)
        # parser counters match too
        cs = dict(c.stages)["json parser"]
        gs = dict(g.stages)["json parser"]
        assert cs == gs


def test_ds_filter(engines, fixture_tree):
    cpu, gpu = engines
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    from dragnet_amd.query import query_load
    q = query_load(breakdown_specs="operation")
    f = {"eq": ["req.method", "GET"]}
    c = cpu.scan([one], [q], ds_filter=f)
    g = gpu.scan([one], [q], ds_filter=f)
    assert_same(c, g)


def test_dates_and_time_filter(engines, fixture_tree):
    cpu, gpu = engines
    files = []
    for root, _dirs, names in os.walk(fixture_tree):
        for n in sorted(names):
            files.append(os.path.join(root, n))
    files.sort()
    from dragnet_amd.query import query_load
    q = query_load(
        breakdown_specs="ts[date,field=time,aggr=lquantize,step=3600]",
        time_after="2014-05-02T06:00:00", time_before="2014-05-04")
    c = cpu.scan(files, [q], time_field="time")
    g = gpu.scan(files, [q], time_field="time")
    assert_same(c, g)
    # every stage's counters agree
    assert c.stages == g.stages or \
        [s for s in c.stages if s[0] != "Aggregator"] == \
        [s for s in g.stages if s[0] != "Aggregator"]


def test_multi_metric(engines, fixture_tree):
    """Build-style fan-out: one parse pass, N aggregations."""
    cpu, gpu = engines
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    from dragnet_amd.query import query_load
    qs = [
        query_load(breakdown_specs="operation,host"),
        query_load(filter={"eq": ["req.method", "GET"]},
                   breakdown_specs="res.statusCode"),
        query_load(breakdown_specs="latency[aggr=quantize]"),
    ]
    c = cpu.scan([one], qs)
    g = gpu.scan([one], qs)
    assert_same(c, g)


def test_skinner_format(engines, fixture_tree, tmp_path):
    cpu, gpu = engines
    one = os.path.join(fixture_tree, "2014", "05-01", "one.log")
    from dragnet_amd.query import query_load
    from dragnet_amd.output import point_json
    q = query_load(breakdown_specs="req.method,res.statusCode")
    base = cpu.scan([one], [q]).aggregators[0].points()
    pfile = tmp_path / "points.ndjson"
    with open(pfile, "w") as f:
        for _ in range(3):
            for p in base:
                f.write(point_json(p) + "\n")
    q2 = query_load(breakdown_specs="req.method")
    c = cpu.scan([str(pfile)], [q2], data_format="json-skinner")
    g = gpu.scan([str(pfile)], [q2], data_format="json-skinner")
    assert_same(c, g)
    assert g.aggregators[0].points()  # non-empty


def test_generated_bulk(engines, tmp_path):
    """200k mktestdata-shaped records (the benchmark workload)."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    path = tmp_path / "bulk.ndjson"
    with open(path, "wb") as f:
        for line in generate_lines(200_000, seed=7):
            f.write(line)
    q = query_load(
        filter={"eq": ["req.method", "GET"]},
        breakdown_specs="req.method,res.statusCode")
    c = cpu.scan([str(path)], [q])
    g = gpu.scan([str(path)], [q])
    assert_same(c, g)
    q2 = query_load(breakdown_specs="req.url")  # high cardinality
    c = cpu.scan([str(path)], [q2])
    g = gpu.scan([str(path)], [q2])
    assert_same(c, g)
    q3 = query_load(breakdown_specs="dataLatency[aggr=lquantize,step=100]")
    c = cpu.scan([str(path)], [q3])
    g = gpu.scan([str(path)], [q3])
    assert_same(c, g)


def test_edge_cases(engines, tmp_path):
    """Malformed lines, empty lines, nested/absent fields, type mix."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    data = b"\n".join([
        b'{"a": 1, "b": {"c": "x"}}',
        b'{"a": "1", "b": {}}',
        b'not json at all',
        b'',
        b'{"a": null, "b": {"c": null}}',
        b'{"a": true, "b": {"c": ["arr"]}}',
        b'{"a": 2.5e3, "b": {"c": {"d": 1}}}',
        b'[1, 2, 3]',
        b'"bare string"',
        b'42',
        b'{"a": 0.125, "b": {"c": "x", "c": "y"}}',
        b'{"dup": 1, "dup": 2}',
        b'{"a": -17}',
    ]) + b"\n"
    path = tmp_path / "edge.ndjson"
    path.write_bytes(data)
    for spec, filt in [
        ("a", None),
        ("b.c", None),
        ("a[aggr=quantize]", None),
        ("dup", None),
        ("a", {"eq": ["a", 1]}),
        ("a", {"eq": ["a", "1"]}),
        ("b.c", {"eq": ["b.c", "x"]}),
        (None, {"lt": ["a", 2]}),
        (None, {"or": [{"eq": ["a", 1]}, {"eq": ["b.c", "x"]}]}),
    ]:
        q = query_load(filter=filt, breakdown_specs=spec)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        assert g.aggregators[0].points() == c.aggregators[0].points(), \
            (spec, filt)
        cs = dict(c.stages)["json parser"]
        gs = dict(g.stages)["json parser"]
        assert cs == gs, (spec, filt)


def test_escaped_string_filter(engines, tmp_path):
    """Filter constants match records carrying the value in escaped
    form (plan.py stores the canonical JSON escaping as an alternate
    compare target)."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    data = (b'{"m": "a\\nb", "x": 1}\n'       # escaped newline
            b'{"m": "a\\u000ab", "x": 2}\n'   # same value, \\u form
            b'{"m": "anb", "x": 3}\n'
            b'{"m": "quote\\"q", "x": 4}\n')
    path = tmp_path / "esc.ndjson"
    path.write_bytes(data)
    for filt in [{"eq": ["m", "a\nb"]}, {"eq": ["m", 'quote"q']}]:
        q = query_load(filter=filt)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        cp = c.aggregators[0].points()
        gp = g.aggregators[0].points()
        # both the canonical escaping (alt-form fast path) and the
        # non-canonical \\u000a form (unescape-compare fallback) match
        if filt == {"eq": ["m", "a\nb"]}:
            assert cp[0]["value"] == 2
        assert gp == cp
    # group keys decode identically regardless of escape form
    q = query_load(breakdown_specs="m")
    c = cpu.scan([str(path)], [q])
    g = gpu.scan([str(path)], [q])
    assert g.aggregators[0].points() == c.aggregators[0].points()


def test_noncanonical_escape_filters(engines, tmp_path):
    """Unescape-on-device compare: \\uXXXX-spelled ASCII, \\/, BMP and
    surrogate-pair escapes, and relational compares over escaped
    spans all match the CPU oracle exactly."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    raw = (b'{"m": "\\u0047ET", "u": "/a\\/b", "s": "x\\u00e9y"}\n'
           b'{"m": "GET", "u": "/a/b", "s": "x\xc3\xa9y"}\n'
           b'{"m": "P\\u004fST", "u": "za", "s": "\\ud83d\\ude00"}\n'
           b'{"m": "GEU", "u": "zb", "s": "\xf0\x9f\x98\x80"}\n')
    path = tmp_path / "nce.ndjson"
    path.write_bytes(raw)
    for filt in [
        {"eq": ["m", "GET"]},            # GET == GET
        {"ne": ["m", "GET"]},
        {"eq": ["u", "/a/b"]},           # \/ == /
        {"eq": ["s", "xéy"]},       # BMP escape == utf-8 bytes
        {"eq": ["s", "\U0001F600"]},     # surrogate pair == utf-8
        {"lt": ["m", "GEU"]},            # relational over escaped span
        {"ge": ["m", "GET"]},
    ]:
        q = query_load(filter=filt)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        assert g.aggregators[0].points() == c.aggregators[0].points(), \
            filt


def test_resident_streaming_passes(engines, tmp_path):
    """The bench's streaming path: stage a pool once, run several
    full passes (ping-pong device pools, sliced copy/kernel overlap,
    pipelined extraction) — every pass must equal the oracle, and
    pipelined decode of pass k must survive pass k+1 running."""
    cpu, gpu = engines
    from dragnet_amd.engine import plan as planmod
    from dragnet_amd.engine.gpu import _ScanContext
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    lines = []
    for line in generate_lines(20000, seed=77):
        lines.append(line)
    pool = b"".join(lines)
    path = tmp_path / "pool.ndjson"
    path.write_bytes(pool)
    q = query_load(filter={"eq": ["req.method", "GET"]},
                   breakdown_specs="req.method,res.statusCode")
    expected = cpu.scan([str(path)], [q]).aggregators[0].points()

    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(gpu, cplan, agg_slots=1 << 14,
                       dict_slots=1 << 14, dict_data_cap=4 << 20)
    ctx.stage_resident(pool)
    prev = None
    results = []
    for _ in range(4):
        ctx.reset()
        ctx.scan_resident(h2d=True)
        ex = ctx.extract_async([q])
        if prev is not None:
            aggs, _ = ctx.decode_extracted(prev, [q])
            results.append(aggs[0].points())
        prev = ex
    aggs, _ = ctx.decode_extracted(prev, [q])
    results.append(aggs[0].points())
    assert len(results) == 4
    for r in results:
        assert r == expected


def test_literal_dotted_keys(engines, tmp_path):
    """Reference lookup split: krill pluck / synthetic sources never
    see a literal "a.b" key, but the aggregation lookup is
    literal-first at the top level (points.lookup).  The kernel routes
    literal-dotted captures to companion slots."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    raw = (b'{"a.b": 1, "m": "w"}\n'
           b'{"a": {"b": 2}, "m": "x"}\n'
           b'{"a.b": 3, "a": {"b": 4}, "m": "y"}\n'
           b'{"x": {"a.b": 5}, "m": "z"}\n'
           b'{"a.b": {"c": 6}, "m": "q"}\n')
    path = tmp_path / "dot.ndjson"
    path.write_bytes(raw)
    for spec, filt in [("a.b", None), ("x.a.b", None), ("a.b.c", None),
                       ("m", {"eq": ["a.b", 2]}),
                       (None, {"gt": ["a.b", 0]}),
                       ("a.b,m", {"eq": ["m", "y"]})]:
        q = query_load(filter=filt, breakdown_specs=spec)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        assert g.aggregators[0].points() == c.aggregators[0].points(), \
            (spec, filt)
        assert g.aggregators[0].ninputs == c.aggregators[0].ninputs


def test_xpose_scan(engines, tmp_path):
    """Wave-transposed staging (scan_kernel_x): length-sorted records
    in 64B-granule interleaved layout must aggregate identically to
    the oracle, including invalid/blank/edge lines."""
    cpu, gpu = engines
    from dragnet_amd.engine import plan as planmod
    from dragnet_amd.engine.gpu import _ScanContext
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    lines = list(generate_lines(30000, seed=99))
    lines += [b"not json\n", b"\n", b'{"a": 1}\n',
              b'{"req": {"method": "GET"}, "res": {"statusCode": 200}}\n',
              b'{"m": "' + b"x" * 500 + b'"}\n']  # long record
    pool = b"".join(lines)
    path = tmp_path / "xp.ndjson"
    path.write_bytes(pool)
    for filt, spec in [({"eq": ["req.method", "GET"]},
                        "req.method,res.statusCode"),
                       (None, "latency[aggr=quantize]")]:
        q = query_load(filter=filt, breakdown_specs=spec)
        exp = cpu.scan([str(path)], [q])
        cplan = planmod.compile_plan([q])
        ctx = _ScanContext(gpu, cplan, agg_slots=1 << 15,
                           dict_slots=1 << 15, dict_data_cap=8 << 20)
        ctx.stage_xpose(pool)
        for _ in range(2):  # repeat pass: reset correctness
            ctx.reset()
            ctx.scan_xpose()
            aggs, stages = ctx.finalize([q])
        assert aggs[0].points() == exp.aggregators[0].points(), \
            (filt, spec)
        assert dict(stages)["json parser"] == \
            dict(exp.stages)["json parser"]


def test_deep_nesting(engines, tmp_path):
    """Nesting to depth 64 parses identically to the oracle (the
    capture-slot window is 12 deep, but deeper containers sit inside
    arrays and are never capture targets)."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    lines = []
    for d in (2, 11, 13, 30, 60):
        deep = '"leaf"'
        for i in range(d):
            deep = ('[%s]' % deep) if i % 2 else ('{"k%d": %s}' % (i, deep))
        lines.append('{"m": "GET", "deep": %s, "v": %d}' % (deep, d))
    path = tmp_path / "deep.ndjson"
    path.write_text("\n".join(lines) + "\n")
    for spec, filt in [("m", None), ("v", {"eq": ["m", "GET"]}),
                       (None, {"gt": ["v", 12]})]:
        q = query_load(filter=filt, breakdown_specs=spec)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        assert g.aggregators[0].points() == c.aggregators[0].points(), \
            (spec, filt)
        assert dict(g.stages)["json parser"] == \
            dict(c.stages)["json parser"]


def test_overflow_regrow(engines, tmp_path, monkeypatch):
    """Tiny tables force C_OVERFLOW; the engine restarts with larger
    capacity and still produces oracle-identical results."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    path = tmp_path / "many.ndjson"
    with open(path, "wb") as f:
        for line in generate_lines(50_000, seed=11):
            f.write(line)
    monkeypatch.setenv("DRAGNET_AGG_SLOTS", "128")
    monkeypatch.setenv("DRAGNET_DICT_SLOTS", "128")
    monkeypatch.setenv("DRAGNET_DICT_DATA_MB", "1")
    from dragnet_amd.engine.gpu import GpuEngine
    g = GpuEngine().scan([str(path)],
                         [query_load(breakdown_specs="req.url")])
    monkeypatch.delenv("DRAGNET_AGG_SLOTS")
    monkeypatch.delenv("DRAGNET_DICT_SLOTS")
    monkeypatch.delenv("DRAGNET_DICT_DATA_MB")
    c = cpu.scan([str(path)], [query_load(breakdown_specs="req.url")])
    assert g.aggregators[0].points() == c.aggregators[0].points()


def test_deterministic_across_runs(engines, tmp_path):
    """Two GPU scans of the same input produce identical results
    (integer counts accumulate exactly in f64; hash-table insert order
    does not affect the decoded aggregate)."""
    _cpu, gpu = engines
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    path = tmp_path / "det.ndjson"
    with open(path, "wb") as f:
        for line in generate_lines(100_000, seed=13):
            f.write(line)
    q = query_load(breakdown_specs="req.url,req.method")
    a = gpu.scan([str(path)], [q]).aggregators[0].points()
    b = gpu.scan([str(path)], [q]).aggregators[0].points()
    assert a == b


def test_many_chunk_boundaries(engines, tmp_path):
    """Tiny chunks force hundreds of boundary tail-carries; record
    framing must survive every one (regression: the chunk padding once
    clobbered carried tail bytes at multi-GB scale)."""
    cpu, gpu = engines
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    path = tmp_path / "bounds.ndjson"
    with open(path, "wb") as f:
        for line in generate_lines(20_000, seed=17):
            f.write(line)
    q = query_load(breakdown_specs="req.method,res.statusCode")
    old = gpu.chunk_bytes
    try:
        gpu.chunk_bytes = 1 << 14  # 16 KB chunks -> ~280 boundaries
        g = gpu.scan([str(path)], [q])
    finally:
        gpu.chunk_bytes = old
    c = cpu.scan([str(path)], [q])
    gs = dict(g.stages)["json parser"]
    assert gs["ninputs"] == 20_000
    assert gs["invalid json"] == 0
    assert g.aggregators[0].points() == c.aggregators[0].points()


def test_dense_mfma_vs_hash_path(engines, tmp_path):
    """The dense-accumulation path (slot directory + per-workgroup
    partial matrix + MFMA f64 column-sum reduce) must equal the atomic
    hash path and the CPU oracle exactly — including near the dense
    directory's capacity and with json-skinner weights."""
    import json as _json
    import random

    from dragnet_amd.engine.gpu import GpuEngine
    from dragnet_amd.query import query_load
    cpu, _ = engines

    rng = random.Random(7)
    lines = []
    # ~3000 distinct (a, b) keys: well into the directory, near the
    # 4096-slot capacity, all slots exercised by the MFMA reduce
    for i in range(120000):
        rec = {"a": "k%d" % rng.randint(0, 120),
               "b": rng.randint(0, 24),
               "lat": rng.choice([3, 17, "99", 2 ** rng.randint(0, 20)])}
        lines.append(_json.dumps(rec))
    f = tmp_path / "dense.log"
    f.write_bytes(("\n".join(lines) + "\n").encode())

    q = query_load(breakdown_specs="a,b,lat[aggr=quantize]")
    c = cpu.scan([str(f)], [q])
    os.environ["DRAGNET_DENSE"] = "1"
    try:
        g_dense = GpuEngine().scan([str(f)], [q])
    finally:
        os.environ["DRAGNET_DENSE"] = "0"
    try:
        g_hash = GpuEngine().scan([str(f)], [q])
    finally:
        os.environ.pop("DRAGNET_DENSE", None)
    assert_same(c, g_dense)
    assert_same(c, g_hash)


def test_dense_overflow_falls_back(engines, tmp_path):
    """Cardinality beyond the dense directory restarts on the hash
    path and still matches the oracle."""
    import json as _json

    cpu, gpu = engines
    lines = [_json.dumps({"u": "id%06d" % i}) for i in range(30000)]
    f = tmp_path / "hi_card.log"
    f.write_bytes(("\n".join(lines) + "\n").encode())
    q = queries_from_case(["-b", "u"])
    c = cpu.scan([str(f)], [q])
    g = gpu.scan([str(f)], [q])
    assert_same(c, g)


def test_xpose_device_vs_host_staging(engines, tmp_path):
    """The device-side transposer (xpose_build_kernel)
    must aggregate identically to the r1 host numpy builder."""
    cpu, gpu = engines
    from dragnet_amd.engine import plan as planmod
    from dragnet_amd.engine.gpu import _ScanContext
    from dragnet_amd.query import query_load
    from dragnet_amd.tools.mktestdata import generate_lines
    lines = list(generate_lines(25000, seed=123))
    lines += [b"junk line\n", b'{"m": "' + b"y" * 700 + b'"}\n']
    pool = b"".join(lines)
    q = query_load(filter={"eq": ["req.method", "GET"]},
                   breakdown_specs="req.method,res.statusCode")
    path = tmp_path / "xpd.ndjson"
    path.write_bytes(pool)
    expected = cpu.scan([str(path)], [q]).aggregators[0].points()

    results = {}
    for mode in ("device", "host"):
        if mode == "host":
            os.environ["DRAGNET_XPOSE_HOST"] = "1"
        try:
            cplan = planmod.compile_plan([q])
            ctx = _ScanContext(gpu, cplan, agg_slots=1 << 15,
                               dict_slots=1 << 15,
                               dict_data_cap=8 << 20)
            ctx.stage_xpose(pool)
            ctx.reset()
            ctx.scan_xpose()
            aggs, stages = ctx.finalize([q])
            results[mode] = (aggs[0].points(),
                             dict(stages)["json parser"])
        finally:
            os.environ.pop("DRAGNET_XPOSE_HOST", None)
    assert results["device"][0] == expected
    assert results["device"] == results["host"]


@pytest.mark.gpu
def test_dense_slots_knob(engines, tmp_path):
    """DRAGNET_DENSE_SLOTS (the documented high-cardinality knob from
    the r2 slot sweep) must not change results — 16384- and
    32768-slot directories equal the default and the oracle."""
    import json as _json
    import random

    from dragnet_amd.engine.gpu import GpuEngine
    from dragnet_amd.query import query_load
    cpu, _ = engines

    rng = random.Random(11)
    lines = []
    for i in range(60000):
        rec = {"a": "k%d" % rng.randint(0, 200),
               "b": rng.randint(0, 30)}
        lines.append(_json.dumps(rec))
    f = tmp_path / "slots.log"
    f.write_bytes(("\n".join(lines) + "\n").encode())

    q = query_load(breakdown_specs="a,b")
    c = cpu.scan([str(f)], [q])
    for slots in ("16384", "32768"):
        os.environ["DRAGNET_DENSE_SLOTS"] = slots
        try:
            g = GpuEngine().scan([str(f)], [q])
        finally:
            os.environ.pop("DRAGNET_DENSE_SLOTS", None)
        assert_same(c, g)
