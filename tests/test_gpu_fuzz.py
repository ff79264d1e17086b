"""
Differential fuzzing: random JSON records + random queries, GPU kernel
vs CPU oracle (pytest -m gpu).

Stays inside the documented parity envelope: numbers <= 15
significant digits, ASCII keys, nesting within the validated depth.
Everything else is fair game: missing fields, nulls, type mixes,
malformed/truncated lines, empty lines, duplicate keys, literal
dotted keys (companion slots), unicode values in BOTH canonical and
backslash-u-escaped spellings (incl. surrogate pairs), deep nesting.
"""

import json
import random

import pytest

pytestmark = pytest.mark.gpu

KEYS = ["a", "b", "c", "req", "res", "xy", "time", "lat", "msg",
        "a.b", "req.a"]  # literal dotted keys: companion-slot path
STRINGS = ["GET", "PUT", "", "hello world", "héllo", "line\nbreak",
           'quo"te', "back\\slash", "tab\there", "200", "1e3", "0x10",
           "  12 ", "Infinity", "naan", "ünïcødé-αβγ", "a" * 120,
           "\U0001F600ok", "x\u00e9y"]


def rand_value(rng, depth):
    r = rng.random()
    if depth < 6 and r < 0.15:
        return {rng.choice(KEYS): rand_value(rng, depth + 1)
                for _ in range(rng.randrange(3))}
    if depth < 6 and r < 0.25:
        return [rand_value(rng, depth + 1)
                for _ in range(rng.randrange(3))]
    if r < 0.40:
        return rng.choice(STRINGS)
    if r < 0.55:
        return rng.randrange(-10000, 10000)
    if r < 0.70:
        return round(rng.uniform(-1e6, 1e6), 6)
    if r < 0.78:
        return rng.choice([1e-30, 2.5e3, 0.125, 1e15, -0.0])
    if r < 0.86:
        return rng.choice([True, False])
    if r < 0.94:
        return None
    return rng.choice(["2014-05-01T12:34:56.789Z", "2014-05-01",
                       "not a date", "2014-13-99"])


def rand_record(rng):
    rec = {}
    for _ in range(rng.randrange(1, 7)):
        rec[rng.choice(KEYS)] = rand_value(rng, 0)
    return rec


def rand_line(rng):
    r = rng.random()
    if r < 0.55:
        return json.dumps(rec_or_scalar(rng),
                          ensure_ascii=False).encode()
    if r < 0.85:
        # backslash-u-escaped spelling of the same values (keys stay
        # ASCII; escaped keys are the remaining documented divergence)
        return json.dumps(rec_or_scalar(rng),
                          ensure_ascii=True).encode()
    if r < 0.90:
        return b""  # empty line
    if r < 0.95:
        good = json.dumps(rand_record(rng)).encode()
        return good[:rng.randrange(len(good))]  # truncated
    return bytes(rng.randrange(33, 126)
                 for _ in range(rng.randrange(1, 30)))  # garbage


def rec_or_scalar(rng):
    if rng.random() < 0.9:
        return rand_record(rng)
    return rand_value(rng, 0)


def rand_query(rng):
    from dragnet_amd.query import query_load
    nbd = rng.randrange(0, 4)
    parts = []
    used = set()
    for _ in range(nbd):
        k = rng.choice(KEYS + ["req.a", "res.b", "a.b"])
        if k in used:
            continue  # duplicate names with mixed attrs: undefined
        used.add(k)
        r = rng.random()
        if r < 0.2:
            parts.append(k + "[aggr=quantize]")
        elif r < 0.35:
            parts.append(k + "[aggr=lquantize,step=%d]"
                         % rng.choice([10, 100, 1000]))
        elif r < 0.45 and k in ("time",):
            parts.append(k + "[date]")
        else:
            parts.append(k)
    filt = None
    if rng.random() < 0.6:
        op = rng.choice(["eq", "ne", "lt", "le", "gt", "ge"])
        k = rng.choice(KEYS + ["req.a"])
        v = rng.choice(["GET", "200", 200, 0, True, None, "héllo",
                        -5.5, "x", "\U0001F600ok"])
        filt = {op: [k, v]}
        if rng.random() < 0.3:
            filt = {rng.choice(["and", "or"]):
                    [filt, {"eq": [rng.choice(KEYS), rng.choice(
                        ["PUT", 1, None])]}]}
    return query_load(filter=filt,
                      breakdown_specs=",".join(parts) or None)


@pytest.mark.parametrize("seed", range(8))
def test_fuzz_differential(seed, tmp_path):
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.engine.gpu import GpuEngine

    rng = random.Random(1000 + seed)
    lines = [rand_line(rng) for _ in range(2000)]
    path = tmp_path / ("fuzz%d.ndjson" % seed)
    path.write_bytes(b"\n".join(lines) + b"\n")

    cpu, gpu = CpuEngine(), GpuEngine()
    for qi in range(6):
        q = rand_query(rng)
        c = cpu.scan([str(path)], [q])
        g = gpu.scan([str(path)], [q])
        desc = (seed, qi, q.filter, [b["name"] for b in q.breakdowns])
        assert g.aggregators[0].points() == c.aggregators[0].points(), \
            desc
        cs = dict(c.stages)["json parser"]
        gs = dict(g.stages)["json parser"]
        assert gs == cs, desc


def test_fuzz_json_skinner_weighted(tmp_path):
    """Weighted json-skinner re-aggregation fuzz: random POINT streams
    (weights, literal dotted keys, junk lines) through both engines —
    the distributed-reduce path (reference reduce phase consumes
    exactly this format)."""
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.engine.gpu import GpuEngine
    cpu, gpu = CpuEngine(), GpuEngine()
    for seed in (11, 12, 13, 14):
        rng = random.Random(seed)
        lines = []
        for i in range(3000):
            r = rng.random()
            if r < 0.75:
                fields = {}
                for _ in range(rng.randrange(1, 5)):
                    fields[rng.choice(KEYS)] = rng.choice(
                        ["GET", "PUT", 200, 404, "h%d" % (i % 9),
                         None, 2 ** (i % 12), "26"])
                w = rng.choice([1, 2, 3, 5, 0.5, 1.25])
                lines.append(json.dumps(
                    {"fields": fields, "value": w}).encode())
            elif r < 0.85:
                # invalid points: missing fields/value, bad types
                lines.append(rng.choice([
                    b'{"fields": {"a": 1}}',
                    b'{"value": 3}',
                    b'{"fields": {"a": 1}, "value": "x"}',
                    b'{"fields": {"a": 1}, "value": true}',
                    b'not json',
                    b'42',
                ]))
            else:
                lines.append(json.dumps(
                    {"fields": {"a.b": rng.randrange(5)},
                     "value": rng.randrange(1, 4)}).encode())
        path = tmp_path / ("sk%d.ndjson" % seed)
        path.write_bytes(b"\n".join(lines) + b"\n")
        for qi in range(3):
            q = rand_query(rng)
            c = cpu.scan([str(path)], [q], data_format="json-skinner")
            g = gpu.scan([str(path)], [q], data_format="json-skinner")
            assert g.aggregators[0].points() == \
                c.aggregators[0].points(), (seed, qi, q.filter)
            assert dict(g.stages)["json parser"] == \
                dict(c.stages)["json parser"], (seed, qi)
