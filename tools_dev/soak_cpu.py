"""CPU-side deep soak battery (no GPU): seeded adversarial sweeps of
the four CPU-trust surfaces.  Found (and now regression-guards) the
date-column non-scalar coercion crash in points.canonical.

    python tools_dev/soak_cpu.py [seeds-multiplier]

Sections:
  1. codec differential  — C++ tagged-point reducer vs the Python loop
  2. pipeline conservation — Σ drops + outputs == inputs at every stage
  3. merge wire format   — encode/rebuild round-trip of the RCCL merge
  4. CLI argv fuzz       — random argv must error cleanly, never raise
  5. skinner conservation — the weighted json-skinner pipeline
  6. config CRUD fuzz    — registry stays valid/atomic under random ops
  7. index round-trip    — write_index -> IndexQuerier == direct table
"""
import io
import json
import os
import random
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

MULT = float(sys.argv[1]) if len(sys.argv) > 1 else 1.0


def n(base):
    return max(1, int(base * MULT))


def soak_codec():
    os.environ.pop("DRAGNET_PY_POINTS", None)
    from test_points_fast import QUERIES, check, tag
    vals = ["s", "x", "", None, True, False, 0, 7, -3, 2**40, 2**64,
            0.5, -1.25, 2.75, "26", " 7 ", "NaN", "café", "0x1f",
            "1e999", "Infinity", "-0x2", [1], {"z": 1}, "a\nb", "\x7f",
            10**25, "  ", "+5", "-", ".5", "5.", "1e", "😀", 1e300,
            -0.0, [None, [2]]]
    bad = 0
    for seed in range(n(300)):
        rng = random.Random(seed)
        lines = []
        for _ in range(150):
            mi = rng.choice([0, 1, 2, 3, -2, "x"])
            fields = {name: rng.choice(vals)
                      for name in ("a", "t", "lat", "ts", "req.method",
                                   "res.statusCode", "junk", "ünïcode")
                      if rng.random() < 0.55}
            value = rng.choice([1, 2, 5, 0.5, 0.25, 3, 1024])
            if isinstance(mi, str):
                lines.append(json.dumps(
                    {"fields": dict(fields), "value": value}).encode())
            else:
                lines.append(tag(mi, fields, value))
            if rng.random() < 0.12:
                lines.append(rng.choice(
                    [b"", b"garbage", b'{"value":1}', b"   ",
                     b'{"fields":{},"value":2}']))
        keep = []
        for ln in lines:
            try:
                v = json.loads(ln.strip() or b"{}")
                if not isinstance(v, dict):
                    continue
                if not isinstance(v.get("fields", {}), dict):
                    continue
                mi = v.get("fields", {}).get("__dn_metric")
                if isinstance(mi, int) and 0 <= mi < len(QUERIES):
                    if "value" not in v or \
                            isinstance(v.get("value"), str):
                        continue
            except ValueError:
                pass
            keep.append(ln)
        try:
            check(keep)
        except AssertionError as e:
            bad += 1
            print("codec SEED %d DIVERGED: %s" % (seed, str(e)[:200]))
    return bad


def soak_pipeline():
    from dragnet_amd.query import query_load
    from dragnet_amd.scan_cpu import ScanPipeline
    QS = [{}, {"breakdown_specs": "a"}, {"breakdown_specs": "req.m,b"},
          {"breakdown_specs": "lat[aggr=quantize]"},
          {"breakdown_specs": "lat[aggr=lquantize,step=7]"},
          {"filter": {"eq": ["a", 1]}},
          {"filter": {"or": [{"lt": ["lat", 100]},
                             {"ne": ["b", "x"]}]},
           "breakdown_specs": "b,a"},
          {"breakdown_specs":
           "t[date,field=time,aggr=lquantize,step=60]"},
          {"breakdown_specs": "t[date,field=time],a"}]
    DROPS = ("invalid json", "nfilteredout", "nfailedeval", "undef",
             "baddate", "nonnumeric")
    VAL = [None, True, False, 0, 1, -5, 2**60, 0.5, "x", "", "26",
           "2014-05-01T00:00:00.000Z", "2014-13-99T99:99:99Z",
           [1, {"a": 2}], {"m": "GET", "deep": {"x": 1}}, "NaN",
           1e308, -0.0, "😀" * 3, "a" * 50]
    KEYS = ["a", "b", "req", "time", "lat", "x.y", "req.m", "", "t",
            "\x7f"]
    bad = 0
    for seed in range(n(250)):
        rng = random.Random(seed)
        lines = []
        for _ in range(60):
            c = rng.random()
            if c < 0.55:
                rec = {rng.choice(KEYS): rng.choice(VAL)
                       for _ in range(rng.randint(0, 5))}
                lines.append(json.dumps(rec).encode())
            elif c < 0.8:
                lines.append(bytes(
                    rng.randrange(1, 256)
                    for _ in range(rng.randint(0, 30)))
                    .replace(b"\n", b"x"))
            else:
                lines.append(b"")
        data = b"".join(ln + b"\n" for ln in lines)
        for qkw in QS:
            p = ScanPipeline(query_load(**qkw))
            p.write_bytes(data)
            p.finish()
            stages = p.counter_stages()
            prev = None
            for name, cnt in stages:
                if name == "Aggregator":
                    continue
                drops = sum(cnt.get(k, 0) for k in DROPS)
                if cnt["ninputs"] != cnt["noutputs"] + drops:
                    bad += 1
                    print("pipeline SEED %d %r %s %r"
                          % (seed, qkw, name, cnt))
                prev = cnt["noutputs"]
            if dict(stages)["Aggregator"]["ninputs"] != prev:
                bad += 1
                print("pipeline SEED %d %r chain break" % (seed, qkw))
    return bad


def soak_wire():
    from dragnet_amd.distributed import _encode_table, _rebuild_table
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    bad = 0
    for seed in range(n(400)):
        rng = random.Random(seed)
        ncols = rng.randint(1, 4)
        q = query_load(breakdown_specs=",".join(
            "c%d" % i for i in range(ncols)))
        agg = Aggregator(q)
        for _ in range(rng.randint(0, 60)):
            key = tuple(rng.choice([
                rng.choice(["", "x", "a" * 40, "ünïcode😀", "null",
                            "undefined", "[object Object]", "1,",
                            "\x7fctl"]),
                rng.randint(-2**45, 2**45),
                rng.randint(-100, 100)]) for _ in range(ncols))
            v = rng.choice([1, 7, 10**14, 0.5, 2.25, 1e12 + 0.25])
            agg.table[key] = agg.table.get(key, 0) + v
        codes, tags, vals, strings = _encode_table(agg, ncols)
        out = _rebuild_table(q, codes, tags, vals, strings)
        want = {k: (int(v) if float(v).is_integer() else float(v))
                for k, v in agg.table.items()}
        if out.table != want:
            bad += 1
            print("wire SEED %d mismatch" % seed)
    return bad


def soak_argv():
    os.environ["DRAGNET_CONFIG"] = tempfile.mktemp()
    os.environ["DRAGNET_ENGINE"] = "cpu"
    from dragnet_amd import cli
    words = ["scan", "query", "build", "datasource-add", "metric-add",
             "index-read", "index-scan", "datasource-list", "-b", "-f",
             "--filter", "--breakdowns", "--path", "--after",
             "--before", "-t", "-n", "-i", "-A", "-B", "--points",
             "--raw", "--gnuplot", "--counters", "--interval", "hour",
             "junk", "", "--", "{bad json", '{"eq":["a",1]}',
             "a[aggr=quantize]", "a[", "x,y,z", "/nonexistent", "-v",
             "--index-config", "%Y/%m", "--time-format",
             "--time-field", "2014-05-01", "99999", "--warnings",
             "--dry-run", "src", "a[date]", "a[aggr=bad]"]
    bad = 0
    for seed in range(n(600)):
        rng = random.Random(seed)
        argv = [rng.choice(words) for _ in range(rng.randint(0, 7))]
        old_out, old_err = sys.stdout, sys.stderr
        sys.stdout = io.StringIO()
        sys.stderr = io.StringIO()
        try:
            code = cli.main(list(argv))
            assert isinstance(code, int)
        except SystemExit:
            pass
        except Exception as e:
            bad += 1
            old_err.write("argv SEED %d %r -> %r\n" % (seed, argv, e))
        finally:
            sys.stdout, sys.stderr = old_out, old_err
    return bad


def soak_skinner():
    from dragnet_amd.query import query_load
    from dragnet_amd.scan_cpu import ScanPipeline
    DROPS = ("invalid json", "nfilteredout", "nfailedeval", "undef",
             "baddate", "nonnumeric")
    QS = [{}, {"breakdown_specs": "a"},
          {"breakdown_specs": "lat[aggr=quantize]"},
          {"filter": {"eq": ["a", "x"]}, "breakdown_specs": "a,b"}]
    bad = 0
    for seed in range(n(200)):
        rng = random.Random(seed)
        lines = []
        for _ in range(50):
            c = rng.random()
            if c < 0.6:
                p = {"fields": {k: rng.choice(
                        ["x", "y", 3, None, "26", [1]])
                        for k in ("a", "b", "lat")
                        if rng.random() < 0.7},
                     "value": rng.choice([1, 2, 0.5, -1, 10**9])}
                lines.append(json.dumps(p).encode())
            elif c < 0.8:
                lines.append(json.dumps(
                    {"notfields": 1, "value": "x"}).encode())
            else:
                lines.append(rng.choice([b"", b"junk {", b"[1]"]))
        data = b"".join(ln + b"\n" for ln in lines)
        for qkw in QS:
            p = ScanPipeline(query_load(**qkw),
                             data_format="json-skinner")
            p.write_bytes(data)
            p.finish()
            stages = p.counter_stages()
            prev = None
            for name, cnt in stages:
                if name == "Aggregator":
                    continue
                drops = sum(cnt.get(k, 0) for k in DROPS)
                if cnt["ninputs"] != cnt["noutputs"] + drops:
                    bad += 1
                    print("skinner SEED %d %r %s" % (seed, qkw, name))
                prev = cnt["noutputs"]
            if dict(stages)["Aggregator"]["ninputs"] != prev:
                bad += 1
                print("skinner SEED %d %r chain" % (seed, qkw))
    return bad


def soak_config():
    bad = 0
    for seed in range(n(150)):
        cfgfile = tempfile.mktemp()
        os.environ["DRAGNET_CONFIG"] = cfgfile
        os.environ["DRAGNET_ENGINE"] = "cpu"
        from dragnet_amd import cli
        rng = random.Random(seed)
        names = ["a", "b", "wëird", "x-y", "s" * 30]
        live = set()
        for step in range(25):
            op = rng.random()
            name = rng.choice(names)
            o, e = sys.stdout, sys.stderr
            sys.stdout, sys.stderr = io.StringIO(), io.StringIO()
            try:
                if op < 0.35:
                    if cli.main(["datasource-add", name,
                                 "--path=/tmp/x%d"
                                 % rng.randint(0, 3)]) == 0:
                        live.add(name)
                elif op < 0.5:
                    if cli.main(["datasource-remove", name]) == 0:
                        live.discard(name)
                elif op < 0.65:
                    cli.main(["datasource-update", name,
                              "--time-field=t%d" % rng.randint(0, 2)])
                elif op < 0.8:
                    cli.main(["metric-add", name,
                              "m%d" % rng.randint(0, 2), "-b",
                              rng.choice(["a", "lat[aggr=quantize]",
                                          "bad["])])
                else:
                    cli.main(["metric-remove", name,
                              "m%d" % rng.randint(0, 2)])
            except Exception as ex:
                bad += 1
                e.write("config SEED %d: %r\n" % (seed, ex))
            finally:
                sys.stdout, sys.stderr = o, e
            if os.path.exists(cfgfile):
                try:
                    json.load(open(cfgfile))
                except ValueError:
                    bad += 1
                    print("config SEED %d: corrupt file" % seed)
            out = io.StringIO()
            o = sys.stdout
            sys.stdout = out
            try:
                assert cli.main(["datasource-list"]) == 0
            finally:
                sys.stdout = o
            listed = {ln.split()[0]
                      for ln in out.getvalue().splitlines()
                      if ln.strip() and not ln.startswith("DATASOURCE")}
            if listed != live:
                bad += 1
                print("config SEED %d: list mismatch" % seed)
                break
        if os.path.exists(cfgfile):
            os.unlink(cfgfile)
    return bad


def soak_index():
    from dragnet_amd.datasource.file import write_index
    from dragnet_amd.index.query import IndexQuerier
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    bad = 0
    for seed in range(n(150)):
        rng = random.Random(seed)
        tmp = tempfile.mkdtemp()
        nbd = rng.randint(1, 4)
        bds = []
        used = set()
        for i in range(nbd):
            name = rng.choice(["a", "b", "c", "lat", "t.s", "x-y"])
            if name in used:
                name += str(i)
            used.add(name)
            kind = rng.random()
            if kind < 0.3:
                bds.append({"name": name, "aggr": "quantize"})
            elif kind < 0.5:
                bds.append({"name": name, "aggr": "lquantize",
                            "step": rng.choice([7, 60, 3600])})
            else:
                bds.append({"name": name})
        metrics = [{"name": "m",
                    "breakdowns": [dict(b) for b in bds]}]
        q = query_load(breakdown_specs=",".join(
            "%s[aggr=%s%s]" % (b["name"], b["aggr"],
                               ",step=%d" % b["step"]
                               if b["aggr"] == "lquantize" else "")
            if "aggr" in b else b["name"] for b in bds))
        src = Aggregator(q)
        for _ in range(rng.randint(1, 120)):
            fields = {}
            for b in bds:
                if "aggr" in b:
                    fields[b["name"]] = rng.choice(
                        [0, 1, 7, 26, 100, 4096, "26", 2.5])
                else:
                    fields[b["name"]] = rng.choice(
                        ["x", "y", "", "z-9", None, 200, True])
            src.write({"fields": dict(fields),
                       "value": rng.randint(1, 9)})
        pts = src.points()
        for p in pts:
            p["fields"]["__dn_metric"] = 0
        write_index(tmp, metrics, "all", pts)
        iq = IndexQuerier(os.path.join(tmp, "all"))
        if iq.run(q).table != src.table:
            bad += 1
            print("index SEED %d mismatch" % seed)
    return bad


def main():
    total = 0
    for name, fn in (("codec", soak_codec),
                     ("pipeline", soak_pipeline),
                     ("wire", soak_wire), ("argv", soak_argv),
                     ("skinner", soak_skinner),
                     ("config", soak_config),
                     ("index", soak_index)):
        bad = fn()
        print("%s: %s" % (name, "CLEAN" if bad == 0
                          else "%d FAILURES" % bad))
        total += bad
    return 1 if total else 0


if __name__ == "__main__":
    sys.exit(main())
