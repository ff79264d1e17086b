"""
Deterministic synthetic muskie-like NDJSON generator.

Same record shape as the reference generator (reference
tools/mktestdata:41-100) — linearly increasing timestamps, correlated
method->operation fields, nullable/omittable nested req.caller,
piecewise-uniform latency distribution, `latency` as a STRING-typed
number (dataLatency is the numeric twin) — but seeded and deterministic
so fixtures and golden outputs are reproducible.  This record shape is
the benchmark workload named in BASELINE.json.
"""

import json
import os
import random

from .. import jsdate

HOSTS = ["ralph", "janey", "kearney", "sherri", "wendell"]
METHODS = ["HEAD", "GET", "PUT", "DELETE"]
OPERATIONS = {
    "HEAD": ["headstorage", "headpublicstorage"],
    "GET": ["getjoberrors", "getpublicstorage", "getstorage"],
    "PUT": ["putdirectory", "putpublicobject", "putobject"],
    "DELETE": ["deletestorage", "deletepublicstorage"],
}
CALLERS = ["admin", "poseidon", None, "__omit__"]
STATUS_CODES = [200, 204, 400, 404, 499, 500, 503]
LATENCY_DIST = [
    (0.4, 1, 5),
    (0.3, 20, 30),
    (0.1, 100, 200),
    (None, 1024, 4096),
]
NURLS = 500

DEFAULT_MIN = "2014-05-31T21:00:00Z"
DEFAULT_MAX = "2014-05-31T23:59:59Z"


def _probdist_value(rng, dist):
    r = rng.random()
    cm = 0.0
    for j in range(len(dist) - 1):
        cm += dist[j][0]
        if cm > r:
            break
    else:
        j = len(dist) - 1
    _, lo, hi = dist[j]
    return int(rng.random() * (hi - lo) + lo + 0.5)


def make_record(rng, j, nrecords, min_ms, max_ms,
                latency_as_string=True):
    ts_ms = round((j / nrecords) * (max_ms - min_ms) + min_ms)
    method = rng.choice(METHODS)
    operation = rng.choice(OPERATIONS[method])
    caller = rng.choice(CALLERS)
    req = {
        "method": method,
        "url": "/random/url/number/%d" % rng.randrange(NURLS),
    }
    if caller != "__omit__":
        req["caller"] = caller
    rec = {
        "time": jsdate.to_iso(ts_ms / 1000.0),
        "host": rng.choice(HOSTS),
        "req": req,
        "operation": operation,
        "res": {"statusCode": rng.choice(STATUS_CODES)},
        # mktestdata's latency is string-typed; the committed reference
        # FIXTURE has it numeric (SURVEY.md §7 hard-parts item 3) — the
        # fixture path passes latency_as_string=False
        "latency": (str(_probdist_value(rng, LATENCY_DIST))
                    if latency_as_string
                    else _probdist_value(rng, LATENCY_DIST)),
        "dataLatency": _probdist_value(rng, LATENCY_DIST),
        "dataSize": int(rng.random() * (1024 ** 3) + 0.5),
    }
    return rec


def generate_lines(nrecords, seed=1, min_time=DEFAULT_MIN,
                   max_time=DEFAULT_MAX):
    """Yield NDJSON lines (bytes, newline-terminated)."""
    rng = random.Random(seed)
    min_ms = jsdate.parse_ms(min_time)
    max_ms = jsdate.parse_ms(max_time)
    for j in range(nrecords):
        rec = make_record(rng, j, nrecords, min_ms, max_ms)
        yield (json.dumps(rec, separators=(",", ":")) + "\n").encode()


def make_fixture_tree(root, seed=1):
    """Write the standard test fixture: a %Y/%m-%d tree spanning
    2014-05-01..05, 2254 lines total: 2250 valid records + 2 invalid
    JSON lines + 1 bad date + 1 missing time field (mirroring the
    reference fixture's drop taxonomy).
    Deterministic for a given seed.  Returns the root."""
    rng = random.Random(seed)
    days = ["2014-05-%02d" % d for d in range(1, 6)]
    per_day = [250, 500, 500, 500, 500]
    nfiles = [1, 2, 2, 2, 2]
    total_written = 0
    for di, day in enumerate(days):
        day_ms = jsdate.parse_ms(day + "T00:00:00Z")
        dirname = os.path.join(root, "2014", day[5:])
        os.makedirs(dirname, exist_ok=True)
        n = per_day[di]
        files = nfiles[di]
        per_file = n // files
        idx = 0
        for fi in range(files):
            name = "one.log" if files == 1 else "f%d.log" % fi
            path = os.path.join(dirname, name)
            with open(path, "wb") as f:
                for j in range(per_file):
                    rec = make_record(
                        rng, idx, n, day_ms, day_ms + 86399000,
                        latency_as_string=False)
                    idx += 1
                    line = json.dumps(
                        rec, separators=(",", ":")) + "\n"
                    f.write(line.encode())
                    total_written += 1
                # inject failures at deterministic spots
                if day == "2014-05-02" and fi == 0:
                    f.write(b'{"this is not valid JSON\n')
                    f.write(b'[1, 2, oops]\n')
                if day == "2014-05-03" and fi == 0:
                    bad = make_record(rng, 0, n, day_ms, day_ms + 1000)
                    bad["time"] = "not-a-date"
                    f.write((json.dumps(bad, separators=(",", ":"))
                             + "\n").encode())
                if day == "2014-05-04" and fi == 0:
                    bad = make_record(rng, 0, n, day_ms, day_ms + 1000)
                    del bad["time"]
                    f.write((json.dumps(bad, separators=(",", ":"))
                             + "\n").encode())
    return root


def main():
    import sys
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 1000
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 1
    out = sys.stdout.buffer
    for line in generate_lines(n, seed=seed):
        out.write(line)


if __name__ == "__main__":
    main()
