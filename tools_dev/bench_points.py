"""Reproduce the point-codec numbers quoted in docs/ROADMAP.md:
C++ tagged-point reduce vs the pure-Python cmd_index_read loop, and
C++ serialize vs per-point json.dumps.  CPU-only."""
import io
import json
import random
import sys
import time

sys.path.insert(0, "/root/repo")

from dragnet_amd.datasource.file import metric_query  # noqa: E402
from dragnet_amd.index import _points  # noqa: E402
from dragnet_amd.output import point_json  # noqa: E402
from dragnet_amd.points import (Aggregator,  # noqa: E402
                                reduce_tagged_stream)

N = int(sys.argv[1]) if len(sys.argv) > 1 else 1_000_000
metric = {"name": "reqs", "breakdowns": [
    {"name": "req.method"}, {"name": "res.statusCode"},
    {"name": "latency", "aggr": "quantize"}]}
q = metric_query(metric, "day", "time")
rng = random.Random(3)
methods = ["GET", "PUT", "DELETE", "HEAD", "POST"]
pts = [{"fields": {"__dn_metric": 0,
                   "__dn_ts": 1400000000 + (i % 5) * 86400,
                   "req.method": rng.choice(methods),
                   "res.statusCode": rng.choice([200, 204, 404, 500]),
                   "latency": rng.choice([1, 2, 4, 8, 16, 32, 129])},
        "value": rng.randint(1, 50)} for i in range(N)]
data = b"".join(json.dumps(p, separators=(",", ":")).encode() + b"\n"
                for p in pts)
print("%d points, %.1f MB" % (N, len(data) / 1e6))

t0 = time.time()
aggs = [Aggregator(q)]
for line in data.split(b"\n"):
    line = line.strip()
    if not line:
        continue
    try:
        p = json.loads(line)
    except ValueError:
        continue
    mi = p.get("fields", {}).get("__dn_metric")
    if not isinstance(mi, int) or not (0 <= mi < 1):
        continue
    aggs[mi].write(p)
t_py = time.time() - t0
print("python reduce: %.2fs  %.0f k pts/s" % (t_py, N / t_py / 1e3))

t0 = time.time()
fast = [Aggregator(q)]
punted = reduce_tagged_stream(io.BytesIO(data), fast, [q])
t_c = time.time() - t0
print("c++ reduce:    %.2fs  %.0f k pts/s  (%.1fx, punted=%d)"
      % (t_c, N / t_c / 1e3, t_py / t_c, len(punted)))
assert fast[0].table == aggs[0].table

t0 = time.time()
blob_py = "".join(point_json(p) + "\n" for p in pts)
t_py = time.time() - t0
print("python emit:   %.2fs  %.0f k pts/s" % (t_py, N / t_py / 1e3))

t0 = time.time()
blob_c = _points.serialize_points(pts, point_json)
t_c = time.time() - t0
print("c++ emit:      %.2fs  %.0f k pts/s  (%.1fx)"
      % (t_c, N / t_c / 1e3, t_py / t_c))
assert blob_c == blob_py.encode()
print("byte-exact")
