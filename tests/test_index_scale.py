"""Index subsystem at the 1M-aggregate scale: write throughput stays
in seconds (pragma synchronous=off + executemany batching,
reference lib/index-sink.js:169-178), the file round-trips through
IndexQuerier, and SUM(value) is conserved."""

import os
import random

from dragnet_amd.index.query import IndexQuerier
from dragnet_amd.index.sink import IndexSink
from dragnet_amd.query import query_load

NROWS = 1_000_000


def test_million_row_index(tmp_path):
    path = str(tmp_path / "big.sqlite")
    metric = {"name": "m0", "filter": None,
              "breakdowns": [
                  {"name": "k", "field": "k", "aggr": None,
                   "step": None, "date": False},
                  {"name": "v", "field": "v", "aggr": "lquantize",
                   "step": 10, "date": False}]}
    sink = IndexSink(path, [metric])
    rng = random.Random(7)
    for _ in range(NROWS):
        sink.write_point({"fields": {"__dn_metric": 0,
                                     "k": "key%05d"
                                          % rng.randrange(90000),
                                     "v": rng.randrange(0, 500)},
                          "value": 1})
    sink.flush()
    assert os.path.exists(path)
    assert sink.nwritten == NROWS

    q = query_load(filter={"eq": ["k", "key00042"]},
                   breakdown_specs="k")
    iq = IndexQuerier(path)
    m = iq.find_metric(q)
    assert m is not None
    agg = iq.run(q)
    pts = agg.points()
    assert len(pts) == 1 and pts[0]["fields"] == {"k": "key00042"}

    # conservation: grouping only by k sums every row's value
    q2 = query_load(breakdown_specs="k")
    agg2 = IndexQuerier(path).run(q2)
    assert sum(p["value"] for p in agg2.points()) == NROWS
