"""
Library facade — the programmatic interface mirroring the reference's
lib/dragnet.js exports (queryLoad, build, indexConfig, indexScan,
indexRead, datasourceForConfig, datasourceForName).
"""

from . import config as mod_config
from .datasource import datasource_for_config
from .query import QueryConfig, QueryError, query_load  # noqa: F401


def datasource_for_name(dsname, cfg=None, engine=None):
    """Instantiate the datasource backend registered under dsname
    (reference lib/dragnet.js:260-304)."""
    cfg = cfg or mod_config.load_config()
    ds = cfg.datasource_get(dsname)
    if ds is None:
        raise mod_config.ConfigError(
            'datasource "%s" does not exist' % dsname)
    return datasource_for_config(ds, engine=engine)


def scan(dsname, filter=None, breakdowns=None, time_after=None,
         time_before=None, cfg=None, engine=None):
    """Scan raw data; returns the list of aggregated points."""
    backend = datasource_for_name(dsname, cfg=cfg, engine=engine)
    q = query_load(filter=filter, breakdown_specs=breakdowns,
                   time_after=time_after, time_before=time_before)
    result = backend.scan(q)
    if result is None or getattr(result, "nonroot", False):
        return None
    return result.aggregators[0].points()


def build(dsname, interval="day", time_after=None, time_before=None,
          cfg=None, engine=None):
    """Materialize indexes for the datasource's configured metrics;
    returns the index files written."""
    cfg = cfg or mod_config.load_config()
    backend = datasource_for_name(dsname, cfg=cfg, engine=engine)
    metrics = [m.serialize(skip_datasource=True)
               for m in cfg.datasource_metrics(dsname)]
    from .query import parse_fields, parse_time_bounds
    for m in metrics:
        m["breakdowns"] = parse_fields(m.get("breakdowns", []))
    after_ms, before_ms = parse_time_bounds(time_after, time_before)
    return backend.build(metrics, interval=interval,
                         after_ms=after_ms, before_ms=before_ms)


def query(dsname, filter=None, breakdowns=None, interval="day",
          time_after=None, time_before=None, cfg=None, engine=None):
    """Answer a query from the datasource's indexes; returns points."""
    backend = datasource_for_name(dsname, cfg=cfg, engine=engine)
    q = query_load(filter=filter, breakdown_specs=breakdowns,
                   time_after=time_after, time_before=time_before)
    result = backend.query(q, interval=interval)
    if result is None or getattr(result, "nonroot", False):
        return None
    return result.aggregators[0].points()


def index_config(dsname, cfg=None):
    """Serialized metric configuration for the datasource."""
    cfg = cfg or mod_config.load_config()
    return {"metrics": [m.serialize(skip_datasource=True)
                        for m in cfg.datasource_metrics(dsname)]}


def check_conservation(stages):
    """Count-conservation audit over pipeline counter stages: every
    stage's inputs must equal its outputs plus its attributed drops
    (the integrity invariant SURVEY.md §5 calls for in place of
    sanitizers).  Returns a list of violation strings (empty = OK)."""
    problems = []
    drop_keys = ("invalid json", "nfilteredout", "nfailedeval",
                 "undef", "baddate", "nonnumeric")
    for name, counters in stages:
        if "ninputs" not in counters or "noutputs" not in counters:
            continue
        if name in ("Aggregator", "Flattener",
                    "Index Result Aggregator") \
                or name.startswith("Find"):
            continue  # aggregation compresses; finders emit subsets
        drops = sum(counters.get(k, 0) for k in drop_keys)
        if counters["ninputs"] != counters["noutputs"] + drops:
            problems.append(
                "%s: ninputs %d != noutputs %d + drops %d"
                % (name, counters["ninputs"], counters["noutputs"],
                   drops))
    return problems
