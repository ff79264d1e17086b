"""
`dn` — the dragnet_amd command-line interface.

Fourteen subcommands matching the reference CLI surface
(reference bin/dn:34-49, share/usage.txt): datasource CRUD, metric CRUD,
scan, build, query, index-config, index-scan, index-read.  Output
formatters: pretty table, DTrace-style histogram, gnuplot, raw JSON,
points (reference bin/dn:941-1274).
"""

import json
import os

import sys
import time as _time

# "require phase" timing (-t; reference bin/dn:8,24,80-83): time spent
# importing the engine modules below, reported separately from total
_t_require0 = _time.monotonic()

from . import config as mod_config
from . import krill
from . import output as mod_output
from .attrs import AttrsError, attrs_parse
from .datasource import datasource_for_config
from .query import QueryConfig, QueryError

_t_require = _time.monotonic() - _t_require0

USAGE = """\
usage: dn SUBCOMMAND [OPTIONS] ARGS

dn datasource-add    [--backend=file|sharded|manta] --path=DATA_PATH
                     [--index-path=INDEX_PATH] [--filter=FILTER]
                     [--time-field=FIELD] [--time-format=TIME_FORMAT]
                     [--data-format=json|json-skinner] [--shards=N]
                     DATASOURCE
dn datasource-update [same options] DATASOURCE
dn datasource-list   [-v]
dn datasource-remove DATASOURCE
dn datasource-show   [-v] DATASOURCE

dn metric-add        [--breakdowns=BREAKDOWN[,...]] [--filter=FILTER]
                     DATASOURCE METRIC
dn metric-list       [-v] DATASOURCE
dn metric-remove     DATASOURCE METRIC

dn build             [--before=START_TIME] [--after=END_TIME]
                     [--interval=hour|day|all] [--index-config=CONFIG_FILE]
                     [--dry-run] DATASOURCE

dn query             [--before=START_TIME] [--after=END_TIME]
                     [--filter=FILTER] [--breakdowns=BREAKDOWN[,...]]
                     [--interval=hour|day|all] [--raw] [--points]
                     [--counters] [--gnuplot] [--dry-run] DATASOURCE

dn scan              [--before=START_TIME] [--after=END_TIME]
                     [--filter=FILTER] [--breakdowns=BREAKDOWN[,...]]
                     [--raw] [--points] [--counters] [--warnings]
                     [--dry-run] DATASOURCE

dn index-config      DATASOURCE
dn index-read        [--index-config=INDEX_CONFIG_FILE]
                     [--interval=hour|day|all] DATASOURCE
dn index-scan        [--index-config=INDEX_CONFIG_FILE]
                     [--interval=hour|day|all]
                     [--before=START_TIME] [--after=END_TIME]
                     [--counters] DATASOURCE
"""


class UsageError(Exception):
    pass


class FatalError(Exception):
    pass


def main(argv=None):
    import time
    t_start = time.monotonic()
    argv = list(sys.argv[1:] if argv is None else argv)
    track_time = False
    if argv and argv[0] == "-t":
        # print phase timings on exit (reference bin/dn:80-83,1291-1296)
        track_time = True
        argv.pop(0)

    from .log import get_logger
    log = get_logger()
    log.debug("dispatch", argv=argv)
    rv = _dispatch(argv)
    log.debug("done", exit_status=rv)
    if track_time:
        total = time.monotonic() - t_start
        sys.stderr.write("timing stats:\n")
        sys.stderr.write("    require:  %.6fs\n" % _t_require)
        sys.stderr.write("    total:    %.6fs\n" % total)
    return rv


def _dispatch(argv):
    if not argv:
        return usage("no command specified", full=True)

    cmdname = argv[0]
    cmds = {
        "datasource-add": cmd_datasource_add,
        "datasource-list": cmd_datasource_list,
        "datasource-remove": cmd_datasource_remove,
        "datasource-update": cmd_datasource_update,
        "datasource-show": cmd_datasource_show,
        "metric-add": cmd_metric_add,
        "metric-list": cmd_metric_list,
        "metric-remove": cmd_metric_remove,
        "build": cmd_build,
        "index-config": cmd_index_config,
        "index-read": cmd_index_read,
        "index-scan": cmd_index_scan,
        "query": cmd_query,
        "scan": cmd_scan,
    }
    if cmdname not in cmds:
        return usage('no such command: "%s"' % cmdname)

    try:
        cmds[cmdname](argv[1:])
        return 0
    except UsageError as e:
        return usage(str(e))
    except (FatalError, QueryError, mod_config.ConfigError,
            krill.KrillError, _sink_error()) as e:
        sys.stderr.write("dn: %s\n" % e)
        return 1
    except BrokenPipeError:
        return 0


def _sink_error():
    from .index.sink import SinkError
    return SinkError


def usage(msg, full=False):
    """Error + short usage line (the reference goldens carry exactly
    these two lines on errors, tst.config.sh.out:7-9); the full
    option table prints only for a bare `dn` (full=True)."""
    sys.stderr.write("dn: %s\n" % msg)
    if full:
        sys.stderr.write(USAGE)
    else:
        sys.stderr.write("usage: dn SUBCOMMAND [OPTIONS] ARGS\n")
    return 2


# ---- option parsing (shared option table, per-command allow-list;
#      reference bin/dn:146-241) ----

_OPTDEFS = {
    "backend": ("--backend", True),
    "path": ("--path", True),
    "index-path": ("--index-path", True),
    "filter": ("--filter", True),
    "time-field": ("--time-field", True),
    "time-format": ("--time-format", True),
    "data-format": ("--data-format", True),
    "shards": ("--shards", True),
    "before": ("--before", True),
    "after": ("--after", True),
    "breakdowns": ("--breakdowns", True),
    "interval": ("--interval", True),
    "index-config": ("--index-config", True),
    "gnuplot": ("--gnuplot", False),
    "raw": ("--raw", False),
    "points": ("--points", False),
    "counters": ("--counters", False),
    "warnings": ("--warnings", False),
    "dry-run": ("--dry-run", False),
    "verbose": ("-v", False),
}

_SHORT = {"-f": "filter", "-b": "breakdowns", "-v": "verbose",
          "-i": "interval", "-A": "after", "-B": "before",
          "-n": "dry-run"}


def parse_args(argv, allowed):
    opts = {}
    args = []
    i = 0
    long_by_flag = {}
    for name in allowed:
        flag, hasval = _OPTDEFS[name]
        long_by_flag[flag] = (name, hasval)
    while i < len(argv):
        a = argv[i]
        if a == "--":
            args.extend(argv[i + 1:])
            break
        name = None
        hasval = False
        val = None
        if a.startswith("--"):
            body = a[2:]
            if "=" in body:
                flag, val = body.split("=", 1)
                flag = "--" + flag
            else:
                flag = a
            if flag not in long_by_flag:
                raise UsageError('unrecognized option: "%s"' % flag)
            name, hasval = long_by_flag[flag]
            if hasval and val is None:
                i += 1
                if i >= len(argv):
                    raise UsageError(
                        'option "%s" requires an argument' % flag)
                val = argv[i]
            if not hasval and val is not None:
                raise UsageError(
                    'option "%s" takes no argument' % flag)
        elif a.startswith("-") and len(a) > 1:
            if a not in _SHORT or _SHORT[a] not in allowed:
                raise UsageError('unrecognized option: "%s"' % a)
            name = _SHORT[a]
            _, hasval = _OPTDEFS[name]
            if hasval:
                i += 1
                if i >= len(argv):
                    raise UsageError(
                        'option "%s" requires an argument' % a)
                val = argv[i]
        else:
            args.append(a)
            i += 1
            continue
        if name == "breakdowns":
            # repeated -b accumulates as a LIST (reference dashdash
            # arrayOfString): each entry is attr-parsed separately so
            # errors quote only the offending value
            opts.setdefault(name, []).append(val)
        else:
            opts[name] = val if hasval else True
        i += 1
    return opts, args


def check_arg_count(args, n):
    if len(args) < n:
        raise UsageError("missing arguments")
    if len(args) > n:
        raise UsageError("extra arguments")


def _parse_filter(s):
    if s is None:
        return None
    try:
        f = json.loads(s)
    except ValueError as e:
        msg = str(e)
        if "Expecting" in msg and "char 0" not in msg:
            msg = "Unexpected end of input" \
                if "line" in msg and s.strip() in ("{", "[") else msg
        # a usage-class error: the reference routes JSON parse
        # failures through usage() (bin/dn dnParseArgs)
        raise UsageError("invalid filter: %s" %
                         ("Unexpected end of input"
                          if _is_truncated_json(s) else msg))
    # predicate semantics are validated where the query is built
    # (QueryConfig), so errors carry the reference's full chain:
    # "invalid query: invalid filter: predicate ...: unknown operator"
    return f


def _is_truncated_json(s):
    try:
        json.loads(s)
        return False
    except ValueError:
        depth = 0
        for c in s:
            if c in "{[":
                depth += 1
            elif c in "}]":
                depth -= 1
        return depth > 0


def _load_config():
    return mod_config.load_config()


def _save_config(cfg):
    mod_config.save_config(cfg)


def _get_datasource(cfg, name):
    ds = cfg.datasource_get(name)
    if ds is None:
        raise FatalError('datasource "%s" does not exist' % name)
    return ds




# ---- datasource commands ----

def _datasource_from_opts(name, opts, existing=None):
    backend = opts.get("backend", existing.backend if existing else "file")
    if backend not in mod_config.VALID_BACKENDS:
        raise FatalError('unsupported backend: "%s"' % backend)
    path = opts.get("path", existing.path if existing else None)
    if path is None:
        raise UsageError('"path" option is required')
    filt = opts.get("filter")
    if filt is not None:
        filt = _parse_filter(filt)
    elif existing is not None:
        filt = existing.filter
    # the data format is validated at USE time, not add time (the
    # reference stores it and `dn scan` fails with
    # 'unsupported format: "junk"' — tst.badargs.sh)
    data_format = opts.get(
        "data-format", existing.data_format if existing else "json")
    nshards = opts.get("shards", existing.nshards if existing else None)
    if nshards is not None:
        nshards = int(nshards)
    return mod_config.Datasource(
        name=name, backend=backend, path=path,
        index_path=opts.get(
            "index-path", existing.index_path if existing else None),
        filter=filt,
        time_field=opts.get(
            "time-field", existing.time_field if existing else None),
        time_format=opts.get(
            "time-format", existing.time_format if existing else None),
        data_format=data_format, nshards=nshards)


def cmd_datasource_add(argv):
    opts, args = parse_args(argv, [
        "backend", "path", "index-path", "filter", "time-field",
        "time-format", "data-format", "shards"])
    check_arg_count(args, 1)
    cfg = _load_config()
    cfg.datasource_add(_datasource_from_opts(args[0], opts))
    _save_config(cfg)


def cmd_datasource_update(argv):
    opts, args = parse_args(argv, [
        "backend", "path", "index-path", "filter", "time-field",
        "time-format", "data-format", "shards"])
    check_arg_count(args, 1)
    cfg = _load_config()
    existing = _get_datasource(cfg, args[0])
    cfg.datasource_update(_datasource_from_opts(args[0], opts, existing))
    _save_config(cfg)


def cmd_datasource_remove(argv):
    opts, args = parse_args(argv, [])
    check_arg_count(args, 1)
    cfg = _load_config()
    if cfg.datasource_get(args[0]) is None:
        raise FatalError('datasource "%s" does not exist' % args[0])
    cfg.datasource_remove(args[0])
    _save_config(cfg)


def _ds_location(ds):
    if ds.backend == "sharded":
        return "sharded:/" + (ds.path or "")
    if ds.backend == "manta":
        # reference renders the Manta host from MANTA_URL
        # (default us-east.manta.joyent.com)
        url = os.environ.get("MANTA_URL",
                             "https://us-east.manta.joyent.com")
        host = url.split("://", 1)[-1].rstrip("/")
        return "manta://" + host + (ds.path or "")
    return "file:/" + (ds.path or "")


def _print_ds(out, ds, verbose):
    out.write("%-20s %-59s\n" % (ds.name, _ds_location(ds)))
    if not verbose:
        return
    if ds.filter is not None:
        out.write("    %-11s %s\n" % (
            "filter:", json.dumps(ds.filter, separators=(",", ":"))))
    out.write("    %-11s %s\n" % ("dataFormat:", '"%s"' % ds.data_format))
    if ds.index_path is not None:
        out.write("    %-11s %s\n" % (
            "indexPath:", '"%s"' % ds.index_path))
    if ds.time_format is not None:
        out.write("    %-11s %s\n" % (
            "timeFormat:", '"%s"' % ds.time_format))
    if ds.time_field is not None:
        out.write("    %-11s %s\n" % (
            "timeField:", '"%s"' % ds.time_field))
    if ds.nshards is not None:
        out.write("    %-11s %d\n" % ("shards:", ds.nshards))


def cmd_datasource_list(argv):
    opts, args = parse_args(argv, ["verbose"])
    check_arg_count(args, 0)
    cfg = _load_config()
    out = sys.stdout
    out.write("%-20s %-59s\n" % ("DATASOURCE", "LOCATION"))
    for ds in cfg.datasource_list():
        _print_ds(out, ds, opts.get("verbose", False))


def cmd_datasource_show(argv):
    opts, args = parse_args(argv, ["verbose"])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    out = sys.stdout
    out.write("%-20s %-59s\n" % ("DATASOURCE", "LOCATION"))
    _print_ds(out, ds, opts.get("verbose", False))


# ---- metric commands ----

def _parse_breakdowns(values):
    """attr-parse each -b occurrence separately (reference
    dnExpandArray, bin/dn; errors quote the offending entry)."""
    parsed = []
    for v in values:
        one = attrs_parse(v)
        if isinstance(one, AttrsError):
            raise UsageError(
                'bad value for "breakdowns" ("%s"): %s' % (v, one))
        parsed.extend(one)
    return parsed


def cmd_metric_add(argv):
    opts, args = parse_args(argv, ["filter", "breakdowns"])
    check_arg_count(args, 2)
    dsname, metname = args
    cfg = _load_config()
    _get_datasource(cfg, dsname)
    filt = _parse_filter(opts.get("filter"))
    breakdowns = []
    if opts.get("breakdowns"):
        from .query import parse_fields
        breakdowns = parse_fields(_parse_breakdowns(opts["breakdowns"]))
    cfg.metric_add(mod_config.Metric(
        name=metname, datasource=dsname, filter=filt,
        breakdowns=breakdowns))
    _save_config(cfg)


def cmd_metric_remove(argv):
    opts, args = parse_args(argv, [])
    check_arg_count(args, 2)
    cfg = _load_config()
    cfg.metric_remove(args[0], args[1])
    _save_config(cfg)


def cmd_metric_list(argv):
    opts, args = parse_args(argv, ["verbose"])
    check_arg_count(args, 1)
    cfg = _load_config()
    out = sys.stdout
    out.write("%-20s %-20s\n" % ("DATASOURCE", "METRIC"))
    for m in cfg.datasource_metrics(args[0]):
        out.write("%-20s %-20s\n" % (m.datasource, m.name))
        if not opts.get("verbose"):
            continue
        if m.filter is not None:
            out.write("    %-11s %s\n" % (
                "filter:", json.dumps(m.filter, separators=(",", ":"))))
        if m.breakdowns:
            out.write("    %-11s %s\n" % (
                "breakdowns:",
                ", ".join(b["name"] for b in m.breakdowns)))


# ---- scan/query shared ----

def _query_from_opts(opts, allow_reserved=False):
    filt = _parse_filter(opts.get("filter"))
    breakdowns = []
    if opts.get("breakdowns"):
        breakdowns = _parse_breakdowns(opts["breakdowns"])
    try:
        qc = QueryConfig(
            filter=filt, breakdowns=breakdowns,
            time_after=opts.get("after"), time_before=opts.get("before"),
            allow_reserved=allow_reserved)
    except QueryError as e:
        raise FatalError("invalid query: %s" % e)
    if opts.get("gnuplot") and len(qc.breakdowns) != 1:
        # validated up front, before any scanning
        # (reference bin/dn:713-716)
        raise FatalError(
            "--gnuplot can only be used with exactly one breakdown")
    return qc


# record-pipeline stages whose counters must conserve records
# (aggregators emit distinct groups, find stages expand directories)
_CONSERVING_STAGES = frozenset([
    "json parser", "SkinnerAdapterStream", "Datasource filter",
    "User filter", "Datetime parser", "Time filter"])
_DROP_KINDS = ("invalid json", "nfilteredout", "nfailedeval", "undef",
               "baddate", "nonnumeric")


def _integrity_guard(result):
    """Premature-exit guard analog (reference bin/dn:1276-1311): the
    reference fails a run that exits 0 without completing its pipeline
    (dropped callbacks).  The synchronous analog of 'work silently
    vanished' is a conservation violation — a stage whose ninputs !=
    noutputs + attributed drops.  On violation: the reference's error
    wording, a full counter dump, exit 1."""
    for name, c in result.stages:
        if name not in _CONSERVING_STAGES:
            continue
        if "ninputs" not in c or "noutputs" not in c:
            continue
        drops = sum(c.get(k, 0) for k in _DROP_KINDS)
        if c["ninputs"] != c["noutputs"] + drops:
            sys.stderr.write("ERROR: internal error: premature exit\n")
            mod_output.dump_counters(result.stages, out=sys.stderr)
            raise FatalError(
                'stage "%s" dropped records without attribution '
                "(ninputs=%d noutputs=%d drops=%d)"
                % (name, c["ninputs"], c["noutputs"], drops))


def _output_result(query, opts, result, title=None):
    """Render a ScanResult per the raw/points/gnuplot/pretty options
    (reference dnOutput, bin/dn:924-967)."""
    _integrity_guard(result)
    agg = result.aggregators[0]
    npoints = agg.noutputs()
    if opts.get("points"):
        mod_output.output_points(agg.points())
    elif opts.get("raw"):
        mod_output.output_raw(query, agg.rows())
    elif opts.get("gnuplot"):
        mod_output.output_gnuplot(query, agg.rows(), title)
    else:
        mod_output.output_pretty(query, agg.rows())

    if opts.get("counters"):
        stages = list(result.stages)
        if not opts.get("points"):
            # the flattener stage exists only when output is
            # re-aggregated to rows; --points streams raw points
            # (reference golden: tst.scan_fileset.sh.out --points
            # --counters section has no Flattener lines)
            stages.append(("Flattener",
                           {"ninputs": npoints, "noutputs": 1}))
        mod_output.dump_counters(stages)

    if opts.get("warnings"):
        # per-record vstream-style warnings with context chains
        # (reference bin/dn warn(): 'warn: <msg>\n    at <label>',
        # bin/dn:135-144); engines that only count drops (the GPU
        # kernel) fall back to per-stage counter summaries in the
        # same shape
        printed = False
        for msg, label in getattr(result, "warnings", []) or []:
            sys.stderr.write("warn: %s\n    at %s\n" % (msg, label))
            printed = True
        if not printed:
            drop_kinds = ("invalid json", "nfailedeval", "undef",
                          "baddate", "nonnumeric")
            for name, counters in result.stages:
                for kind in drop_kinds:
                    v = counters.get(kind, 0)
                    if v:
                        sys.stderr.write(
                            "warn: %d record%s dropped (%s)\n"
                            "    at %s\n"
                            % (v, "s" if v != 1 else "", kind, name))


def cmd_scan(argv):
    opts, args = parse_args(argv, [
        "before", "after", "filter", "breakdowns", "raw", "points",
        "counters", "warnings", "gnuplot", "dry-run"])
    check_arg_count(args, 1)
    if opts.get("warnings"):
        # engines check this to collect per-record warning context
        os.environ["DRAGNET_WARNINGS"] = "1"
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    backend = datasource_for_config(ds)
    query = _query_from_opts(opts)
    try:
        result = backend.scan(query, dry_run=bool(opts.get("dry-run")))
    except ValueError as e:
        raise FatalError(str(e))
    if result is None or getattr(result, "nonroot", False):
        return  # dry run or non-rank-0 of a distributed scan
    _output_result(query, opts, result, title=args[0])


def cmd_query(argv):
    opts, args = parse_args(argv, [
        "before", "after", "filter", "breakdowns", "interval", "raw",
        "points", "counters", "gnuplot", "dry-run"])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    backend = datasource_for_config(ds)
    query = _query_from_opts(opts)
    interval = opts.get("interval", "day")
    try:
        result = backend.query(query, interval=interval,
                               dry_run=bool(opts.get("dry-run")))
    except ValueError as e:
        raise FatalError(str(e))
    if result is None or getattr(result, "nonroot", False):
        return
    if getattr(result, "errors", None):
        for path, msg in result.errors:
            sys.stderr.write("warn: %s: %s\n" % (path, msg))
        if len(result.errors) == len(result.files):
            raise FatalError(result.errors[0][1])
    _output_result(query, opts, result, title=args[0])


# ---- index commands ----

def _metrics_for_index(cfg, ds, opts):
    """Configured metrics, or --index-config override
    (reference lib/dragnet.js:573-598, bin/dn:617-677)."""
    path = opts.get("index-config")
    if path is not None:
        try:
            if path == "-":
                parsed = json.load(sys.stdin)
            else:
                with open(path) as f:
                    parsed = json.load(f)
        except (OSError, ValueError) as e:
            raise FatalError('index config "%s": %s' % (path, e))
        metrics = parsed.get("metrics", [])
    else:
        metrics = [m.serialize(skip_datasource=True)
                   for m in cfg.datasource_metrics(ds.name)]
    if not metrics:
        raise FatalError(
            'no metrics configured for datasource "%s"' % ds.name)
    from .query import parse_fields
    for m in metrics:
        m["breakdowns"] = parse_fields(m.get("breakdowns", []))
    return metrics


def cmd_build(argv):
    opts, args = parse_args(argv, [
        "before", "after", "interval", "index-config", "dry-run"])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    backend = datasource_for_config(ds)
    metrics = _metrics_for_index(cfg, ds, opts)
    from .query import parse_time_bounds
    after_ms, before_ms = parse_time_bounds(
        opts.get("after"), opts.get("before"))
    interval = opts.get("interval", "day")
    try:
        backend.build(metrics, interval=interval, after_ms=after_ms,
                      before_ms=before_ms,
                      dry_run=bool(opts.get("dry-run")))
    except ValueError as e:
        raise FatalError(str(e))


def cmd_index_config(argv):
    opts, args = parse_args(argv, [])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    metrics = [m.serialize(skip_datasource=True)
               for m in cfg.datasource_metrics(ds.name)]
    sys.stdout.write(json.dumps({"metrics": metrics}, indent=4) + "\n")


def cmd_index_scan(argv):
    """Map phase of a distributed build: emit tagged aggregated points."""
    opts, args = parse_args(argv, [
        "index-config", "interval", "before", "after", "filter",
        "breakdowns", "counters"])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    backend = datasource_for_config(ds)
    metrics = _metrics_for_index(cfg, ds, opts)
    from .query import parse_time_bounds
    after_ms, before_ms = parse_time_bounds(
        opts.get("after"), opts.get("before"))
    interval = opts.get("interval", "day")
    try:
        rv = backend.index_scan_points(
            metrics, interval, after_ms=after_ms, before_ms=before_ms)
    except ValueError as e:
        raise FatalError(str(e))
    points, stages = rv
    mod_output.output_points(points)
    if opts.get("counters"):
        mod_output.dump_counters(stages)


def cmd_index_read(argv):
    """Reduce phase of a distributed build: read tagged points from
    stdin, materialize the index tree at the datasource's index path."""
    opts, args = parse_args(argv, ["index-config", "interval"])
    check_arg_count(args, 1)
    cfg = _load_config()
    ds = _get_datasource(cfg, args[0])
    if not ds.index_path:
        raise FatalError(
            'datasource is missing "indexPath" for index operations')
    metrics = _metrics_for_index(cfg, ds, opts)
    interval = opts.get("interval", "day")

    from .points import Aggregator
    from .datasource.file import metric_query, write_index
    queries = [metric_query(m, interval, "__dn_ts")
               for m in metrics]
    # Re-aggregate incoming tagged points per metric: the C++ reducer
    # (index/_points) takes the flat-scalar fast shape and returns the
    # rest for the Python oracle below (DRAGNET_PY_POINTS=1 forces the
    # all-Python path).
    aggs = [Aggregator(q) for q in queries]
    from .points import reduce_tagged_stream
    try:
        lines = reduce_tagged_stream(sys.stdin.buffer, aggs, queries)
    except ImportError:
        lines = sys.stdin.buffer
    for line in lines:
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
    points = []
    for mi, agg in enumerate(aggs):
        for p in agg.points():
            p["fields"]["__dn_metric"] = mi
            points.append(p)
    write_index(ds.index_path, metrics, interval, points)


if __name__ == "__main__":
    sys.exit(main())
