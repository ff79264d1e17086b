"""
Result rendering: raw JSON, points, pretty table, DTrace-style histogram,
gnuplot script.

Formats are byte-compatible with the reference CLI's renderers
(reference bin/dn:924-1274; goldens tests/dn/local/*.out): uppercase column
headers, content-sized columns, right-aligned numbers, 40-char '@' histogram
bars, bucket-minimum / ISO-date expansion of flattened rows.
"""

import sys

from . import jsdate
from .points import js_num_str


def _sort_rows(rows):
    """Elementwise sort: strings lexicographic, numbers numeric
    (reference dnOutputSortRows, bin/dn:980-999)."""
    def keyf(row):
        return tuple(
            (0, 0, v) if isinstance(v, (int, float)) else (1, v, 0)
            for v in row)
    return sorted(rows, key=keyf)


def expand_values(query, rows):
    """Replace ordinal bucket indices with bucket minimums and date
    seconds with ISO strings — except the LAST column when it is
    aggregated (kept ordinal for the histogram renderer)
    (reference dnOutputExpandValues, bin/dn:1000-1031)."""
    coldefs = query.breakdowns
    quantized = len(coldefs) > 0 and "aggr" in coldefs[-1]
    for j, c in enumerate(coldefs):
        if quantized and j == len(coldefs) - 1:
            continue
        bk = query.bucketizers.get(c["name"])
        if bk is not None:
            for row in rows:
                row[j] = bk.bucket_min(row[j])
        if "date" in c:
            for row in rows:
                row[j] = jsdate.to_iso(row[j])
    return rows


def _cell_str(v):
    if isinstance(v, str):
        return v
    return js_num_str(v)


def output_pretty(query, rows, out=None):
    out = out or sys.stdout
    coldefs = query.breakdowns
    rows = [list(r) if isinstance(r, list) else r for r in rows]
    expand_values(query, rows)

    quantized = len(coldefs) > 0 and "aggr" in coldefs[-1]
    if quantized:
        output_pretty_quantized(query, rows, out)
        return

    labels = [c["name"].upper() for c in coldefs] + ["VALUE"]
    widths = [len(s) for s in labels]
    aligns = ["l"] * len(coldefs) + ["r"]

    if len(rows) == 0:
        return
    if len(rows) == 1 and isinstance(rows[0], (int, float)):
        rows[0] = [rows[0]]

    for row in rows:
        for j, v in enumerate(row):
            if isinstance(v, (int, float)):
                aligns[j] = "r"
            w = len(_cell_str(v))
            if widths[j] < w:
                widths[j] = w

    def emit(cells):
        parts = []
        for j, cell in enumerate(cells):
            if aligns[j] == "r":
                parts.append(cell.rjust(widths[j]))
            else:
                parts.append(cell.ljust(widths[j]))
        out.write(" ".join(parts).rstrip() + "\n")

    emit(labels)
    for row in _sort_rows(rows):
        emit([_cell_str(v) for v in row])


def output_pretty_quantized(query, rows, out):
    """Group rows on the leading discrete columns; render one DTrace-style
    distribution per group (reference dnOutputPrettyQuantized,
    bin/dn:1093-1147)."""
    coldefs = query.breakdowns
    qcol = coldefs[-1]
    bucketizer = query.bucketizers[qcol["name"]]
    asdate = "date" in qcol

    groups = {}
    for row in rows:
        label = ", ".join(_cell_str(v) for v in row[:-2]) + "\n"
        groups.setdefault(label, []).append([row[-2], row[-1]])

    for i, label in enumerate(sorted(groups.keys())):
        if i != 0:
            out.write("\n")
        out.write(label)
        distr = sorted(groups[label], key=lambda d: d[0])
        print_distribution(out, distr, bucketizer, asdate)


def print_distribution(out, distr, bucketizer, asdate=False):
    """(reference dnPrintDistribution, bin/dn:1149-1199)"""
    if asdate:
        out.write("          ")
    out.write("           ")
    out.write("value  ------------- Distribution ------------- count\n")

    if len(distr) == 0:
        return

    total = sum(d[1] for d in distr)

    # Skip leading empty buckets for large values (e.g. timestamps)
    bi = distr[0][0] if distr[0][0] > 100 else 0

    di = 0
    while di < len(distr) + 1:
        if di == len(distr):
            count = 0
            di += 1
        elif distr[di][0] == bi:
            count = distr[di][1]
            di += 1
        else:
            count = 0

        normalized = int(40.0 * count / total + 0.5) if total else 0
        dots = "@" * normalized + " " * (40 - normalized)
        mn = bucketizer.bucket_min(bi)
        label = jsdate.to_iso(mn) if asdate else js_num_str(mn)
        if asdate:
            out.write("  %s |%s %s\n" % (label.rjust(24), dots,
                                         js_num_str(count)))
        else:
            out.write("%s |%s %s\n" % (label.rjust(16), dots,
                                       js_num_str(count)))
        bi += 1


def output_raw(query, rows, out=None):
    """JSON.stringify of the flattened rows array (reference bin/dn:972)."""
    import json
    out = out or sys.stdout
    out.write(json.dumps(rows, separators=(",", ":")) + "\n")


def point_json(point):
    import json
    return json.dumps(
        {"fields": point["fields"], "value": point["value"]},
        separators=(",", ":"))


def output_points(points, out=None):
    """Emit points as NDJSON — through the C++ serializer
    (index/_points, byte-exact json.dumps; any point with a
    non-scalar value falls back to point_json in place) unless
    DRAGNET_PY_POINTS=1 or the extension is absent."""
    import os
    out = out or sys.stdout
    if os.environ.get("DRAGNET_PY_POINTS") != "1":
        try:
            from .index import _points
        except ImportError:
            pass
        else:
            data = _points.serialize_points(list(points), point_json)
            buf = getattr(out, "buffer", None)
            if buf is not None:
                out.flush()  # keep text-layer writes ordered
                buf.write(data)
            else:
                out.write(data.decode("utf-8"))
            return
    for p in points:
        out.write(point_json(p) + "\n")


def output_gnuplot(query, rows, title, out=None):
    """Emit a gnuplot script (reference dnOutputGnuplot, bin/dn:1204-1274)."""
    out = out or sys.stdout
    coldefs = query.breakdowns
    rows = [list(r) for r in rows]

    out.write("#\n")
    out.write("# This is a GNUplot input file generated automatically\n")
    out.write('# by the Dragnet "dn" command.  You can use it to create\n')
    out.write('# a graph as a PNG image (as file "graph.png") using:\n')
    out.write("#\n")
    out.write("#     gnuplot < this_file > graph.png\n")
    out.write("#\n")
    out.write("set terminal png size 1200,600\n")
    out.write('set title "' + title + '"\n')

    if "date" in coldefs[0]:
        out.write("# Configure plots to use the x-axis as time.\n")
        out.write("set xdata time;\n")
        out.write('set timefmt "%s";\n')
        out.write('set format x "%m/%d\\n%H:%MZ"\n')

    out.write("# Add 10% padding at the top of the graph.\n")
    out.write("set offsets graph 0, 0, 0.1, 0\n")
    out.write("# The y-axis should always start at zero.\n")
    out.write("set yrange [0:*]\n")
    out.write('set ylabel "Count"\n')
    out.write("set ytics\n")

    assert len(coldefs) == 1
    xquant = coldefs[0]["name"] in query.bucketizers
    if xquant:
        out.write('plot "-" using 1:2 with linespoints title "Value"\n')
    else:
        out.write('plot "-" using (column(0)):2:xtic(1) '
                  'with linespoints title "Value"\n')

    for row in _sort_rows(rows):
        if xquant:
            b = query.bucketizers[coldefs[0]["name"]]
            x = b.bucket_min(row[0])
        else:
            x = row[0]
        out.write("\t" + _cell_str(x) + " " + _cell_str(row[1]) + "\n")

    out.write("\te\n")


def dump_counters(stages, out=None):
    """Per-stage counter dump in the vstream vsDumpCounters format
    (observed: tests/dn/local/tst.scan_fileset.sh.out:2464-2486).

    stages: list of (stage_name, {counter: value}) in pipeline order.
    """
    out = out or sys.stderr
    for name, counters in stages:
        for cname in sorted(counters.keys()):
            v = counters[cname]
            if v == 0:
                continue
            out.write("%-18s %-15s%6d\n" % (name, cname + ":", v))
