"""
Replay the reference's OWN golden test outputs byte-for-byte.

The reference ships bash test harnesses whose `.out` files interleave
`# dn scan <args>` marker lines with the exact expected stdout
(tools/catest compares verbatim).  These tests parse those goldens,
run OUR `dn` with the same arguments against the reference's own
fixture data (/root/reference/tests/data), and require identical
output — the strongest possible CLI-parity evidence, using goldens we
did not produce.

Skipped when the reference checkout is not present.
"""

import os
import re
import subprocess

import pytest

REF = "/root/reference/tests"
DATA = os.path.join(REF, "data")

pytestmark = pytest.mark.skipif(
    not os.path.isdir(DATA), reason="reference checkout not present")


def parse_sections(outfile):
    """[(marker_args_str, expected_text), ...] from a .sh.out file."""
    text = open(outfile).read()
    parts = re.split(r"^# dn (.*)$", text, flags=re.M)
    # parts[0] is any preamble; then alternating (args, body)
    sections = []
    for i in range(1, len(parts), 2):
        body = parts[i + 1]
        # the body piece starts with the newline terminating the
        # marker line; histogram sections genuinely begin with a
        # further blank line — strip exactly one
        if body.startswith("\n"):
            body = body[1:]
        sections.append((parts[i], body))
    return sections


def split_args(argstr):
    """The harness echoes `"# dn scan" "$@"`, which flattens the
    quoted JSON filter into space-separated words; re-join a brace-
    balanced run after -f/--filter (the .sh sources single-space the
    JSON, so the reconstruction is exact)."""
    out = []
    i = 0
    n = len(argstr)
    while i < n:
        while i < n and argstr[i] == " ":
            i += 1
        if i >= n:
            break
        if argstr[i] == "{":
            depth = 0
            j = i
            while j < n:
                if argstr[j] == "{":
                    depth += 1
                elif argstr[j] == "}":
                    depth -= 1
                    if depth == 0:
                        j += 1
                        break
                j += 1
            out.append(argstr[i:j])
            i = j
        else:
            j = argstr.find(" ", i)
            if j < 0:
                j = n
            out.append(argstr[i:j])
            i = j
    return out


def sort_d(text):
    """Reproduce the harness's `| sort -d` (dictionary order, C
    locale — the reference CI's collation for JSON point lines)."""
    r = subprocess.run(["sort", "-d"], input=text, text=True,
                       capture_output=True,
                       env=dict(os.environ, LC_ALL="C"))
    return r.stdout


def run_section(dn, argstr, datasource="test_file", strip_prefix=None):
    # the harness's scan()/query() append the datasource last:
    # `dn scan "$@" test_file`
    args = split_args(argstr) + [datasource]
    res = dn(*args)
    assert res.code == 0, (argstr, res.err)
    out = res.out
    if strip_prefix:
        out = out.replace(strip_prefix.rstrip("/") + "/", "")
    if args and "--points" in args:
        out = sort_d(out)
    return out


def expected_body(body):
    """A marked section's own output: the harness echoes one blank
    line after each dn invocation."""
    return body.rstrip("\n") + "\n" if body.strip() else ""


def section_own_output(body):
    """Everything up to the blank separator (unmarked trailing output
    from markerless dn invocations is handled by the caller)."""
    return body


def norm(text):
    return text.rstrip("\n") + "\n" if text.strip() else ""


def test_scan_fileset_goldens(dn):
    """tst.scan_fileset.sh.out: the scan_testcases over the whole
    tree (incl. invalid-JSON lines), unmarked gnuplot outputs glued
    to section 25, then dry-run/--counters sections with time-bound
    file pruning (stderr merged by the harness's 2>&1; paths
    sed-stripped to tests/data/...)."""
    sections = parse_sections(
        os.path.join(REF, "dn", "local", "tst.scan_fileset.sh.out"))
    assert len(sections) == 38
    prefix = os.path.dirname(REF.rstrip("/"))  # /root/reference

    assert dn("datasource-add", "test_input", "--path=" + DATA,
              "--time-format=%Y/%m-%d", "--time-field=time").code == 0
    for argstr, body in sections[:25]:
        got = run_section(dn, argstr, datasource="test_input")
        assert got == expected_body(body), "# dn " + argstr

    # section 25: last scan_testcases section, with the two unmarked
    # `dn scan ... --gnuplot` outputs appended by the harness
    argstr, body = sections[25]
    got = run_section(dn, argstr, datasource="test_input")
    g1 = dn("scan", "-b",
            "timestamp[field=time,date,aggr=lquantize,step=86400]",
            "--gnuplot", "test_input")
    g2 = dn("scan", "-b", "req.method", "--gnuplot", "test_input")
    assert g1.code == 0 and g2.code == 0
    assert norm(got + "\n" + g1.out + g2.out) == norm(body), \
        "# dn " + argstr + " (+gnuplot)"

    # dry-run / counters sections: stdout then stderr (2>&1), paths
    # relative to the workspace root (sed)
    for argstr, body in sections[26:]:
        args = split_args(argstr) + ["test_input"]
        res = dn(*args)
        assert res.code == 0, (argstr, res.err)
        if "--points" in args:
            # `... 2>&1 | sort -d`: sort buffers stdout to EOF, so
            # the unpiped stderr (counters/dry-run list) lands FIRST
            merged = res.err + sort_d(res.out)
        else:
            merged = res.out + res.err
        merged = merged.replace(prefix.rstrip("/") + "/", "")
        assert norm(merged) == norm(body), "# dn " + argstr


def test_scan_file_goldens(dn):
    """tst.scan_file.sh.out: 26 scan_testcases sections against
    one.log, then 4 sections under a datasource filter
    (reference tests/dn/local/tst.scan_file.sh)."""
    sections = parse_sections(
        os.path.join(REF, "dn", "local", "tst.scan_file.sh.out"))
    assert len(sections) == 30
    one = os.path.join(DATA, "2014", "05-01", "one.log")

    assert dn("datasource-add", "test_file", "--path=" + one).code == 0
    for argstr, body in sections[:26]:
        got = run_section(dn, argstr)
        assert got == expected_body(body), "# dn " + argstr

    assert dn("datasource-remove", "test_file").code == 0
    assert dn("datasource-add", "test_file", "--path=" + one,
              "--filter",
              '{ "eq": [ "req.method", "GET" ] }').code == 0
    for argstr, body in sections[26:]:
        got = run_section(dn, argstr)
        assert got == expected_body(body), "# dn " + argstr
