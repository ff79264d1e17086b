"""Structured logging (the bunyan analog): LOG_LEVEL filtering and
bunyan record shape (reference bin/dn:68-71 attaches a bunyan logger,
level from LOG_LEVEL, default warn)."""

import io
import json

from dragnet_amd.log import LEVELS, Logger


def recs(buf):
    return [json.loads(line) for line in buf.getvalue().splitlines()]


def test_default_level_is_warn():
    buf = io.StringIO()
    log = Logger("dragnet", level="warn", stream=buf)
    log.debug("nope")
    log.info("nope")
    log.warn("yes")
    log.error("also")
    out = recs(buf)
    assert [r["msg"] for r in out] == ["yes", "also"]
    assert [r["level"] for r in out] == [40, 50]


def test_bunyan_record_shape():
    buf = io.StringIO()
    log = Logger("dragnet", level="info", stream=buf)
    log.info("hello", extra_field=7)
    (r,) = recs(buf)
    for k in ("name", "hostname", "pid", "level", "msg", "time", "v"):
        assert k in r
    assert r["v"] == 0 and r["name"] == "dragnet"
    assert r["extra_field"] == 7
    assert r["time"].endswith("Z")


def test_child_component():
    buf = io.StringIO()
    log = Logger("dragnet", level="debug", stream=buf)
    log.child("datasource-file").debug("scan starting")
    (r,) = recs(buf)
    assert r["component"] == "datasource-file"


def test_level_spec_parsing():
    assert Logger("x", level="trace").level == LEVELS["trace"]
    assert Logger("x", level="30").level == 30
    assert Logger("x", level="bogus").level == LEVELS["warn"]
