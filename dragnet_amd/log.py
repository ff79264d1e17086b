"""
Structured logging — the bunyan analog.

The reference attaches a bunyan logger to every component, level
selected by the LOG_LEVEL environment variable, default "warn"
(reference bin/dn:68-71; child loggers per component,
lib/datasource-file.js:102, 224).  This module reproduces the
observable behavior: newline-JSON records on stderr with bunyan's
field set ({name, component, hostname, pid, level, msg, time, v:0})
and numeric level scale (trace=10 … fatal=60), so existing bunyan
tooling can consume `LOG_LEVEL=debug dn scan …` output unchanged.
"""

import json
import os
import socket
import sys
import time

LEVELS = {"trace": 10, "debug": 20, "info": 30,
          "warn": 40, "error": 50, "fatal": 60}

_BUNYAN_V = 0


def _resolve_level(spec):
    if spec is None:
        return LEVELS["warn"]
    spec = str(spec).strip().lower()
    if spec in LEVELS:
        return LEVELS[spec]
    try:
        return int(spec)
    except ValueError:
        return LEVELS["warn"]


class Logger(object):
    """Minimal bunyan-shaped logger (JSON lines on stderr)."""

    def __init__(self, name, component=None, level=None, stream=None):
        self.name = name
        self.component = component
        self.level = (_resolve_level(os.environ.get("LOG_LEVEL"))
                      if level is None else _resolve_level(level))
        self.stream = stream if stream is not None else sys.stderr

    def child(self, component):
        """Per-component child logger (the reference's log.child)."""
        c = Logger(self.name, component=component, level=self.level,
                   stream=self.stream)
        return c

    def _emit(self, level, msg, extra):
        if level < self.level:
            return
        rec = {
            "name": self.name,
            "hostname": socket.gethostname(),
            "pid": os.getpid(),
            "level": level,
            "msg": msg,
            "time": time.strftime("%Y-%m-%dT%H:%M:%S",
                                  time.gmtime())
                    + (".%03dZ" % int((time.time() % 1) * 1000)),
            "v": _BUNYAN_V,
        }
        if self.component is not None:
            rec["component"] = self.component
        if extra:
            rec.update(extra)
        try:
            self.stream.write(json.dumps(rec) + "\n")
            self.stream.flush()
        except (OSError, ValueError):
            pass  # logging must never take the process down

    def trace(self, msg, **extra):
        self._emit(10, msg, extra)

    def debug(self, msg, **extra):
        self._emit(20, msg, extra)

    def info(self, msg, **extra):
        self._emit(30, msg, extra)

    def warn(self, msg, **extra):
        self._emit(40, msg, extra)

    def error(self, msg, **extra):
        self._emit(50, msg, extra)

    def fatal(self, msg, **extra):
        self._emit(60, msg, extra)


_root = None


def get_logger():
    """The process-wide root logger (name "dragnet", LOG_LEVEL env)."""
    global _root
    if _root is None:
        _root = Logger("dragnet")
    return _root
