"""
File enumeration for datasources.

Replaces the reference's recursive FindStream pipeline
(reference lib/fs-find.js:32-205): given root paths (possibly produced by
the time-based path enumerator), walk directories recursively and yield
regular files and character devices; missing paths and stat errors are
warn-and-skip.  Traversal is sorted for deterministic output.

Counter model mirrors the reference find stages so --counters output has
the same shape (observed tests/dn/local/tst.scan_fileset.sh.out:2464-2472).
"""

import os
import stat as _stat


class FindCounters(object):
    def __init__(self):
        self.npathenum = None  # noutputs incl. the EOF push, when
                               # strftime pruning ran (vstream counts
                               # the stream-end: tst.scan_fileset
                               # golden shows npaths+1)
        self.nstarts = 0
        self.nstatted = 0
        self.ndirectories = 0
        self.nregfiles = 0
        self.nchrdevs = 0
        self.noutputs = 0

    def stages(self):
        # The statter/traverser also pass the generation-numbered EOF
        # markers the feedback loop cycles: one per expanded directory
        # plus ONE global end-of-stream marker (reference
        # lib/fs-find.js:185-205) — pinned by the reference goldens:
        # tst.empty.sh.out statter ninputs 2 for one char device,
        # tst.scan_fileset.sh.out 24 for 16 paths + 7 directories,
        # tst.index_fileset.sh.out 25 for 24 enumerated index files.
        cycled = self.nstatted + self.ndirectories + 1
        head = []
        if self.npathenum is not None:
            head.append(("PathEnumerator",
                         {"noutputs": self.npathenum}))
        return head + [
            ("FindStart", {"ninputs": self.nstarts,
                           "noutputs": self.nstarts}),
            ("FindStatter", {"ninputs": cycled,
                             "noutputs": cycled}),
            ("FindTraverser", {"ninputs": cycled,
                               "noutputs": cycled}),
            ("FindFeedback", {"ninputs": cycled,
                              "noutputs": self.noutputs,
                              "ndirectories": self.ndirectories,
                              "nregfiles": self.nregfiles,
                              "nchrdevs": self.nchrdevs}),
        ]


def find_files(roots, counters=None, warn=None):
    """Yield (path, stat) for every regular file / char device under each
    root, depth-first in sorted order.  `warn` is called with (path,
    message) for stat failures."""
    counters = counters if counters is not None else FindCounters()
    for root in roots:
        counters.nstarts += 1
        stack = [root]
        while stack:
            path = stack.pop()
            counters.nstatted += 1
            try:
                st = os.stat(path)
            except OSError as e:
                if warn:
                    warn(path, str(e))
                continue
            mode = st.st_mode
            if _stat.S_ISDIR(mode):
                counters.ndirectories += 1
                try:
                    entries = sorted(os.listdir(path), reverse=True)
                except OSError as e:
                    if warn:
                        warn(path, str(e))
                    continue
                for name in entries:
                    stack.append(os.path.join(path, name))
            elif _stat.S_ISREG(mode) or _stat.S_ISCHR(mode):
                if _stat.S_ISREG(mode):
                    counters.nregfiles += 1
                else:
                    counters.nchrdevs += 1
                counters.noutputs += 1
                yield (path, st)
            # other types are silently skipped


def find_data_files(root, timeformat=None, after_ms=None, before_ms=None,
                    counters=None, warn=None):
    """Enumerate the files a scan should read.

    With a timeformat and both time bounds, expand root/timeformat over
    the bounds and walk only those paths (pruning, reference
    lib/datasource-file.js:218-246); otherwise walk root.
    """
    from . import pathenum
    if before_ms is not None and timeformat is not None:
        pats = list(pathenum.enumerate_paths(
            os.path.join(root, timeformat), after_ms, before_ms))
        if counters is not None:
            # vstream counts the enumerator's end-of-stream push as an
            # output UNLESS the last path's push hit the Node stream
            # high-water mark (16 objects), in which case the null is
            # pushed by the drained-_read branch uncounted.  Golden
            # observations: 1 path -> 2 (x3 occurrences), 24 paths ->
            # 24 (tst.index_fileset.sh.out).
            n = len(pats)
            counters.npathenum = n + 1 if n < 16 else n
        return find_files(pats, counters=counters, warn=warn)
    return find_files([root], counters=counters, warn=warn)
