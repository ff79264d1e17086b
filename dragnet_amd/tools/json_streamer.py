"""
json_streamer: read NDJSON on stdin and print the count of valid JSON
objects found, exercising the line parser (reference
tools/json_streamer).  Progress lines go to stderr every 2000 records.

    ... | python -m dragnet_amd.tools.json_streamer [--stop]

--stop: flow-control probe — read only a bounded amount and report the
internal buffer length once per second, demonstrating that an
unconsumed stream does not grow memory (the reference monitors the
Node stream's _readableState/_writableState lengths the same way).
"""

import sys
import time

from ..scan_cpu import parse_json_line

BUF_CAP = 64 * 1024


def main(argv=None):
    argv = sys.argv[1:] if argv is None else argv
    stdin = sys.stdin.buffer

    if argv and argv[0] == "--stop":
        buf = stdin.read(BUF_CAP)  # bounded: no further reads
        try:
            while True:
                sys.stderr.write("%d\n" % len(buf))
                sys.stderr.flush()
                time.sleep(1)
        except KeyboardInterrupt:
            return 0

    count = 0
    partial = b""
    while True:
        chunk = stdin.read(1 << 20)
        if not chunk:
            break
        data = partial + chunk
        lines = data.split(b"\n")
        partial = lines.pop()
        for line in lines:
            count = _bump(count, line)
    if partial:
        count = _bump(count, partial)
    print("%d" % count)
    return 0


def _count_line(line):
    try:
        parse_json_line(line.decode("utf-8"))
        return True
    except (ValueError, UnicodeDecodeError) as e:
        sys.stderr.write("warn: %s\n" % e)
        return False


def _bump(count, line):
    if not _count_line(line):
        return count
    count += 1
    if count % 2000 == 0:
        sys.stderr.write("processed %d lines\n" % count)
    return count


if __name__ == "__main__":
    sys.exit(main())
