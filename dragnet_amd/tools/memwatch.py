"""
memwatch: given a pid, wait for it to exit, sampling memory usage;
print the maximum RSS and VSZ seen (KB), space-separated — the shape
the reference's resource-bound test consumes (reference
tools/memwatch, tests/dn/local/tst.scan_250k.sh).

    python -m dragnet_amd.tools.memwatch PID [interval_s]
"""

import sys
import time


def sample(pid):
    """(rss_kb, vsz_kb) from /proc, or None if the process is gone
    (a zombie counts as gone: its memory is released)."""
    try:
        with open("/proc/%d/status" % pid) as f:
            rss = vsz = 0
            for line in f:
                if line.startswith("State:") and "Z" in line.split()[1]:
                    return None
                if line.startswith("VmRSS:"):
                    rss = int(line.split()[1])
                elif line.startswith("VmSize:"):
                    vsz = int(line.split()[1])
            return rss, vsz
    except (OSError, ValueError):
        return None


def watch(pid, interval=3.0):
    max_rss = max_vsz = 0
    while True:
        s = sample(pid)
        if s is None:
            break
        max_rss = max(max_rss, s[0])
        max_vsz = max(max_vsz, s[1])
        time.sleep(interval)
    return max_rss, max_vsz


def main(argv=None):
    argv = sys.argv[1:] if argv is None else argv
    if not argv:
        sys.stderr.write("usage: memwatch PID [interval_s]\n")
        return 2
    pid = int(argv[0])
    interval = float(argv[1]) if len(argv) > 1 else 3.0
    max_rss, max_vsz = watch(pid, interval)
    print("%d %d" % (max_rss, max_vsz))
    return 0


if __name__ == "__main__":
    sys.exit(main())
