"""
Time-based path enumeration: expand a strftime-like pattern over a time
range into concrete paths.

Supports exactly %Y %m %d %H and %% (reference lib/path-enum.js:27-32;
semantics pinned by tests/lib/tst.path_enum.js): the start time is floored
to the smallest unit present in the pattern, paths are emitted
calendar-correctly, and enumeration continues while the next aligned time
is strictly before the end time (the floored start is always emitted).
"""

from . import jsdate


class PathEnumError(Exception):
    pass


_UNITS = {"Y": 0, "m": 1, "d": 2, "H": 3}  # larger value = smaller unit


def parse_pattern(pattern):
    """Returns (segments, smallest_unit) where segments is a list of
    literal strings and single-char conversion codes; smallest_unit is
    one of None,'Y','m','d','H'."""
    segs = []
    lit = []
    smallest = None
    smallest_rank = -1
    i = 0
    n = len(pattern)
    while i < n:
        c = pattern[i]
        if c != "%":
            lit.append(c)
            i += 1
            continue
        if i + 1 >= n:
            raise PathEnumError('unexpected "%%" at char %d' % (i + 1))
        conv = pattern[i + 1]
        if conv == "%":
            lit.append("%")
            i += 2
            continue
        if conv not in _UNITS:
            raise PathEnumError(
                'unsupported conversion "%%%s" at char %d' % (conv, i + 1))
        if lit:
            segs.append("".join(lit))
            lit = []
        segs.append(("conv", conv))
        if _UNITS[conv] > smallest_rank:
            smallest_rank = _UNITS[conv]
            smallest = conv
        i += 2
    if lit:
        segs.append("".join(lit))
    return segs, smallest


def _format(segs, y, mo, d, h):
    out = []
    for s in segs:
        if isinstance(s, str):
            out.append(s)
        else:
            conv = s[1]
            if conv == "Y":
                out.append("%04d" % y)
            elif conv == "m":
                out.append("%02d" % mo)
            elif conv == "d":
                out.append("%02d" % d)
            else:
                out.append("%02d" % h)
    return "".join(out)


def enumerate_paths(pattern, start_ms, end_ms):
    """Expand pattern over [start, end).  start/end in epoch ms.

    The floored start path is always emitted; subsequent paths while the
    aligned time is strictly < end.
    """
    if start_ms is None:
        raise PathEnumError('"timeStart" is not a valid date')
    if end_ms is None:
        raise PathEnumError('"timeEnd" is not a valid date')
    if start_ms > end_ms:
        raise PathEnumError('"timeStart" may not be after "timeEnd"')

    segs, smallest = parse_pattern(pattern)
    if smallest is None:
        return [_format(segs, 0, 0, 0, 0)]

    days = start_ms // 1000 // 86400
    y, mo, d = jsdate.civil_from_days(days)
    h = (start_ms // 1000 % 86400) // 3600

    # floor to the smallest unit
    if smallest == "Y":
        mo, d, h = 1, 1, 0
    elif smallest == "m":
        d, h = 1, 0
    elif smallest == "d":
        h = 0

    out = []
    while True:
        out.append(_format(segs, y, mo, d, h))
        # calendar-correct increment by the smallest unit
        if smallest == "Y":
            y += 1
        elif smallest == "m":
            mo += 1
            if mo > 12:
                mo = 1
                y += 1
        elif smallest == "d":
            t = jsdate.days_from_civil(y, mo, d) + 1
            y, mo, d = jsdate.civil_from_days(t)
        else:
            h += 1
            if h > 23:
                h = 0
                t = jsdate.days_from_civil(y, mo, d) + 1
                y, mo, d = jsdate.civil_from_days(t)
        t_ms = (jsdate.days_from_civil(y, mo, d) * 86400 + h * 3600) * 1000
        if t_ms >= end_ms:
            break
    return out


def main():
    """CLI harness (the reference tools/pathenum analog):
    pathenum PATTERN START END"""
    import sys

    from . import jsdate
    if len(sys.argv) != 4:
        sys.stderr.write("usage: pathenum PATTERN START END\n")
        return 2
    try:
        for p in enumerate_paths(sys.argv[1],
                                 jsdate.parse_ms(sys.argv[2]),
                                 jsdate.parse_ms(sys.argv[3])):
            print(p)
    except PathEnumError as e:
        sys.stderr.write("pathenum: %s\n" % e)
        return 1
    return 0


if __name__ == "__main__":
    import sys
    sys.exit(main())
