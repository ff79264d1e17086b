"""
JavaScript-compatible date parsing/formatting (the subset the engine needs).

The reference engine leans on V8's Date.parse for synthetic date fields
(reference lib/stream-synthetic.js:58-80) and on `new Date(ms).toISOString()`
for display (reference bin/dn:1017-1030).  We implement the ISO-8601 subset
(ES5 semantics: a missing timezone means UTC, as in the Node 0.10 era the
reference targeted) plus RFC-1123-ish fallbacks are NOT supported: all of
the engine's own data paths emit ISO-8601.
"""

import re

_ISO_RE = re.compile(
    r"^(\d{4})"
    r"(?:-(\d{2})"
    r"(?:-(\d{2})"
    r"(?:[T ](\d{2}):(\d{2})"
    r"(?::(\d{2})"
    r"(?:\.(\d{1,9}))?"
    r")?"
    r"(Z|[+-]\d{2}:?\d{2})?"
    r")?)?)?$"
)

_DAYS_PER_MONTH = [31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31]


def _is_leap(y):
    return y % 4 == 0 and (y % 100 != 0 or y % 400 == 0)


def days_from_civil(y, m, d):
    """Days since 1970-01-01 for a proleptic Gregorian date.

    Howard Hinnant's days_from_civil algorithm; also used (in HIP form)
    by the date kernels so host and device agree bit-for-bit.
    """
    y -= m <= 2
    era = (y if y >= 0 else y - 399) // 400
    yoe = y - era * 400
    doy = (153 * (m + (-3 if m > 2 else 9)) + 2) // 5 + d - 1
    doe = yoe * 365 + yoe // 4 - yoe // 100 + doy
    return era * 146097 + doe - 719468


def civil_from_days(z):
    """Inverse of days_from_civil: (year, month, day)."""
    z += 719468
    era = (z if z >= 0 else z - 146096) // 146097
    doe = z - era * 146097
    yoe = (doe - doe // 1460 + doe // 36524 - doe // 146096) // 365
    y = yoe + era * 400
    doy = doe - (365 * yoe + yoe // 4 - yoe // 100)
    mp = (5 * doy + 2) // 153
    d = doy - (153 * mp + 2) // 5 + 1
    m = mp + (3 if mp < 10 else -9)
    return (y + (m <= 2), m, d)


def parse_ms(s):
    """Parse an ISO-8601-ish date string to milliseconds since the epoch.

    Returns None where JS Date.parse would return NaN.
    """
    if not isinstance(s, str):
        return None
    m = _ISO_RE.match(s.strip())
    if m is None:
        return None
    year = int(m.group(1))
    month = int(m.group(2) or 1)
    day = int(m.group(3) or 1)
    hh = int(m.group(4) or 0)
    mm = int(m.group(5) or 0)
    ss = int(m.group(6) or 0)
    frac = m.group(7) or ""
    ms = int((frac + "000")[:3]) if frac else 0
    tz = m.group(8)

    if not (1 <= month <= 12):
        return None
    dim = _DAYS_PER_MONTH[month - 1] + (
        1 if (month == 2 and _is_leap(year)) else 0)
    if not (1 <= day <= dim):
        return None
    if hh > 24 or mm > 59 or ss > 59:
        return None
    if hh == 24 and (mm or ss or ms):
        # V8 accepts hour 24 only as exactly 24:00:00.000
        return None

    days = days_from_civil(year, month, day)
    total = ((days * 24 + hh) * 60 + mm) * 60 + ss
    total_ms = total * 1000 + ms

    if tz and tz != "Z":
        sign = 1 if tz[0] == "+" else -1
        tzh = int(tz[1:3])
        tzm = int(tz[-2:])
        total_ms -= sign * (tzh * 60 + tzm) * 60 * 1000
    return total_ms


def to_iso(seconds):
    """Format unix seconds as JS `new Date(s*1000).toISOString()`."""
    ms_total = round(seconds * 1000)
    ms = ms_total % 1000
    secs = ms_total // 1000
    days = secs // 86400
    rem = secs % 86400
    y, mo, d = civil_from_days(days)
    hh = rem // 3600
    mm = (rem % 3600) // 60
    ss = rem % 60
    return "%04d-%02d-%02dT%02d:%02d:%02d.%03dZ" % (y, mo, d, hh, mm, ss, ms)
