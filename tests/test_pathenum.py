"""Path enumerator tests (ported table:
reference tests/lib/tst.path_enum.js:10-170)."""

import pytest

from dragnet_amd import jsdate
from dragnet_amd.pathenum import PathEnumError, enumerate_paths


def enum(pattern, start, end):
    return enumerate_paths(pattern, jsdate.parse_ms(start),
                           jsdate.parse_ms(end))


CASES = [
    ("my_pattern", "2010-01-01T00:00:00Z", "2010-01-10T00:00:00Z",
     ["my_pattern"]),
    ("my_%%pattern", "2010-01-01T00:00:00Z", "2010-01-10T00:00:00Z",
     ["my_%pattern"]),
    ("my_pattern%%", "2010-01-01T00:00:00Z", "2010-01-10T00:00:00Z",
     ["my_pattern%"]),
    ("%Y", "2010-12-03T01:23:45.678Z", "2013-01-01T00:00:00.000",
     ["2010", "2011", "2012"]),
    ("%Y", "2010-01-01T00:00:00.000Z", "2013-01-01T00:00:00.001",
     ["2010", "2011", "2012", "2013"]),
    ("%Y", "2014-02-01T00:00:00.000Z", "2014-02-01T00:00:00.000Z",
     ["2014"]),
    ("%Y", "2014-12-31T23:59:59.999Z", "2015-01-01T00:00:00.001Z",
     ["2014", "2015"]),
    ("%Y-%m", "2010-06-01T00:00:00Z", "2012-08-01T00:00:00Z",
     ["2010-06", "2010-07", "2010-08", "2010-09", "2010-10", "2010-11",
      "2010-12", "2011-01", "2011-02", "2011-03", "2011-04", "2011-05",
      "2011-06", "2011-07", "2011-08", "2011-09", "2011-10", "2011-11",
      "2011-12", "2012-01", "2012-02", "2012-03", "2012-04", "2012-05",
      "2012-06", "2012-07"]),
    ("%Y-%m", "2010-10-30T00:00:00Z", "2011-05-01T00:00:00Z",
     ["2010-10", "2010-11", "2010-12", "2011-01", "2011-02", "2011-03",
      "2011-04"]),
    ("%Y/%m", "2014-02-01T00:00:00.000Z", "2014-02-01T00:00:00.000Z",
     ["2014/02"]),
    ("%Y/%m", "2014-01-31T23:59:59.999Z", "2014-02-01T00:00:00.001Z",
     ["2014/01", "2014/02"]),
    ("%d", "2010-06-12T03:05:06Z", "2010-06-18T00:00:00Z",
     ["12", "13", "14", "15", "16", "17"]),
    ("year_%Y/month_%m/day_%d/some/other/stuff", "2014-02-26",
     "2014-03-03",
     ["year_2014/month_02/day_26/some/other/stuff",
      "year_2014/month_02/day_27/some/other/stuff",
      "year_2014/month_02/day_28/some/other/stuff",
      "year_2014/month_03/day_01/some/other/stuff",
      "year_2014/month_03/day_02/some/other/stuff"]),
    ("%m/%d", "2014-02-01T00:00:00.000Z", "2014-02-01T00:00:00.000Z",
     ["02/01"]),
    ("%m/%d", "2014-01-31T23:59:59.999Z", "2014-02-01T00:00:00.001Z",
     ["01/31", "02/01"]),
    ("%H", "2010-06-12T03:05:06Z", "2010-06-12T09:00:00Z",
     ["03", "04", "05", "06", "07", "08"]),
    ("%Y/%m/%d/%H", "2014-02-28T20:00:00Z", "2014-03-01T04:00:00Z",
     ["2014/02/28/20", "2014/02/28/21", "2014/02/28/22",
      "2014/02/28/23", "2014/03/01/00", "2014/03/01/01",
      "2014/03/01/02", "2014/03/01/03"]),
    ("%d/%H", "2014-02-01T00:00:00.000Z", "2014-02-01T00:00:00.000Z",
     ["01/00"]),
    ("%d/%H", "2014-01-31T23:59:59.999Z", "2014-02-01T00:00:00.001Z",
     ["31/23", "01/00"]),
]


def test_enumeration():
    for pattern, start, end, expected in CASES:
        assert enum(pattern, start, end) == expected, pattern


def test_errors():
    with pytest.raises(PathEnumError, match='unexpected "%" at char 11'):
        enum("my_pattern%", "2010-01-01", "2010-01-10")
    with pytest.raises(PathEnumError,
                       match='unsupported conversion "%T" at char 11'):
        enum("my_pattern%T", "2010-01-01", "2010-01-10")
    with pytest.raises(PathEnumError,
                       match='"timeStart" is not a valid date'):
        enumerate_paths("%Y", None, 100)
    with pytest.raises(PathEnumError,
                       match='"timeStart" may not be after "timeEnd"'):
        enum("%Y", "2010-01-11", "2010-01-10")


def test_month_only_pattern():
    got = enum("%m", "2010-06-01T00:00:00Z", "2012-08-01T00:00:00Z")
    assert len(got) == 26
    assert got[0] == "06" and got[-1] == "07"
