"""Breakdown attr-parser tests (ported table:
reference tests/lib/tst.attrsparse.js:7-107)."""

from dragnet_amd.attrs import AttrsError, attrs_parse

CASES = [
    ("foo", [{"name": "foo"}]),
    ("foo,bar", [{"name": "foo"}, {"name": "bar"}]),
    ("foo[b]", [{"name": "foo", "b": ""}]),
    ("foo[boolprop]", [{"name": "foo", "boolprop": ""}]),
    ("foo[myprop=one]", [{"name": "foo", "myprop": "one"}]),
    ("foo[myprop=one],bar",
     [{"name": "foo", "myprop": "one"}, {"name": "bar"}]),
    ("foo[p1=one,p2,p3=three],bar",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    (",foo[p1=one,p2,p3=three],bar",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    ("foo[p1=one,p2,p3=three],bar,",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    ("foo[p1=one,p2,p3=three],,bar",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    ("foo[p1=one,p2,,p3=three],,bar",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    ("foo[p1=one,p2,p3=three],bar[]",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar"}]),
    ("foo[p1=one,p2,p3=three],bar[,p4]",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar", "p4": ""}]),
    ("foo[p1=one,p2,p3=three],bar[,p4=]",
     [{"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "bar", "p4": ""}]),
    ("bar,foo[p1=one,p2,p3=three],baz,qant[p1=onetwo],junk[p5]",
     [{"name": "bar"},
      {"name": "foo", "p1": "one", "p2": "", "p3": "three"},
      {"name": "baz"},
      {"name": "qant", "p1": "onetwo"},
      {"name": "junk", "p5": ""}]),
]

ERRORS = [
    ("foo[", "unexpected end of string"),
    ("foo[foo", "unexpected end of string"),
    ("foo[foo=", "unexpected end of string"),
    ("foo[=]", "missing attribute name"),
    ("foo[=bar]", "missing attribute name"),
    ("foo,[]", "missing field name"),
    ("foo,[bar=baz]", "missing field name"),
]


def test_parse_cases():
    for s, expected in CASES:
        assert attrs_parse(s) == expected, s


def test_error_cases():
    for s, msg in ERRORS:
        rv = attrs_parse(s)
        assert isinstance(rv, AttrsError), s
        assert str(rv) == msg, s


def test_trailing_single_char_field():
    # divergence from the reference's off-by-one (documented in attrs.py)
    assert attrs_parse("a") == [{"name": "a"}]
    assert attrs_parse("foo[p],b") == [{"name": "foo", "p": ""},
                                       {"name": "b"}]
