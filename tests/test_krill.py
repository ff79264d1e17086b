"""Predicate language semantics (JS-coercion rules pinned by reference
goldens; see dragnet_amd/krill.py docstring)."""

import pytest

from dragnet_amd import krill


def ev(pred, fields):
    return krill.create_predicate(pred).eval(fields)


def test_empty_matches_all():
    assert ev({}, {"x": 1}) is True


def test_eq_strings():
    assert ev({"eq": ["m", "GET"]}, {"m": "GET"})
    assert not ev({"eq": ["m", "GET"]}, {"m": "PUT"})


def test_eq_loose_number_string():
    # "200" (filter) matches numeric 200 (record):
    # tst.scan_file.sh.out:1578-1590
    assert ev({"eq": ["c", "200"]}, {"c": 200})
    assert ev({"eq": ["c", 200]}, {"c": "200"})
    assert not ev({"eq": ["c", "200"]}, {"c": 204})
    assert ev({"eq": ["c", 200.0]}, {"c": 200})


def test_eq_null():
    assert not ev({"eq": ["c", "poseidon"]}, {"c": None})
    assert ev({"eq": ["c", None]}, {"c": None})
    assert not ev({"eq": ["c", None]}, {"c": 0})
    assert not ev({"eq": ["c", 0]}, {"c": None})


def test_eq_bool():
    assert ev({"eq": ["audit", True]}, {"audit": True})
    assert ev({"eq": ["audit", True]}, {"audit": 1})
    assert not ev({"eq": ["audit", True]}, {"audit": "true"})
    assert ev({"eq": ["audit", True]}, {"audit": "1"})


def test_missing_field_drops():
    with pytest.raises(krill.MissingFieldError):
        ev({"eq": ["req.caller", "x"]}, {"req": {}})
    with pytest.raises(krill.MissingFieldError):
        ev({"eq": ["nope", 1]}, {})


def test_nested_pluck():
    assert ev({"eq": ["req.method", "GET"]},
              {"req": {"method": "GET"}})


def test_relational():
    assert ev({"lt": ["v", 10]}, {"v": 9})
    assert not ev({"lt": ["v", 10]}, {"v": 10})
    assert ev({"le": ["v", 10]}, {"v": 10})
    assert ev({"gt": ["v", 10]}, {"v": 11})
    assert ev({"ge": ["v", 10]}, {"v": 10})
    # string-vs-number coerces numerically
    assert ev({"lt": ["v", 10]}, {"v": "9"})
    assert ev({"gt": ["v", "10"]}, {"v": 11})
    # string-vs-string is lexicographic (JS)
    assert ev({"lt": ["v", "b"]}, {"v": "a"})
    assert ev({"gt": ["v", "10"]}, {"v": "9"})  # "9" > "10" lexically
    # NaN makes comparisons false
    assert not ev({"lt": ["v", 10]}, {"v": "zzz"})
    assert not ev({"ge": ["v", 10]}, {"v": "zzz"})


def test_and_or():
    p = {"and": [{"eq": ["a", 1]}, {"or": [{"eq": ["b", 2]},
                                           {"eq": ["b", 3]}]}]}
    assert ev(p, {"a": 1, "b": 3})
    assert not ev(p, {"a": 1, "b": 4})
    assert not ev(p, {"a": 2, "b": 2})


def test_fields():
    p = krill.create_predicate(
        {"and": [{"eq": ["a", 1]}, {"lt": ["b.c", 2]},
                 {"ge": ["a", 0]}]})
    assert p.fields() == ["a", "b.c"]


def test_validate_errors():
    for bad in [{"xx": ["a", 1]},
                {"eq": ["a"]},
                {"eq": "a"},
                {"and": []},
                {"eq": ["a", 1], "ne": ["b", 2]},
                "notadict"]:
        with pytest.raises(krill.KrillError):
            krill.create_predicate(bad)


def test_to_sql():
    p = krill.create_predicate(
        {"and": [{"eq": ["req.method", "GET"]},
                 {"ge": ["latency", 100]}]})
    sql = p.to_sql({"req.method": "req_method"})
    assert sql == '("req_method" = \'GET\') AND ("latency" >= 100)'


def test_filter_and():
    assert krill.filter_and(None, None) is None
    f = {"eq": ["a", 1]}
    assert krill.filter_and(f, None) == f
    assert krill.filter_and(f, f) == {"and": [f, f]}


def test_to_number():
    from math import isnan
    assert krill.to_number("") == 0.0
    assert krill.to_number("  12 ") == 12.0
    assert krill.to_number("0x10") == 16.0
    assert isnan(krill.to_number("12px"))
    assert isnan(krill.to_number("inf"))
    assert krill.to_number("Infinity") == float("inf")
    assert krill.to_number(None) == 0.0
    assert krill.to_number(True) == 1.0
    # JS Number() rejects signed hex
    assert isnan(krill.to_number("-0x10"))
    assert isnan(krill.to_number("+0x10"))
    assert isnan(krill.to_number(" -0x10 "))
