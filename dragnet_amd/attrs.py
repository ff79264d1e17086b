"""
Breakdown mini-language parser.

Parses comma-separated field lists where each field may carry a bracketed
attribute list:

    field1
    field1,field2
    field1[attr1=value1,attr2=value2,attr3],field2[attr2],field3

Semantics match the reference parser (reference: lib/attr-parser.js:1-77 and
its unit-test table tests/lib/tst.attrsparse.js): empty list elements are
skipped, a bare attribute gets value '', `[=x]` is a "missing attribute name"
error, an unterminated bracket is "unexpected end of string", and a bracket
with no preceding field name is "missing field name".  Errors are RETURNED
as AttrsError (not raised) so callers can wrap them.
"""


class AttrsError(Exception):
    pass


def attrs_parse(s):
    """Parse a breakdown spec string into a list of dicts.

    Each dict has at least {'name': <field>}; attributes are extra keys
    whose values are strings ('' for bare flags).  Returns AttrsError on
    malformed input.
    """
    propname = None
    props = None
    rv = []
    i = 0
    j = 0
    n = len(s)
    while i < n:
        c = s[i]
        if propname is None:
            assert props is None
            if c == ",":
                if i - j > 0:
                    rv.append({"name": s[j:i]})
                j = i + 1
            elif c == "[":
                if i - j == 0:
                    return AttrsError("missing field name")
                propname = s[j:i]
                props = {"name": propname}
                j = i + 1
            i += 1
            continue

        assert props is not None
        if c == "," or c == "]":
            if i - j > 0:
                propdef = s[j:i]
                eq = propdef.find("=")
                if eq == -1:
                    props[propdef] = ""
                elif eq == 0:
                    return AttrsError("missing attribute name")
                else:
                    props[propdef[:eq]] = propdef[eq + 1:]
            if c == "]":
                rv.append(props)
                propname = None
                props = None
            j = i + 1
        i += 1

    if propname is not None:
        return AttrsError("unexpected end of string")

    # Trailing bare field after the last ']' or ','.  (The reference's
    # equivalent check is `j < len-1`, which silently drops a trailing
    # single-character field; we use the correct `j < len`.)
    if j < n:
        rv.append({"name": s[j:]})

    return rv
