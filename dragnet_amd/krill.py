"""
"krill" JSON predicate language.

Re-implementation of the predicate semantics the reference engine exposes
through node-krill (reference call sites: lib/krill-skinner-stream.js:29-52,
lib/index-query.js:419-453; serialized ops observed: eq ne lt le gt ge
combined with and/or arrays).

A predicate is a plain JSON structure:

    {}                                    -> matches everything
    {"eq": ["req.method", "GET"]}         -> leaf: field op constant
    {"and": [p1, p2, ...]}                -> conjunction
    {"or":  [p1, p2, ...]}                -> disjunction

Evaluation semantics (verified against reference golden outputs,
tests/dn/local/tst.scan_file.sh.out:1578-1590 where the string filter "200"
matches numeric 200):

  * A MISSING field raises MissingFieldError -> the caller drops the record
    (the reference counts this as the 'nfailedeval' warn-drop,
    lib/krill-skinner-stream.js:37-49).
  * eq/ne use JavaScript loose equality: number-vs-numeric-string compares
    numerically, null equals only null, booleans coerce to numbers.
  * lt/le/gt/ge use JavaScript relational semantics: string-vs-string
    compares lexicographically, anything else coerces to number, and a NaN
    on either side makes the comparison false.
"""

import math

LEAF_OPS = ("eq", "ne", "lt", "le", "gt", "ge")
_NAN = float("nan")


class KrillError(Exception):
    """Invalid predicate structure."""


class MissingFieldError(Exception):
    """A field referenced by the predicate is absent from the record."""


def _validate(pred):
    if not isinstance(pred, dict):
        raise KrillError("predicate must be an object")
    if len(pred) == 0:
        return
    if len(pred) != 1:
        raise KrillError("predicate must have exactly one key")
    (key, val), = pred.items()
    if key in ("and", "or"):
        if not isinstance(val, list) or len(val) == 0:
            raise KrillError('"%s" requires a non-empty array' % key)
        for sub in val:
            _validate(sub)
        return
    if key not in LEAF_OPS:
        raise KrillError('predicate %s: unknown operator "%s"'
                         % (js_inspect(pred), key))
    if (not isinstance(val, list) or len(val) != 2
            or not isinstance(val[0], str)):
        raise KrillError('"%s" requires [fieldname, value]' % key)
    v = val[1]
    if not (v is None or isinstance(v, (str, bool, int, float))):
        raise KrillError('"%s" value must be a scalar' % key)


def to_number(v):
    """JavaScript ToNumber for the value types that can appear here."""
    if v is None:
        return 0.0
    if isinstance(v, bool):
        return 1.0 if v else 0.0
    if isinstance(v, (int, float)):
        return float(v)
    if isinstance(v, str):
        s = v.strip()
        if s == "":
            return 0.0
        if s in ("Infinity", "+Infinity"):
            return math.inf
        if s == "-Infinity":
            return -math.inf
        low = s.lower()
        # Python's float() accepts forms JS rejects.
        if low in ("nan", "inf", "+inf", "-inf", "infinity",
                   "+infinity", "-infinity"):
            return _NAN
        if low.startswith("0x"):
            try:
                return float(int(s, 16))
            except ValueError:
                return _NAN
        if low.startswith("-0x") or low.startswith("+0x"):
            # JS Number() rejects signed hex ('-0x10' -> NaN)
            return _NAN
        if low.endswith("j") or "_" in s:
            return _NAN
        try:
            return float(s)
        except ValueError:
            return _NAN
    return _NAN  # objects/arrays: ToPrimitive not modeled


def loose_eq(a, b):
    """JavaScript `==` over the scalar types JSON can carry."""
    if a is None or b is None:
        return a is None and b is None
    if isinstance(a, bool):
        return loose_eq(1 if a else 0, b)
    if isinstance(b, bool):
        return loose_eq(a, 1 if b else 0)
    a_num = isinstance(a, (int, float))
    b_num = isinstance(b, (int, float))
    if a_num and b_num:
        return float(a) == float(b)
    if isinstance(a, str) and isinstance(b, str):
        return a == b
    if a_num and isinstance(b, str):
        n = to_number(b)
        return not math.isnan(n) and float(a) == n
    if b_num and isinstance(a, str):
        n = to_number(a)
        return not math.isnan(n) and float(b) == n
    return False  # object/array operands


def _relational(a, b, op):
    """JavaScript relational (<, <=, >, >=)."""
    if isinstance(a, str) and isinstance(b, str):
        if op == "lt":
            return a < b
        if op == "le":
            return a <= b
        if op == "gt":
            return a > b
        return a >= b
    x = to_number(a)
    y = to_number(b)
    if math.isnan(x) or math.isnan(y):
        return False
    if op == "lt":
        return x < y
    if op == "le":
        return x <= y
    if op == "gt":
        return x > y
    return x >= y


def js_inspect(obj):
    """Node util.inspect rendering of a predicate (the reference's
    error messages embed it: tst.badargs.sh.out:9)."""
    if obj is None:
        return "null"
    if obj is True:
        return "true"
    if obj is False:
        return "false"
    if isinstance(obj, str):
        return "'" + obj.replace("\\", "\\\\").replace("'", "\\'") + "'"
    if isinstance(obj, (int, float)):
        return js_num_repr(obj)
    if isinstance(obj, list):
        if not obj:
            return "[]"
        return "[ " + ", ".join(js_inspect(x) for x in obj) + " ]"
    if isinstance(obj, dict):
        if not obj:
            return "{}"
        return "{ " + ", ".join(
            "%s: %s" % (k, js_inspect(v)) for k, v in obj.items()) + " }"
    return repr(obj)


def js_num_repr(v):
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, int) or (isinstance(v, float) and v.is_integer()):
        return str(int(v))
    return repr(v)


def pluck(fields, path):
    """Look up a dotted path in a nested dict.

    Returns the sentinel MISSING if any step is absent or a non-dict is
    traversed (mirrors jsprim.pluck returning undefined,
    reference lib/stream-synthetic.js:47).
    """
    cur = fields
    for part in path.split("."):
        if not isinstance(cur, dict) or part not in cur:
            return MISSING
        cur = cur[part]
    return cur


class _Missing(object):
    __slots__ = ()

    def __repr__(self):
        return "<missing>"


MISSING = _Missing()


class Predicate(object):
    """A compiled predicate: eval / fields / SQL rendering."""

    def __init__(self, pred):
        _validate(pred)
        self.pred = pred
        self._fields = []
        self._collect_fields(pred, self._fields)

    @staticmethod
    def _collect_fields(pred, out):
        if len(pred) == 0:
            return
        (key, val), = pred.items()
        if key in ("and", "or"):
            for sub in val:
                Predicate._collect_fields(sub, out)
        else:
            if val[0] not in out:
                out.append(val[0])

    def fields(self):
        return list(self._fields)

    def eval(self, fields):
        """Evaluate against a record's fields dict.

        Field references are dotted paths into the (possibly nested)
        record.  Raises MissingFieldError when a referenced field is
        absent (callers drop the record, counting 'nfailedeval').
        """
        return self._eval(self.pred, fields)

    def _eval(self, pred, fields):
        if len(pred) == 0:
            return True
        (key, val), = pred.items()
        if key == "and":
            return all(self._eval(sub, fields) for sub in val)
        if key == "or":
            return any(self._eval(sub, fields) for sub in val)
        fieldval = pluck(fields, val[0])
        if fieldval is MISSING:
            raise MissingFieldError(val[0])
        const = val[1]
        if key == "eq":
            return loose_eq(fieldval, const)
        if key == "ne":
            return not loose_eq(fieldval, const)
        return _relational(fieldval, const, key)

    # -- SQL rendering (for index queries; reference lib/index-query.js
    #    renders leaves via krill's toCStyleString at :453) --

    _SQL_OPS = {"eq": "=", "ne": "<>", "lt": "<", "le": "<=",
                "gt": ">", "ge": ">="}

    def to_sql(self, name_map=None):
        """Render as a SQLite WHERE expression.

        name_map optionally maps field names to column names.
        """
        return self._sql(self.pred, name_map or {})

    def _sql(self, pred, name_map):
        if len(pred) == 0:
            return "1"
        (key, val), = pred.items()
        if key in ("and", "or"):
            joiner = " AND " if key == "and" else " OR "
            return joiner.join(
                "(%s)" % self._sql(sub, name_map) for sub in val)
        field = name_map.get(val[0], val[0])
        col = '"%s"' % field.replace('"', '""')
        const = val[1]
        if const is None:
            if key == "eq":
                return "%s IS NULL" % col
            if key == "ne":
                return "%s IS NOT NULL" % col
            const = 0
        if isinstance(const, bool):
            const = 1 if const else 0
        if isinstance(const, str):
            lit = "'%s'" % const.replace("'", "''")
        elif isinstance(const, int):
            lit = str(const)
        else:
            lit = repr(float(const))
        return "%s %s %s" % (col, self._SQL_OPS[key], lit)


def create_predicate(pred):
    """Compile a predicate; raises KrillError on invalid structure."""
    return Predicate(pred)


def filter_and(*filters):
    """Combine N filter JSON values with AND, ignoring Nones.

    (reference lib/dragnet-impl.js:332-343)
    """
    fs = [f for f in filters if f is not None]
    if not fs:
        return None
    if len(fs) == 1:
        return fs[0]
    return {"and": fs}
