"""
JSON-schema validation for the user-facing index/query config shapes
(reference schema/user-index.js, schema/user-query.js, schema/common.js
and jsprim.validateJsonObject semantics: draft-v3-style 'required' on
properties, union types, enums).

Only the subset those schemas use is implemented: type (incl. union
lists and inline object schemas inside them), required, enum,
properties, array items.
"""


class SchemaError(Exception):
    pass


def _type_name(v):
    if v is None:
        return "null"
    if isinstance(v, bool):
        return "boolean"
    if isinstance(v, (int, float)):
        return "number"
    if isinstance(v, str):
        return "string"
    if isinstance(v, list):
        return "array"
    if isinstance(v, dict):
        return "object"
    return type(v).__name__


def _check_type(schema_type, value, path):
    """True if value matches schema_type (a name, an inline schema
    object, or a union list of either)."""
    if isinstance(schema_type, list):
        return any(_check_type(t, value, path) for t in schema_type)
    if isinstance(schema_type, dict):
        try:
            _validate(schema_type, value, path)
            return True
        except SchemaError:
            return False
    if schema_type == "any":
        return True
    if schema_type == "integer":
        return (isinstance(value, (int, float))
                and not isinstance(value, bool)
                and float(value).is_integer())
    name = _type_name(value)
    if schema_type == "number":
        return name == "number"
    return name == schema_type


def _validate(schema, value, path="input"):
    st = schema.get("type")
    if st is not None and not _check_type(st, value, path):
        raise SchemaError(
            'property "%s": value has wrong type (expected %s, got %s)'
            % (path, st, _type_name(value)))
    if "enum" in schema and value not in schema["enum"]:
        raise SchemaError(
            'property "%s": unsupported value: %r (expected one of %s)'
            % (path, value, schema["enum"]))
    if isinstance(value, dict):
        for name, sub in schema.get("properties", {}).items():
            sub_path = "%s.%s" % (path, name)
            if name not in value:
                if sub.get("required") or (
                        isinstance(sub, dict)
                        and sub.get("required") is True):
                    raise SchemaError(
                        'property "%s": missing and required'
                        % sub_path)
                continue
            _validate(sub, value[name], sub_path)
    if isinstance(value, list) and "items" in schema:
        for i, item in enumerate(value):
            _validate(schema["items"], item,
                      "%s[%d]" % (path, i))
    return value


def validate(schema, value):
    """Validate `value` against `schema`; raises SchemaError."""
    if schema.get("required") and value is None:
        raise SchemaError("input is required")
    return _validate(schema, value)


def _t(name, required=False):
    rv = {"type": name}
    if required:
        rv["required"] = True
    return rv


def _enum(values, required=False):
    rv = {"type": _type_name(values[0]), "enum": list(values)}
    if required:
        rv["required"] = True
    return rv


# reference schema/user-index.js
USER_INDEX = {
    "type": "object",
    "properties": {
        "name": _t("string", required=True),
        "fsroot": _t("string"),
        "mantaroot": _t("string"),
        "format": _enum(["json"], required=True),
        "filter": {"type": "object"},
        "primaryKey": _t("string"),
        "columns": {
            "type": "array",
            "required": True,
            "items": {
                "type": ["string", {
                    "type": "object",
                    "properties": {
                        "name": _t("string", required=True),
                        "field": _t("string", required=True),
                        "aggr": _enum(["quantize"]),
                    },
                }],
            },
        },
    },
}

# reference schema/user-query.js
USER_QUERY = {
    "type": "object",
    "properties": {
        "index": _t("string", required=True),
        "timeStart": _t("string"),
        "timeEnd": _t("string"),
        "timeResolution": _t("number"),
        "filter": {"type": "object"},
        "breakdowns": {
            "type": "array",
            "items": _t("string"),
        },
    },
}

SCHEMAS = {"user-index": USER_INDEX, "user-query": USER_QUERY}
