"""
validate-schema: validate an input JSON file against one of the
user-facing schemas (reference tools/validate-schema).

    python -m dragnet_amd.tools.validate_schema SCHEMA_NAME INPUT_FILE
"""

import json
import sys

from ..schema import SCHEMAS, SchemaError, validate

ARG0 = "validate-schema"


def usage():
    sys.stderr.write("usage: %s SCHEMA_NAME INPUT_FILE\n" % ARG0)
    sys.stderr.write("available schemas: %s\n"
                     % ", ".join(sorted(SCHEMAS)))
    return 2


def main(argv=None):
    argv = sys.argv[1:] if argv is None else argv
    if len(argv) != 2:
        return usage()
    schema_name, input_file = argv
    if schema_name not in SCHEMAS:
        return usage()
    try:
        with open(input_file) as f:
            data = json.load(f)
    except (OSError, ValueError) as e:
        sys.stderr.write('%s: read "%s": %s\n' % (ARG0, input_file, e))
        return 1
    try:
        validate(SCHEMAS[schema_name], data)
    except SchemaError as e:
        sys.stderr.write("%s: %s\n" % (ARG0, e))
        return 1
    print("%s okay" % input_file)
    return 0


if __name__ == "__main__":
    sys.exit(main())
