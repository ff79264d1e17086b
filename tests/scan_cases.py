"""Shared scan test cases (ported from the reference's shared fragment,
tests/dn/scan_testcases.sh:1-40) — run against multiple backends/modes
with golden outputs."""

SCAN_CASES = [
    # Count everything.
    (),
    # Break down results by operation.
    ("-b", "operation"),
    # Nested property + host.
    ("-b", "operation,req.method,host"),
    # Nullable / undefined field.
    ("-b", "req.caller"),
    ("-b", "operation,req.caller"),
    # Count filtered on request method.
    ("-f", '{ "eq": [ "req.method", "GET" ] }'),
    ("-f", '{ "eq": [ "req.method", "GET" ] }',
     "-b", "operation,req.method,host"),
    # Filter on nullable field.
    ("-f", '{ "eq": [ "req.caller", "poseidon" ] }'),
    ("-f", '{ "eq": [ "req.caller", "poseidon" ] }', "-b", "req.caller"),
    # Quantization alone (histogram).
    ("-b", "latency[aggr=quantize]"),
    # Quantization followed by normal fields: table.
    ("-b", "latency[aggr=quantize],operation,host"),
    # Ends with quantization: grouped histograms.
    ("-b", "host,operation,latency[aggr=quantize]"),
    # Linear quantization.
    ("-b", "latency[aggr=lquantize,step=100]"),
]

# Cases whose aggregation spec matches the index big_metric exactly
# (the lquantize case re-buckets quantized minimums and is not
# scan-equivalent; the reference's index goldens differ there too).
INDEX_EQUIV_CASES = SCAN_CASES[:-1]
