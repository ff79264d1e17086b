"""
dragnet_amd: an MI355X-native event-stream analytics engine.

A from-scratch rebuild of the capabilities of TritonDataCenter/dragnet
(reference surveyed in SURVEY.md): the `dn` CLI scans newline-separated-JSON
event logs, filters them with a JSON predicate language (see krill.py),
aggregates counts grouped by (possibly nested, possibly bucketized) fields,
materializes aggregates into per-interval SQLite index files, and answers
later queries from the indexes.

The hot path (NDJSON tokenize -> predicate filter -> date parse -> bucketize
-> multi-field hash-aggregate) runs as a single fused hand-written HIP/CDNA4
kernel on MI355X GPUs (dragnet_amd/ops/hip/), with data-parallel fan-out
across the 8 GPUs of a node and RCCL merge of per-GPU partial aggregates
over xGMI.  A pure-Python CPU oracle (scan_cpu.py) implements identical
semantics and is the differential-test reference for every kernel.
"""

__version__ = "0.1.0"
