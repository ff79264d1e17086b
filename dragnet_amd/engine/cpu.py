"""
CPU scan engine: wraps the oracle pipeline (scan_cpu.py) behind the
engine interface.  One parse pass feeds N per-metric pipelines
(mirroring the reference's one-parse/N-StreamScan fan-out,
lib/datasource-file.js:386-432).
"""

from ..scan_cpu import ScanPipeline


class CpuEngine(object):
    name = "cpu"

    def scan(self, files, queries, ds_filter=None, time_field=None,
             data_format="json", byte_source=None):
        """Run `queries` over the concatenated bytes of `files` (or over
        `byte_source`, an iterable of byte chunks).  Returns a
        ScanResult-compatible object."""
        import os

        from ..datasource.file import ScanResult

        collect = os.environ.get("DRAGNET_WARNINGS") == "1"
        pipelines = [
            ScanPipeline(q, ds_filter=ds_filter, time_field=time_field,
                         data_format=data_format,
                         collect_warnings=collect)
            for q in queries
        ]
        primary = pipelines[0]

        def feed_line(line):
            # Parse once in the primary pipeline; errors are counted
            # there.  Remaining pipelines consume the parsed record
            # (sharing the fields dict is safe: synthetic stages only
            # add name-keyed values that agree across pipelines).
            n_out_before = primary.parser_counters["noutputs"]
            consumed = primary.write_line(line)
            if primary.parser_counters["noutputs"] == n_out_before:
                return  # parse error; already counted
            point = primary.last_point
            for p in pipelines[1:]:
                p.parser_counters["ninputs"] += 1
                p.parser_counters["noutputs"] += 1
                p.write_point({"fields": point["fields"],
                               "value": point["value"]})
            return consumed

        if byte_source is None:
            byte_source = _read_files(files, pipelines=pipelines)

        partial = b""
        for chunk in byte_source:
            data = partial + chunk
            lines = data.split(b"\n")
            partial = lines.pop()
            for line in lines:
                feed_line(line)
        if partial:
            feed_line(partial)

        stages = pipelines[0].counter_stages()
        result = ScanResult([p.aggr for p in pipelines], stages)
        result.warnings = primary.warnings
        return result


def _read_files(files, chunk_size=8 * 1024 * 1024, pipelines=None):
    for path in files:
        # warning context: which file the following records came from
        for p in pipelines or []:
            p.context_file = path
        with open(path, "rb", buffering=0) as f:
            while True:
                chunk = f.read(chunk_size)
                if not chunk:
                    break
                yield chunk
