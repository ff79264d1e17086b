"""
Sharded datasource: the distributed (Manta-analog) backend.

Where the reference fans a scan out as one Manta map task per input
object with a reduce phase re-aggregating points
(reference lib/datasource-manta.js:151-219), this backend fans the file
list out across the ranks of a torch.distributed world — one process
per GPU over RCCL/xGMI — and merges per-rank aggregates.

Run it under torchrun (one rank per GPU) or single-process (world=1,
where it behaves like the file backend).  Rank 0 returns results;
other ranks return None (the CLI suppresses their output).
"""


from ..distributed import (dist_env, init_process_group,
                           merge_aggregators, merge_counter_stages,
                           shard_files)
from .file import FileDatasource, ScanResult


def _patch_agg_stage(stages, merged_aggs):
    """After the cross-rank merge, the Aggregator stage's noutputs is
    the MERGED group count, not the sum of per-rank counts."""
    out = []
    for name, c in stages:
        if name == "Aggregator" and merged_aggs:
            c = dict(c)
            c["noutputs"] = merged_aggs[0].noutputs()
        out.append((name, c))
    return out


class ShardedDatasource(FileDatasource):
    def __init__(self, ds, engine=None):
        super().__init__(ds, engine=engine)
        self.rank, self.world, self.local_rank = dist_env()
        self.dist = init_process_group() if self.world > 1 else None

    def _shard(self, files):
        if self.world <= 1:
            return files
        return shard_files(files, self.rank, self.world)

    def scan(self, query, dry_run=False, out=None):
        from ..fsfind import FindCounters
        counters = FindCounters()
        files = list(self._find_data(query, counters=counters))
        if dry_run:
            if self.rank == 0:
                import sys
                o = out or sys.stderr
                o.write("would scan files (%d shards x %d ranks):\n"
                        % (len(files), self.world))
                for path, _ in files:
                    o.write("    %s\n" % path)
            return None
        mine = self._shard([p for p, _ in files])
        result = self.engine().scan(
            files=mine, queries=[query],
            ds_filter=self.ds.filter,
            time_field=self.ds.time_field,
            data_format=self.ds.data_format)
        if self.world <= 1:
            result.stages = counters.stages() + result.stages
            return result
        merged = merge_aggregators(result.aggregators, [query])
        stages = _patch_agg_stage(
            merge_counter_stages(result.stages), merged)
        if self.rank != 0:
            return ScanResult([], [], nonroot=True)
        return ScanResult(merged, counters.stages() + stages,
                          [p for p, _ in files])

    def scan_multi(self, queries, after_ms, before_ms, dry_run=False,
                   out=None):
        from ..fsfind import FindCounters
        from ..query import QueryConfig
        from .file import _ms_to_iso
        pseudo = QueryConfig(time_after=_ms_to_iso(after_ms),
                             time_before=_ms_to_iso(before_ms))
        counters = FindCounters()
        files = list(self._find_data(pseudo, counters=counters))
        if dry_run:
            if self.rank == 0:
                import sys
                o = out or sys.stderr
                o.write("would scan files:\n")
                for path, _ in files:
                    o.write("    %s\n" % path)
            return None
        mine = self._shard([p for p, _ in files])
        result = self.engine().scan(
            files=mine, queries=queries,
            ds_filter=self.ds.filter,
            time_field=self.ds.time_field,
            data_format=self.ds.data_format)
        if self.world <= 1:
            result.stages = counters.stages() + result.stages
            return result
        merged = merge_aggregators(result.aggregators, queries)
        stages = _patch_agg_stage(
            merge_counter_stages(result.stages), merged)
        if self.rank != 0:
            # every rank holds the full merge (allreduce semantics);
            # build() overrides consume it for partitioned index writes
            return ScanResult(merged, stages, nonroot=True)
        return ScanResult(merged, counters.stages() + stages,
                          [p for p, _ in files])

    def build(self, metrics, interval="day", after_ms=None,
              before_ms=None, dry_run=False):
        """Distributed build: every rank scans its shard (the map
        phase); the dense RCCL merge gives EVERY rank the full
        aggregate, so the reduce phase — materializing the index
        tree — is partitioned by interval bucket across ranks
        (disjoint files written in parallel; rank 0 returns the full
        tree listing)."""
        from .file import write_index
        if not self.ds.index_path:
            raise ValueError(
                'datasource is missing "indexPath" for index operations')
        queries = self.metric_queries(metrics, interval, after_ms,
                                      before_ms)
        result = self.scan_multi(queries, after_ms, before_ms,
                                 dry_run=dry_run)
        if result is None:
            return None
        points = []
        for qi, agg in enumerate(result.aggregators):
            for p in agg.points():
                p["fields"]["__dn_metric"] = qi
                points.append(p)
        partition = (self.rank, self.world) if self.world > 1 else None
        written = write_index(self.ds.index_path, metrics, interval,
                              points, partition=partition)
        if self.dist is not None:
            self.dist.barrier()  # all files in place before any return
        return None if result.nonroot else written

    def query(self, query, interval="day", dry_run=False, out=None):
        """Index queries run on rank 0 only (indexes are tiny relative
        to raw data; the reference also queries indexes serially)."""
        if self.world > 1 and self.rank != 0:
            return ScanResult([], [], nonroot=True)
        return super().query(query, interval=interval, dry_run=dry_run,
                             out=out)
