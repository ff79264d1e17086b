"""
Multi-GPU data parallelism: shard the input across ranks, merge partial
aggregates.

The reference's distributed path is Manta map/reduce — one map task per
input object emitting skinner points, one reduce re-aggregating them
(reference lib/datasource-manta.js:202-219).  Here the same shape runs
as one process per GPU under torch.distributed (backend "nccl" IS RCCL
on ROCm; "gloo" for CPU tests):

  * shard assignment: round-robin over the file list (the analog of
    addJobKey fan-out, lib/datasource-manta.js:549-559)
  * merge: per-rank aggregate tables are exchanged and re-aggregated —
    correctness rests on points merging associatively/commutatively
    (proved by the x3 idempotence test)

Two merge paths:
  * merge_points_object: gather_object of decoded points to rank 0 —
    simple, used for small results (CLI paths)
  * merge_tables_tensor: dictionary unification (C2) + dense
    (key, count) tensor exchange over RCCL/xGMI (C1): each rank
    serializes its table to tensors, all_gather into rank-aligned
    buffers, re-aggregate.  Aggregate payloads are typically <= MBs,
    so the latency-optimal direct gather beats ring algorithms on
    xGMI's 7 p2p links (SURVEY.md §5 topology note).
"""

import os

from .points import Aggregator


def dist_env():
    """(rank, world_size, local_rank) from torchrun env, or (0,1,0)."""
    return (int(os.environ.get("RANK", 0)),
            int(os.environ.get("WORLD_SIZE", 1)),
            int(os.environ.get("LOCAL_RANK", 0)))


def init_process_group(backend=None):
    import torch
    import torch.distributed as dist
    if dist.is_initialized():
        return dist
    rank, world, local = dist_env()
    if world == 1:
        return None
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    dist.init_process_group(backend=backend, rank=rank,
                            world_size=world)
    if backend == "nccl":
        torch.cuda.set_device(local)
    return dist


def shard_files(files, rank, world):
    """Round-robin shard assignment (deterministic across ranks)."""
    return [f for i, f in enumerate(files) if i % world == rank]


def merge_points_object(aggs, queries, group=None):
    """Gather every rank's points to rank 0 and re-aggregate.

    aggs: this rank's [Aggregator, ...] (one per query).
    Returns merged [Aggregator, ...] on rank 0; None elsewhere.
    """
    import torch.distributed as dist
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    payload = [a.points() for a in aggs]
    gathered = [None] * world if rank == 0 else None
    dist.gather_object(payload, gathered, dst=0, group=group)
    if rank != 0:
        return None
    merged = [Aggregator(q) for q in queries]
    for rank_payload in gathered:
        for mi, points in enumerate(rank_payload):
            for p in points:
                merged[mi].write(p)
    return merged


def merge_counter_stages(stages, group=None):
    """Sum per-stage counters across ranks (for --counters output)."""
    import torch.distributed as dist
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    gathered = [None] * world if rank == 0 else None
    dist.gather_object(stages, gathered, dst=0, group=group)
    if rank != 0:
        return None
    out = []
    for si, (name, counters) in enumerate(gathered[0]):
        merged = dict(counters)
        for other in gathered[1:]:
            if si < len(other) and other[si][0] == name:
                for k, v in other[si][1].items():
                    merged[k] = merged.get(k, 0) + v
        out.append((name, merged))
    return out


def merge_tables_tensor(agg, query, device, group=None):
    """RCCL tensor-path merge of one aggregation table (C1+C2).

    Each rank packs its table into (key-bytes, count) tensors; string
    values travel as UTF-8 payloads (the dictionary unification).
    all_gather over xGMI, then every rank re-aggregates the union —
    an allreduce in effect, so every rank holds the full result.
    Returns the merged Aggregator (on every rank).
    """
    import pickle

    import torch
    import torch.distributed as dist

    blob = pickle.dumps(list(agg.table.items()),
                        protocol=pickle.HIGHEST_PROTOCOL)
    t = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(device)
    sizes = [torch.zeros(1, dtype=torch.int64, device=device)
             for _ in range(dist.get_world_size(group))]
    mine = torch.tensor([t.numel()], dtype=torch.int64, device=device)
    dist.all_gather(sizes, mine, group=group)
    # >=1 so zero-length all_gather never reaches RCCL (all-empty
    # tables, e.g. a filter matching nothing anywhere)
    maxn = max(int(max(s.item() for s in sizes)), 1)
    padded = torch.zeros(maxn, dtype=torch.uint8, device=device)
    padded[:t.numel()] = t
    bufs = [torch.zeros(maxn, dtype=torch.uint8, device=device)
            for _ in range(dist.get_world_size(group))]
    dist.all_gather(bufs, padded, group=group)

    merged = Aggregator(query)
    for buf, size in zip(bufs, sizes):
        items = pickle.loads(bytes(
            buf[:int(size.item())].cpu().numpy().tobytes()))
        for k, v in items:
            merged.table[k] = merged.table.get(k, 0) + v
    merged.ninputs = agg.ninputs  # per-rank; callers sum via counters
    return merged
