"""
Multi-GPU data parallelism: shard the input across ranks, merge partial
aggregates over RCCL/xGMI.

The reference's distributed path is Manta map/reduce — one map task per
input object emitting skinner points, one reduce re-aggregating them
(reference lib/datasource-manta.js:202-219).  Here the same shape runs
as one process per GPU under torch.distributed (backend "nccl" IS RCCL
on ROCm; "gloo" for CPU tests):

  * shard assignment: round-robin over the file list (the analog of
    addJobKey fan-out, lib/datasource-manta.js:549-559)
  * merge: per-rank aggregate tables are exchanged as DENSE TENSORS and
    re-aggregated — correctness rests on points merging associatively/
    commutatively (proved by the x3 idempotence test)

The merge (C1+C2 of SURVEY.md §2c) is fully tensor-typed — no object
serialization crosses the wire:

  C2 dictionary unification: each rank's string dictionary travels as a
     UTF-8 byte blob + offset tensor; every rank builds the global
     dictionary and remaps its peers' string codes into it.
  C1 table exchange: each rank's table flattens to an int64 key-code
     matrix [n, nk] (per-element tag: int / string-id / f64-bits packed
     into a companion column) and an f64 value vector.  Payloads are
     typically <= MBs, so the latency-optimal direct all_gather beats
     ring algorithms on xGMI's 7 p2p links (SURVEY.md §5 topology
     note); every rank re-aggregates the union (allreduce semantics),
     vectorized via np.unique row grouping + bincount.
  Pathological cardinality (> _PARTITION_ROWS local rows): hash-
     partition the keys with all_to_all_single so each rank owns and
     merges one key range (reduce-scatter shape), then all_gather the
     merged partitions.
"""

import os

import numpy as np

from .points import Aggregator

# per-element tag values in the packed tag column
_EL_INT = 0   # int64 payload (ordinals, dates, integral counts)
_EL_STR = 1   # payload = index into the rank's string dictionary
_EL_F64 = 2   # payload = IEEE-754 bits of a float key element

_PARTITION_ROWS = 1 << 20  # switch to hash-partition all_to_all above


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
        # DRAGNET_DIST_BACKEND=gloo lets N ranks rehearse on one GPU
        # (RCCL refuses duplicate devices); production defaults to
        # RCCL when CUDA/HIP devices are visible
        backend = os.environ.get("DRAGNET_DIST_BACKEND") or \
            ("nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    if backend == "nccl":
        torch.cuda.set_device(local)
    dist.init_process_group(backend=backend, rank=rank,
                            world_size=world)
    return dist


def shard_files(files, rank, world):
    """Round-robin shard assignment (deterministic across ranks)."""
    return [f for i, f in enumerate(files) if i % world == rank]


def _coll_device(group=None):
    """Device collectives must use: CUDA for RCCL, CPU for gloo."""
    import torch
    import torch.distributed as dist
    if dist.get_backend(group) == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


# ---- dense table encode/decode ----------------------------------------

def _encode_table(agg, nk):
    """Flatten an Aggregator's table to dense arrays:
    (codes i64 [n, nk], tagpack i64 [n], vals f64 [n], strings list).
    Key elements are int (ordinal/date), str, or (defensively) float;
    strings are dictionary-encoded per rank (C2 unifies them)."""
    n = len(agg.table)
    codes = np.zeros((n, nk), dtype=np.int64)
    tagpack = np.zeros(n, dtype=np.int64)
    vals = np.empty(n, dtype=np.float64)
    strings = []
    sidx = {}
    for i, (k, v) in enumerate(agg.table.items()):
        vals[i] = v
        tp = 0
        for j, el in enumerate(k):
            if isinstance(el, str):
                si = sidx.get(el)
                if si is None:
                    si = sidx[el] = len(strings)
                    strings.append(el)
                codes[i, j] = si
                tp |= _EL_STR << (2 * j)
            elif isinstance(el, float) and not el.is_integer():
                codes[i, j] = np.float64(el).view(np.int64)
                tp |= _EL_F64 << (2 * j)
            else:
                iv = int(el)
                if -(1 << 63) <= iv < (1 << 63):
                    codes[i, j] = iv
                else:
                    # lquantize ordinals of absurd magnitudes exceed
                    # int64: ship as f64 bits (JS-float key semantics
                    # — the reference's ordinals are floats there
                    # anyway); every rank rebuilds the same float key
                    codes[i, j] = np.float64(float(iv)).view(np.int64)
                    tp |= _EL_F64 << (2 * j)
        tagpack[i] = tp
    return codes, tagpack, vals, strings


def _strings_blob(strings):
    bs = [s.encode("utf-8") for s in strings]
    offs = np.zeros(len(bs) + 1, dtype=np.int64)
    if bs:
        np.cumsum([len(b) for b in bs], out=offs[1:])
    return b"".join(bs), offs


def _decode_strings(blob_bytes, offs):
    return [blob_bytes[offs[i]:offs[i + 1]].decode("utf-8")
            for i in range(len(offs) - 1)]


def _rebuild_table(query, codes, tagpack, vals, strings):
    """Vectorized re-aggregation: group identical key rows with
    np.unique, sum values with bincount, decode each UNIQUE key to the
    Aggregator's canonical Python tuple once."""
    merged = Aggregator(query)
    nk = len(query.breakdowns)
    if nk == 0:
        total = float(vals.sum()) if vals.size else 0.0
        if vals.size:
            merged.table[()] = int(total) if total.is_integer() \
                else total
        return merged
    X = np.concatenate([codes, tagpack[:, None]], axis=1)
    uniq, inv = np.unique(X, axis=0, return_inverse=True)
    sums = np.bincount(inv, weights=vals, minlength=uniq.shape[0])
    for r in range(uniq.shape[0]):
        tp = int(uniq[r, nk])
        key = []
        for j in range(nk):
            tag = (tp >> (2 * j)) & 3
            c = int(uniq[r, j])
            if tag == _EL_STR:
                key.append(strings[c])
            elif tag == _EL_F64:
                key.append(float(np.int64(c).view(np.float64)))
            else:
                key.append(c)
        v = float(sums[r])
        merged.table[tuple(key)] = int(v) if v.is_integer() else v
    return merged


# ---- collectives -------------------------------------------------------

def _all_gather_padded(t, sizes, device, group):
    """all_gather a variable-length 1-D tensor: pad to the global max
    (>=1 so zero-length never reaches RCCL), return per-rank tensors
    trimmed to their true sizes."""
    import torch
    import torch.distributed as dist
    world = dist.get_world_size(group)
    maxn = max(max(sizes), 1)
    padded = torch.zeros(maxn, dtype=t.dtype, device=device)
    if t.numel():
        padded[:t.numel()] = t
    bufs = [torch.empty(maxn, dtype=t.dtype, device=device)
            for _ in range(world)]
    dist.all_gather(bufs, padded, group=group)
    return [b[:sizes[r]] for r, b in enumerate(bufs)]


def merge_tables_tensor(agg, query, device=None, group=None):
    """Dense RCCL merge of one aggregation table (C1+C2); every rank
    returns the fully-merged Aggregator (allreduce semantics).  See
    module docstring for the wire format; no object serialization."""
    import torch
    import torch.distributed as dist

    if device is None:
        device = _coll_device(group)
    nk = len(query.breakdowns)
    codes, tagpack, vals, strings = _encode_table(agg, nk)
    blob, soffs = _strings_blob(strings)

    world = dist.get_world_size(group)
    meta = torch.tensor([codes.shape[0], len(strings), len(blob)],
                        dtype=torch.int64, device=device)
    metas = [torch.empty(3, dtype=torch.int64, device=device)
             for _ in range(world)]
    dist.all_gather(metas, meta, group=group)
    metas = torch.stack(metas).cpu().numpy()
    ns_rows = metas[:, 0]
    ns_strs = metas[:, 1]
    ns_blob = metas[:, 2]
    # path choice is COLLECTIVE (computed from the gathered sizes, so
    # every rank takes the same branch): pathological cardinality goes
    # through the hash-partition reduce-scatter shape instead of
    # shipping every row everywhere
    if int(ns_rows.max()) > _PARTITION_ROWS:
        return _merge_partitioned(query, codes, tagpack, vals, strings,
                                  device, group, agg)

    def dev(arr, dtype):
        t = torch.from_numpy(np.ascontiguousarray(arr))
        return t.to(device=device, dtype=dtype)

    g_codes = _all_gather_padded(
        dev(codes.reshape(-1), torch.int64), list(ns_rows * nk),
        device, group)
    g_tags = _all_gather_padded(
        dev(tagpack, torch.int64), list(ns_rows), device, group)
    g_vals = _all_gather_padded(
        dev(vals, torch.float64), list(ns_rows), device, group)
    g_soffs = _all_gather_padded(
        dev(soffs, torch.int64), list(ns_strs + 1), device, group)
    g_blob = _all_gather_padded(
        dev(np.frombuffer(blob, dtype=np.uint8).copy(), torch.uint8),
        list(ns_blob), device, group)

    # C2: global dictionary + per-rank remap, then concatenate
    gdict = {}
    gstrings = []
    all_codes = []
    all_tags = []
    all_vals = []
    for r in range(world):
        sb = bytes(g_blob[r].cpu().numpy().tobytes())
        so = g_soffs[r].cpu().numpy()
        rstr = _decode_strings(sb, so)
        remap = np.empty(max(len(rstr), 1), dtype=np.int64)
        for i, s in enumerate(rstr):
            gi = gdict.get(s)
            if gi is None:
                gi = gdict[s] = len(gstrings)
                gstrings.append(s)
            remap[i] = gi
        rc = g_codes[r].cpu().numpy().reshape(-1, nk).copy() \
            if nk else np.zeros((int(ns_rows[r]), 0), dtype=np.int64)
        rt = g_tags[r].cpu().numpy()
        for j in range(nk):
            is_str = ((rt >> (2 * j)) & 3) == _EL_STR
            if is_str.any():
                rc[is_str, j] = remap[rc[is_str, j]]
        all_codes.append(rc)
        all_tags.append(rt)
        all_vals.append(g_vals[r].cpu().numpy())

    merged = _rebuild_table(
        query,
        np.concatenate(all_codes) if all_codes else
        np.zeros((0, nk), np.int64),
        np.concatenate(all_tags), np.concatenate(all_vals), gstrings)
    merged.ninputs = agg.ninputs  # per-rank; callers sum via counters
    merged.ndropped_nonnumeric = agg.ndropped_nonnumeric
    return merged


def _merge_partitioned(query, codes, tagpack, vals, strings, device,
                       group, agg):
    """Pathological-cardinality path: hash-partition key rows across
    ranks with all_to_all_single (reduce-scatter shape — each rank
    merges one key range), then all_gather the merged partitions so
    every rank still ends with the full table.

    Strings cannot be hashed consistently pre-unification, so the
    global dictionary is built FIRST (strings are few relative to
    rows in any realistic pathological-cardinality table), then rows
    are partitioned on their remapped codes.
    """
    import torch
    import torch.distributed as dist

    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    nk = len(query.breakdowns)

    # C2 first: unify dictionaries globally
    blob, soffs = _strings_blob(strings)
    meta = torch.tensor([len(strings), len(blob)], dtype=torch.int64,
                        device=device)
    metas = [torch.empty(2, dtype=torch.int64, device=device)
             for _ in range(world)]
    dist.all_gather(metas, meta, group=group)
    metas = torch.stack(metas).cpu().numpy()

    def dev(arr, dtype):
        return torch.from_numpy(np.ascontiguousarray(arr)).to(
            device=device, dtype=dtype)

    g_soffs = _all_gather_padded(dev(soffs, torch.int64),
                                 list(metas[:, 0] + 1), device, group)
    g_blob = _all_gather_padded(
        dev(np.frombuffer(blob, dtype=np.uint8).copy(), torch.uint8),
        list(metas[:, 1]), device, group)
    gdict = {}
    gstrings = []
    remaps = []
    for r in range(world):
        sb = bytes(g_blob[r].cpu().numpy().tobytes())
        rstr = _decode_strings(sb, g_soffs[r].cpu().numpy())
        remap = np.empty(max(len(rstr), 1), dtype=np.int64)
        for i, s in enumerate(rstr):
            gi = gdict.get(s)
            if gi is None:
                gi = gdict[s] = len(gstrings)
                gstrings.append(s)
            remap[i] = gi
        remaps.append(remap)
    for j in range(nk):
        is_str = ((tagpack >> (2 * j)) & 3) == _EL_STR
        if is_str.any():
            codes[is_str, j] = remaps[rank][codes[is_str, j]]

    # partition rows by key hash
    h = np.zeros(codes.shape[0], dtype=np.uint64)
    for j in range(nk):
        cj = np.ascontiguousarray(codes[:, j]).view(np.uint64)
        h = (h ^ cj) * np.uint64(0x100000001B3)
    h ^= tagpack.view(np.uint64)
    part = (h % np.uint64(world)).astype(np.int64)
    order = np.argsort(part, kind="stable")
    counts = np.bincount(part, minlength=world)

    row = np.concatenate(
        [codes, tagpack[:, None], vals.view(np.int64)[:, None]],
        axis=1)[order]  # [n, nk+2] i64 rows, grouped by target rank
    send = dev(row.reshape(-1), torch.int64)
    in_counts = torch.from_numpy(counts * (nk + 2)).to(device)
    out_counts = torch.empty_like(in_counts)
    dist.all_to_all_single(out_counts, in_counts, group=group)
    recv = torch.empty(int(out_counts.sum().item()), dtype=torch.int64,
                       device=device)
    dist.all_to_all_single(
        recv, send, list(out_counts.cpu().numpy()),
        list(in_counts.cpu().numpy()), group=group)
    mine = recv.cpu().numpy().reshape(-1, nk + 2)

    # merge my partition, then share the merged partitions
    my_codes = mine[:, :nk]
    my_tags = mine[:, nk]
    my_vals = mine[:, nk + 1].view(np.float64)
    X = np.concatenate([my_codes, my_tags[:, None]], axis=1)
    uniq, inv = np.unique(X, axis=0, return_inverse=True)
    sums = np.bincount(inv, weights=my_vals, minlength=uniq.shape[0])
    out = np.concatenate([uniq, sums.view(np.int64)[:, None]], axis=1)

    sz = torch.tensor([out.shape[0] * (nk + 2)], dtype=torch.int64,
                      device=device)
    szs = [torch.empty(1, dtype=torch.int64, device=device)
           for _ in range(world)]
    dist.all_gather(szs, sz, group=group)
    szs = [int(s.item()) for s in szs]
    parts = _all_gather_padded(dev(out.reshape(-1), torch.int64), szs,
                               device, group)
    allrows = np.concatenate(
        [p.cpu().numpy().reshape(-1, nk + 2) for p in parts])
    merged = _rebuild_table(
        query, allrows[:, :nk], allrows[:, nk],
        allrows[:, nk + 1].view(np.float64), gstrings)
    merged.ninputs = agg.ninputs
    merged.ndropped_nonnumeric = agg.ndropped_nonnumeric
    return merged


def merge_aggregators(aggs, queries, device=None, group=None):
    """Dense merge of every query's table; every rank returns the
    fully-merged [Aggregator, ...]."""
    return [merge_tables_tensor(a, q, device=device, group=group)
            for a, q in zip(aggs, queries)]


def merge_points_object(aggs, queries, group=None):
    """Rank-0 merge view (legacy name): dense tensor merge under the
    hood; returns merged [Aggregator, ...] on rank 0, None elsewhere."""
    import torch.distributed as dist
    merged = merge_aggregators(aggs, queries, group=group)
    return merged if dist.get_rank(group) == 0 else None


def merge_counter_stages(stages, group=None):
    """Sum per-stage counters across ranks (for --counters output).

    Every rank runs the identical pipeline, so the stage list shape
    and per-stage counter keys are identical everywhere; the values
    ride in one int64 tensor through all_reduce(SUM) — no object
    serialization.  Every rank returns the merged stages."""
    import torch
    import torch.distributed as dist
    device = _coll_device(group)
    vals = []
    for _name, c in stages:
        vals.extend(int(c[k]) for k in c)
    t = torch.tensor(vals or [0], dtype=torch.int64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    flat = t.cpu().numpy()
    out = []
    i = 0
    for name, c in stages:
        merged = {}
        for k in c:
            merged[k] = int(flat[i])
            i += 1
        out.append((name, merged))
    return out
