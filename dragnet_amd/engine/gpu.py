"""
MI355X GPU scan engine — host orchestration.

Streams NDJSON bytes through the fused CDNA4 scan kernel
(ops/hip/scan_kernels.hip): chunked reads into pinned host buffers,
async H2D on the current HIP stream, torch-side newline indexing, one
fused kernel launch per chunk, then table/dictionary extraction and
host-side decode into the same canonical aggregate representation the
CPU oracle produces (points.Aggregator) — so every downstream consumer
(formatters, index sink, RCCL merge) is engine-agnostic.

Capacity handling: the hash tables and dictionaries are sized up front;
on overflow (C_OVERFLOW counter) the whole scan is retried with 8x the
capacity (aggregation is a pure function of the input, so a restart is
correct).
"""

import json
import os

import numpy as np

from ..points import Aggregator
from . import plan as planmod

COUNTER_NAMES = ["lines", "invalid_json", "parsed", "ds_filtered",
                 "ds_failedeval", "overflow"]
C_GLOBAL_N = 8
CM_N = 8
(CM_FILTER_IN, CM_FILTERED, CM_FAILEDEVAL, CM_UNDEF, CM_BADDATE,
 CM_TIME_OUT, CM_AGG_IN, CM_NONNUMERIC) = range(8)

MAX_KEY = 8


def _pad(n):
    """Pad a chunk length so the cursor's 8-byte windows never read
    past the buffer (>= 8 bytes of newline slack, 16B aligned)."""
    return (n + 16 + 15) & ~15


def _env_int(name, default):
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


class GpuEngine(object):
    name = "gpu"

    def __init__(self, device=None):
        import torch
        self.torch = torch
        if not torch.cuda.is_available():
            raise RuntimeError("no GPU visible (torch.cuda unavailable)")
        from ..ops import load_ops
        self.ops = load_ops(required=True)
        self.device = device if device is not None else \
            torch.device("cuda", torch.cuda.current_device())
        self.chunk_bytes = _env_int("DRAGNET_CHUNK_MB", 256) * 1024 * 1024
        # chunk addressing is uint32 on-device: offsets/positions must
        # fit 4 GiB per chunk (arbitrarily large INPUTS stream through
        # many chunks; only a single RECORD larger than the chunk is
        # rejected, at scan time)
        if self.chunk_bytes > (3 << 30):
            raise ValueError(
                "DRAGNET_CHUNK_MB too large: device chunk addressing "
                "is 32-bit (max 3072 MB per chunk)")

    # ---- public engine interface ----

    def scan(self, files, queries, ds_filter=None, time_field=None,
             data_format="json", byte_source=None):
        from ..datasource.file import ScanResult

        cplan = planmod.compile_plan(
            queries, ds_filter=ds_filter, time_field=time_field,
            data_format=data_format)

        agg_slots = _env_int("DRAGNET_AGG_SLOTS", 1 << 20)
        dict_slots = _env_int("DRAGNET_DICT_SLOTS", 1 << 20)
        dict_data = _env_int("DRAGNET_DICT_DATA_MB", 256) * 1024 * 1024

        # byte sources cannot restart, so they never risk the dense
        # directory (its capacity bounds key cardinality)
        dense = False if byte_source is not None else None
        for attempt in range(4):
            ctx = _ScanContext(self, cplan, agg_slots, dict_slots,
                               dict_data, dense=dense)
            if byte_source is not None:
                if attempt > 0:
                    raise RuntimeError(
                        "table overflow on non-restartable byte source")
                for chunk in byte_source:
                    ctx.feed(chunk)
                ctx.flush()
            else:
                ctx.scan_files(files)
            if ctx.overflowed():
                from ..log import get_logger
                if ctx.dense:
                    # dense directory overflowed (high cardinality):
                    # restart on the atomic hash path at full size
                    get_logger().child("gpu-engine").warn(
                        "dense directory overflow; restarting on the "
                        "hash path", agg_slots=agg_slots)
                    dense = False
                    continue
                get_logger().child("gpu-engine").warn(
                    "aggregation table overflow; restarting scan",
                    agg_slots=agg_slots * 8, dict_slots=dict_slots * 8)
                dense = False
                agg_slots *= 8
                dict_slots *= 8
                dict_data *= 4
                continue
            aggs, stages = ctx.finalize(queries)
            return ScanResult(aggs, stages)
        raise RuntimeError("aggregation tables overflowed after retries")

    def columnar_query(self, query, filt, params, kinds, cols, vals):
        """K7: evaluate `query` over one index metric's typed columns
        on-device (columnar_query_kernel) — no NDJSON round trip
        (reference lib/index-query.js:303-338 semantics; VERDICT r1
        #6).

        query:  the effective QueryConfig (drives decode: date and
                bucket canonicalization)
        filt:   combined query ∧ time-bounds krill filter (or None)
        params: the metric's param dicts (column name order)
        kinds:  'n'/'s' per column (matches read_columns)
        cols:   per column f64 numpy array or (blob, offs, lens)
        vals:   f64 numpy array of row weights
        """
        torch = self.torch
        dev = self.device
        from ..query import QueryConfig

        # kernel plan: each breakdown reads the column keyed by its
        # FIELD name — the reference's deserializeRow reads
        # row[escape(field.field)] (lib/index-query.js:395-400), so a
        # renamed date breakdown (field != name) reads a column that
        # was never selected -> undefined -> dropped by the
        # bucketizer.  Stored columns are keyed by NAME, so mapping
        # the kernel slot by field reproduces exactly that; the
        # caller's query does the date/bucket decode.
        kq_bds = []
        for b in query.breakdowns:
            src = b.get("field", b["name"])
            nb = {"name": src, "field": src}
            if "aggr" in b:
                nb["aggr"] = b["aggr"]
                if "step" in b:
                    nb["step"] = b["step"]
            kq_bds.append(nb)
        kq = QueryConfig(filter=None, breakdowns=kq_bds,
                         allow_reserved=True)
        cplan = planmod.compile_plan([kq], ds_filter=filt)

        # slot -> column mapping; companion slots stay missing
        by_name = {p["name"]: i for i, p in enumerate(params)}
        nf = len(cplan.field_sigs)
        slot_kinds = [0] * nf
        keep = []  # device tensors referenced by the descriptor array
        nums, soffs, slens = ([torch.empty(0)] * nf for _ in range(3))
        blobs = []
        bias = 0
        biases = {}
        for ci, c in enumerate(cols):
            if kinds[ci] == "s":
                biases[ci] = bias
                blobs.append(c[0])
                bias += len(c[0])
        blob = b"".join(blobs) or b"\0"
        blob_t = torch.from_numpy(
            np.frombuffer(blob, dtype=np.uint8).copy()).to(dev)
        nrows = int(vals.shape[0])
        for si, (path, _raw) in enumerate(cplan.fields.paths):
            ci = by_name.get(path)
            if ci is None:
                continue
            if kinds[ci] == "n":
                slot_kinds[si] = 1
                t = torch.from_numpy(
                    np.ascontiguousarray(cols[ci])).to(dev)
                nums[si] = t
                keep.append(t)
            else:
                slot_kinds[si] = 2
                off = cols[ci][1].astype(np.uint32) + \
                    np.uint32(biases[ci])
                to = torch.from_numpy(off.view(np.int32)).to(dev)
                tl = torch.from_numpy(
                    cols[ci][2].view(np.int32).copy()).to(dev)
                soffs[si] = to
                slens[si] = tl
                keep.extend([to, tl])
        vals_t = torch.from_numpy(
            np.ascontiguousarray(vals)).to(dev)
        descs = self.ops.col_descs_host(slot_kinds, nums, soffs,
                                        slens).to(dev)

        agg_slots = _env_int("DRAGNET_AGG_SLOTS", 1 << 20)
        dict_slots = _env_int("DRAGNET_DICT_SLOTS", 1 << 20)
        dict_cap = max(len(blob) * 2 + (1 << 20), 8 << 20)
        for attempt in range(2):
            ctx = _ScanContext(self, cplan, agg_slots, dict_slots,
                               dict_cap, dense=False)
            self.ops.columnar_query(
                descs, blob_t, vals_t, nrows,
                ctx.field_sigs, ctx.comp_slot, cplan.nf_match,
                ctx.prog_nodes, ctx.prog_bounds,
                ctx.const_meta, ctx.const_dvals, ctx.const_bytes,
                ctx.metric_rows, ctx.synth_req,
                ctx.bd_rows, ctx.bd_steps,
                ctx.table_descs,
                ctx.sd["state"], ctx.sd["hash"], ctx.sd["id"],
                ctx.sd["off"], ctx.sd["len"], ctx.sd["data"],
                ctx.sd["used"], ctx.sd["next"],
                ctx.nd["state"], ctx.nd["bits"], ctx.nd["id"],
                ctx.nd["next"], ctx.counters)
            aggs, _stages = ctx.finalize([query])
            if not ctx.overflowed():
                return aggs[0]
            agg_slots *= 8
            dict_slots *= 8
            dict_cap *= 4
        raise RuntimeError("columnar query overflowed after retry")


PROWS = 2048  # dense partial rows == the scan kernel's grid cap


class _ScanContext(object):
    def __init__(self, eng, cplan, agg_slots, dict_slots, dict_data_cap,
                 dense=None):
        torch = eng.torch
        self.eng = eng
        self.t = torch
        self.cplan = cplan
        dev = eng.device
        # Dense-accumulation mode (default on): the aggregate table is
        # a small slot directory, per-workgroup counts go to a dense
        # [PROWS, slots] partial matrix, and the MFMA f64 column-sum
        # reduce folds it before extraction.  High
        # cardinality overflows the directory and the engine restarts
        # with dense off (the atomic hash path).
        if dense is None:
            dense = _env_int("DRAGNET_DENSE", 1) == 1
        self.dense = dense
        if dense:
            agg_slots = _env_int("DRAGNET_DENSE_SLOTS", 8192)
        i32 = dict(dtype=torch.int32, device=dev)
        f64 = dict(dtype=torch.float64, device=dev)
        u8 = dict(dtype=torch.uint8, device=dev)

        def dev_i32(np_arr):
            return torch.from_numpy(
                np.ascontiguousarray(np_arr.astype(np.int32))).to(dev)

        def dev_f64(np_arr):
            return torch.from_numpy(
                np.ascontiguousarray(np_arr.astype(np.float64))).to(dev)

        # plan buffers
        self.field_sigs = torch.from_numpy(
            cplan.field_sigs.view(np.int64)).to(dev)
        self.comp_slot = torch.from_numpy(cplan.comp_slot).to(dev)
        progs, bounds = cplan.programs
        self.prog_nodes = dev_i32(progs)
        self.prog_bounds = dev_i32(bounds)
        self.const_meta = dev_i32(cplan.const_meta)
        self.const_dvals = dev_f64(cplan.const_dvals)
        self.const_bytes = torch.from_numpy(
            np.ascontiguousarray(cplan.const_bytes)).to(dev)
        self.synth_slots = dev_i32(cplan.synthetic)
        metrics, sreq = cplan.metrics
        self.metric_rows = dev_i32(metrics)
        self.synth_req = dev_i32(sreq)
        bds, steps = cplan.breakdown_descs
        self.bd_rows = dev_i32(bds)
        self.bd_steps = dev_f64(steps)
        self.nm = metrics.shape[0]

        # tables
        self.tables = []
        self.partials = []
        for m in range(self.nm):
            state = torch.zeros(agg_slots, **i32)
            keys = torch.zeros(agg_slots * MAX_KEY, **i32)
            count = torch.zeros(agg_slots, **f64)
            self.tables.append((state, keys, count))
            if dense:
                self.partials.append(
                    torch.zeros((PROWS, agg_slots), **f64))
        descs = eng.ops.agg_descs_host(
            [t[0] for t in self.tables], [t[1] for t in self.tables],
            [t[2] for t in self.tables], self.partials)
        self.table_descs = descs.to(dev)

        # dictionaries
        self.sd = dict(
            state=torch.zeros(dict_slots, **i32),
            hash=torch.zeros(dict_slots, dtype=torch.int64, device=dev),
            id=torch.zeros(dict_slots, **i32),
            off=torch.zeros(dict_slots, **i32),
            len=torch.zeros(dict_slots, **i32),
            data=torch.zeros(dict_data_cap, **u8),
            used=torch.zeros(1, **i32),
            next=torch.zeros(1, **i32),
        )
        self.nd = dict(
            state=torch.zeros(dict_slots, **i32),
            bits=torch.zeros(dict_slots, dtype=torch.int64, device=dev),
            id=torch.zeros(dict_slots, **i32),
            next=torch.zeros(1, **i32),
        )
        self.counters = torch.zeros(
            C_GLOBAL_N + self.nm * CM_N, dtype=torch.int64, device=dev)

        self._partial = b""
        self._pinned = None
        self._copy_ev = None  # guards pinned-buffer reuse across chunks
        self.agg_slots = agg_slots
        self.dict_slots = dict_slots
        self._fin_stream = None  # lazy: D2H lane for pipelined decode

    # ---- chunk feeding ----

    def feed(self, chunk):
        """Consume a byte chunk (records split on newlines; partial
        trailing line carried to the next chunk)."""
        data = self._partial + chunk
        cut = data.rfind(b"\n")
        if cut < 0:
            self._partial = data
            return
        self._partial = data[cut + 1:]
        self._run(data[:cut + 1])

    def flush(self):
        if self._partial:
            self._run(self._partial + b"\n")
            self._partial = b""

    def _ensure_buffers(self, padded):
        torch = self.t
        dev = self.eng.device
        if self._pinned is not None and self._pinned.numel() >= padded:
            return
        cap = max(padded, self.eng.chunk_bytes + (1 << 20))
        cap = (cap + 15) & ~15
        self._pinned = torch.empty(cap, dtype=torch.uint8,
                                   pin_memory=True)
        self._dev_data = torch.empty(cap, dtype=torch.uint8, device=dev)
        # worst case: every byte is a newline
        self._pos = torch.empty(cap + 2, dtype=torch.int32, device=dev)
        self._nlines = torch.zeros(1, dtype=torch.int32, device=dev)
        self._segs = torch.empty(cap // 2048 + 2, dtype=torch.int32,
                                 device=dev)

    def _run(self, buf):
        torch = self.t
        n = len(buf)
        if n == 0:
            return
        padded = _pad(n)
        self._ensure_buffers(padded)
        pin = self._pinned
        # the previous chunk's async H2D copy must complete before the
        # pinned buffer is overwritten
        if self._copy_ev is not None:
            self._copy_ev.synchronize()
        pin[:n] = torch.frombuffer(bytearray(buf), dtype=torch.uint8)
        pin[n:padded] = 10  # newline padding for the 16B cursor window

        dev_data = self._dev_data
        dev_data[:padded].copy_(pin[:padded], non_blocking=True)
        if self._copy_ev is None:
            self._copy_ev = torch.cuda.Event()
        self._copy_ev.record()
        self.eng.ops.newline_index(dev_data, 0, n, self._segs,
                                   self._pos, self._nlines)
        self._scan_call(dev_data, 0)

    def overflowed(self):
        return int(self.counters[5].item()) > 0

    # ---- zero-copy file scanning ----

    def scan_files(self, files):
        """Scan the concatenated bytes of `files` reading directly
        into ping-pong pinned buffers (no intermediate Python bytes):
        each chunk is cut at its last newline and the tail is carried
        into the head of the other buffer while the GPU works on the
        previous chunk.  Regular-file chunks are filled with parallel
        preadv calls (page-cache/tmpfs reads are single-core bound
        at ~10 GB/s; microbench 52/67/55 GB/s at 8/16/32 readers;
        end-to-end scans measured most stable at 24 — the default.
        DRAGNET_CHUNK_MB=1024 adds ~10% on >=50 GB scans at the cost
        of slower pinned-buffer startup)."""
        import concurrent.futures as cf
        import stat as _stat

        torch = self.t
        cap = self.eng.chunk_bytes
        # adaptive chunk size: big corpora amortize per-chunk overhead
        # with 1 GiB chunks (+~10% measured at 24 readers); small scans
        # keep the cheap-startup default
        if (self._pinned is None
                and os.environ.get("DRAGNET_CHUNK_MB") is None):
            try:
                total = sum(os.stat(p).st_size for p in files)
            except OSError:
                total = 0
            if total > (16 << 30):
                cap = self.eng.chunk_bytes = 1 << 30
        if self._pinned is None:
            self._ensure_buffers(_pad(cap))
        pins = [self._pinned,
                torch.empty(self._pinned.numel(), dtype=torch.uint8,
                            pin_memory=True)]
        views = [memoryview(p.numpy()) for p in pins]
        evs = [None, None]
        pool = cf.ThreadPoolExecutor(
            max_workers=_env_int("DRAGNET_READERS", 24))

        def pread_full(fd, mv, off):
            """preadv until mv is full (a single preadv may legally
            return short; a short section would leave a HOLE of stale
            bytes that corrupts record framing — observed at the
            multi-GB scale)."""
            done = 0
            while done < len(mv):
                got = os.preadv(fd, [mv[done:]], off + done)
                if got <= 0:
                    break
                done += got
            return done

        def fill_from(fd, fpos, fsize, view, at, want, seq_file):
            """Read up to `want` bytes of fd@fpos into view[at:].
            Returns bytes read (contiguous).  Parallel preadv for
            regular files."""
            want = min(want, fsize - fpos)
            if want <= 0:
                return 0
            if seq_file is not None:  # char device / pipe: sequential
                return seq_file.readinto(view[at:at + want]) or 0
            if want < (8 << 20):
                return pread_full(fd, view[at:at + want], fpos)
            nsec = _env_int("DRAGNET_READERS", 24)
            sec = (want + nsec - 1) // nsec
            futs = []
            for s in range(0, want, sec):
                e = min(s + sec, want)
                futs.append((e - s, pool.submit(
                    pread_full, fd, view[at + s:at + e], fpos + s)))
            # only the contiguous prefix counts
            total = 0
            for length, fut in futs:
                got = fut.result()
                total += got
                if got < length:
                    break
            return total

        def last_newline(buf, n):
            probe = max(0, n - (1 << 16))
            cut = bytes(buf[probe:n]).rfind(b"\n")
            if cut >= 0:
                return probe + cut
            return bytes(buf[:probe]).rfind(b"\n")

        cur = 0
        head = 0
        fiter = iter(files)
        fobj = None     # (fd, pos, size, seq_file or None)
        try:
            while True:
                if evs[cur] is not None:
                    evs[cur].synchronize()
                    evs[cur] = None
                n = head
                eof = False
                while n < cap:
                    if fobj is None:
                        try:
                            path = next(fiter)
                        except StopIteration:
                            eof = True
                            break
                        st = os.stat(path)
                        if _stat.S_ISREG(st.st_mode):
                            fd = os.open(path, os.O_RDONLY)
                            fobj = (fd, 0, st.st_size, None)
                        else:
                            sf = open(path, "rb", buffering=0)
                            fobj = (sf.fileno(), 0, 1 << 62, sf)
                    fd, fpos, fsize, sf = fobj
                    got = fill_from(fd, fpos, fsize, views[cur], n,
                                    cap - n, sf)
                    if got <= 0:
                        if sf is not None:
                            sf.close()
                        else:
                            os.close(fd)
                        fobj = None
                        continue
                    fobj = (fd, fpos + got, fsize, sf)
                    n += got
                if n == 0:
                    break
                cut = last_newline(views[cur], n)
                if eof and cut < n - 1:
                    views[cur][n:n + 1] = b"\n"
                    n += 1
                    cut = n - 1
                nxt = 1 - cur
                if cut < 0:
                    if n >= cap:
                        raise RuntimeError(
                            "record exceeds the chunk buffer (%d "
                            "bytes); raise DRAGNET_CHUNK_MB" % cap)
                    tail = n
                else:
                    tail = n - (cut + 1)
                # the tail must be carried over BEFORE the chunk is
                # launched: _run_pinned pads pin[cut+1:] with newlines
                # and would clobber the carried bytes (record-splitting
                # corruption observed at multi-GB scale)
                if tail:
                    if evs[nxt] is not None:
                        evs[nxt].synchronize()
                        evs[nxt] = None
                    views[nxt][:tail] = views[cur][n - tail:n]
                if cut >= 0:
                    evs[cur] = self._run_pinned(pins[cur], cut + 1)
                head = tail
                cur = nxt
                if eof:
                    break
        finally:
            if fobj is not None:
                _fd, _p, _s, sf = fobj
                if sf is not None:
                    sf.close()
                else:
                    os.close(_fd)
            pool.shutdown(wait=False)

    def _run_pinned(self, pin, n):
        """H2D + kernels for pin[:n] (already newline-terminated).

        Ping-pong DEVICE buffers with the H2D on a dedicated copy
        stream: chunk k+1's copy overlaps chunk k's kernels, so the
        steady state is max(fill, copy, kernels) instead of their sum
        (the serial form measured 12-15 GB/s end-to-end vs the 67 GB/s
        the parallel readers deliver; see profiles/r02 notes).
        Returns the H2D-complete event — the caller must wait on it
        before refilling this pinned buffer."""
        torch = self.t
        padded = _pad(n)
        pin[n:padded] = 10
        if not hasattr(self, "_fs_dev"):
            self._fs_copy_stream = torch.cuda.Stream(
                device=self.eng.device)
            self._fs_dev = [self._dev_data,
                            torch.empty_like(self._dev_data)]
            self._fs_done = [torch.cuda.Event(), torch.cuda.Event()]
            self._fs_idx = 0
        idx = self._fs_idx
        self._fs_idx ^= 1
        dev_data = self._fs_dev[idx]
        main = torch.cuda.current_stream(self.eng.device)
        cs = self._fs_copy_stream
        # last kernel reader of this device buffer must finish first
        cs.wait_event(self._fs_done[idx])
        ev = torch.cuda.Event()
        with torch.cuda.stream(cs):
            dev_data[:padded].copy_(pin[:padded], non_blocking=True)
            ev.record(cs)
        main.wait_event(ev)
        self.eng.ops.newline_index(dev_data, 0, n, self._segs,
                                   self._pos, self._nlines)
        self._scan_call(dev_data, 0)
        self._fs_done[idx].record(main)
        return ev

    # ---- resident-pool path (bench / repeated scans) ----

    def stage_resident(self, buf, n_slices=None):
        """Stage a byte pool into the pinned buffer once; later
        scan_resident() calls re-run the H2D + kernels without the
        host-side copy.  The pool is split into newline-aligned
        slices so the H2D copy of slice k+1 overlaps the scan of
        slice k (copy stream + events)."""
        torch = self.t
        if n_slices is None:
            n_slices = _env_int("DRAGNET_SLICES", 6)
        n = len(buf)
        padded = _pad(n)
        self._ensure_buffers(padded)
        pin = self._pinned
        pin[:n] = torch.frombuffer(bytearray(buf), dtype=torch.uint8)
        pin[n:padded] = 10

        # newline-aligned slice boundaries
        bounds = [0]
        for k in range(1, n_slices):
            target = n * k // n_slices
            cut = buf.rfind(b"\n", 0, target)
            start = cut + 1 if cut >= 0 else 0
            if start > bounds[-1]:
                bounds.append(start)
        bounds.append(n)
        self._slices = [(bounds[i], bounds[i + 1])
                        for i in range(len(bounds) - 1)
                        if bounds[i + 1] > bounds[i]]
        self._resident = (n, padded)
        self._copy_stream = torch.cuda.Stream(device=self.eng.device)
        self._slice_evs = [torch.cuda.Event()
                           for _ in self._slices]
        # Ping-pong device pools for streaming passes: pass k+1's H2D
        # lands in the buffer pass k is NOT reading, so the copy
        # stream runs back-to-back at link rate instead of stalling
        # behind pass k's kernel tail + extraction + reset.
        self._dev_data_pp = torch.empty_like(self._dev_data)
        self._pass_done = [torch.cuda.Event(), torch.cuda.Event()]
        self._db_idx = 0

        # Pre-index line positions per slice: the pool bytes are
        # identical on every pass, so the line index is computed once
        # here (a re-scanning engine retains its line index the same
        # way the reference retains its on-disk indexes).
        self._dev_data[:padded].copy_(pin[:padded])
        self._slice_pos = []
        for (s, e) in self._slices:
            self.eng.ops.newline_index(self._dev_data, s, e,
                                       self._segs, self._pos,
                                       self._nlines)
            cnt = int(self._nlines.item())
            self._slice_pos.append(
                (self._pos[:cnt].clone(), self._nlines.clone()))

    def scan_resident(self, h2d=True):
        """One full streaming pass over the staged pool: sliced async
        H2D on a copy stream overlapping newline-index + fused scan on
        the compute stream.  h2d=False skips the copies (device-
        resident re-scan: data already in HBM from a previous pass)."""
        torch = self.t
        n, padded = self._resident
        dev_data = self._dev_data
        pin = self._pinned
        main = torch.cuda.current_stream(self.eng.device)
        if h2d:
            # ping-pong: write the buffer the PREVIOUS pass did not
            # read; wait only for the pass-before-last (the last
            # reader of this buffer), which has long finished — so
            # the copies chain back-to-back across passes
            idx = self._db_idx
            self._db_idx ^= 1
            dev_data = self._dev_data if idx == 0 else self._dev_data_pp
            ev_done = self._pass_done[idx]
            self._copy_stream.wait_event(ev_done)
            with torch.cuda.stream(self._copy_stream):
                for k, (s, e) in enumerate(self._slices):
                    s16 = s & ~15
                    e16 = min((e + 15) & ~15, padded)
                    dev_data[s16:e16].copy_(pin[s16:e16],
                                            non_blocking=True)
                    self._slice_evs[k].record(self._copy_stream)
        # kernels per slice on the compute stream.  Streaming passes
        # (h2d=True) model fresh data and re-index newlines each pass;
        # device-resident re-scans reuse the staged line index (a
        # re-scanning engine retains it like the reference retains its
        # on-disk indexes).
        for k, (s, e) in enumerate(self._slices):
            if h2d:
                main.wait_event(self._slice_evs[k])
                self.eng.ops.newline_index(dev_data, s, e, self._segs,
                                           self._pos, self._nlines)
                self._scan_call(dev_data, s)
            else:
                pos_k, nlines_k = self._slice_pos[k]
                self._scan_call(dev_data, s, pos_k, nlines_k)
        if h2d:
            ev_done.record(main)  # last reader of this pass's buffer

    def _scan_call(self, dev_data, first_start, pos=None, nlines=None,
                   x=None):
        self.eng.ops.scan_chunk(
            dev_data,
            pos if pos is not None else self._pos,
            nlines if nlines is not None else self._nlines,
            first_start,
            self.field_sigs, self.comp_slot, self.cplan.nf_match,
            self.cplan.sig_bloom, self.cplan.fields_parent_sig,
            self.prog_nodes, self.prog_bounds,
            self.const_meta, self.const_dvals, self.const_bytes,
            self.synth_slots, self.cplan.n_synth,
            self.metric_rows, self.synth_req,
            self.bd_rows, self.bd_steps,
            self.cplan.value_slot, self.cplan.fields_slot,
            self.cplan.data_format == "json-skinner",
            _env_int("DRAGNET_LDS_STAGE", 0),
            self.table_descs,
            self.sd["state"], self.sd["hash"], self.sd["id"],
            self.sd["off"], self.sd["len"], self.sd["data"],
            self.sd["used"], self.sd["next"],
            self.nd["state"], self.nd["bits"], self.nd["id"],
            self.nd["next"],
            self.counters,
            x["xdata"] if x else dev_data,
            x["wave_base"] if x else self.counters,
            x["rec_len"] if x else self.counters,
            x["n_slots"] if x else 0)

    # ---- wave-transposed staging (prototype; DRAGNET_XPOSE) ----

    def stage_xpose(self, buf=None):
        """Wave-transposed staging for scan_kernel_x: records
        length-sorted (aggregation is order-independent), 64
        consecutive sorted records form a wave, each record's bytes
        split into granules interleaved so granule g of lane l sits at
        wave_base + g*64*gran + l*gran — a wave's window refills touch
        64 CONSECUTIVE granules (coalesced) instead of 64 scattered
        records.

        Default: built ON DEVICE from the resident pool (torch sort +
        cumsum + the xpose_build_kernel scatter) — no host staging
        pass.  DRAGNET_XPOSE_HOST=1 selects the r1 numpy builder
        (A/B + layout unit tests)."""
        if os.environ.get("DRAGNET_XPOSE_HOST") == "1":
            return self._stage_xpose_host(buf)
        torch = self.t
        dev = self.eng.device
        gran = _env_int("DRAGNET_XGRAN", 64)
        if gran not in (32, 64, 128):
            # the scan kernel is templated on the granule size; an
            # unsupported value would mismatch the staged layout
            raise ValueError("DRAGNET_XGRAN must be 32, 64 or 128")
        glog = gran.bit_length() - 1
        if not hasattr(self, "_resident"):
            if buf is None:
                raise ValueError("no resident pool to transpose")
            self.stage_resident(buf)
        n, _padded = self._resident
        # full-pool line index (the staged slice indexes cover slices)
        self.eng.ops.newline_index(self._dev_data, 0, n, self._segs,
                                   self._pos, self._nlines)
        nrec = int(self._nlines.item())  # one-time staging sync
        if nrec == 0:
            raise ValueError("no records")
        pos = self._pos[:nrec].to(torch.int64)
        starts = torch.empty_like(pos)
        starts[0] = 0
        starts[1:] = pos[:-1] + 1
        lens_sorted, order = torch.sort(pos - starts)
        starts_sorted = starts[order]
        nslots = (nrec + 63) & ~63
        nw = nslots // 64
        slot_len = torch.full((nslots,), -1, dtype=torch.int32,
                              device=dev)
        slot_len[:nrec] = lens_sorted.to(torch.int32)
        sstart = torch.zeros(nslots, dtype=torch.int32, device=dev)
        sstart[:nrec] = starts_sorted.to(torch.int32)
        # per-wave granule count from each wave's longest (last) lane
        last = torch.clamp(
            torch.arange(nw, device=dev, dtype=torch.int64) * 64 + 63,
            max=nrec - 1)
        gwl = torch.clamp((lens_sorted[last] + gran - 1) // gran, min=1)
        stride = 64 * gran
        wbase = torch.zeros(nw + 1, dtype=torch.int64, device=dev)
        torch.cumsum(gwl * stride, 0, out=wbase[1:])
        total = int(wbase[-1].item()) + stride  # +slack
        xb = torch.empty(total, dtype=torch.uint8, device=dev)
        self.eng.ops.xpose_build(self._dev_data, sstart, slot_len,
                                 wbase, nslots, glog, xb)
        self._x = {
            "xdata": xb,
            "wave_base": wbase,  # kernel reads [0, nw); +1 harmless
            "rec_len": slot_len,
            "n_slots": nslots,
        }
        self._x_nrec = nrec

    def _stage_xpose_host(self, buf):
        """r1 host-side numpy builder (kept for A/B and layout tests)."""
        torch = self.t
        dev = self.eng.device
        gran = _env_int("DRAGNET_XGRAN", 64)
        if gran not in (32, 64, 128):
            raise ValueError("DRAGNET_XGRAN must be 32, 64 or 128")
        xb, wave_base, slot_len, nslots, n = _build_xpose_layout(
            buf, gran)
        self._x = {
            "xdata": torch.from_numpy(xb).to(dev),
            "wave_base": torch.from_numpy(wave_base).to(dev),
            "rec_len": torch.from_numpy(
                slot_len.view(np.int32)).to(dev),
            "n_slots": nslots,
        }
        self._x_nrec = n

    def scan_xpose(self):
        """One scan pass over the wave-transposed staging."""
        rl = self._x["rec_len"]
        self._scan_call(self._x["xdata"], 0, pos=rl, nlines=rl,
                        x=self._x)

    def make_graph(self, h2d=True, xpose=False):
        """Capture reset + the whole scan pass into a hipGraph so a
        step replays as one launch (the per-slice kernel+copy launch
        overhead otherwise costs ~0.5-1 ms per step).  Returns a
        replayable graph or None if capture fails."""
        torch = self.t

        def pass_():
            self.reset()
            if xpose:
                self.scan_xpose()
            else:
                self.scan_resident(h2d=h2d)

        try:
            # warm up the exact op sequence outside capture
            pass_()
            torch.cuda.synchronize(self.eng.device)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                pass_()
            torch.cuda.synchronize(self.eng.device)
            return g
        except Exception:
            return None

    def reset(self):
        """Zero tables/dictionaries/counters for a fresh scan job
        (one binding call; ~10 separate .zero_() dispatches otherwise
        show up at 5 ms/step)."""
        self.eng.ops.scan_reset(
            [t[0] for t in self.tables], [t[2] for t in self.tables],
            self.sd["state"], self.sd["used"], self.sd["next"],
            self.nd["state"], self.nd["next"], self.counters,
            self.partials)

    # ---- results ----

    def finalize(self, queries):
        return self.decode_extracted(self.extract_async(queries),
                                     queries)

    def extract_async(self, queries):
        """Enqueue the table/dictionary extraction kernels and snapshot
        everything decode needs into fresh device tensors, WITHOUT any
        host sync.  The returned handle is safe across a subsequent
        reset()/scan of this context (the snapshots are ordered on the
        current stream before the reset's zeroing), which lets a caller
        software-pipeline host-side decode of step k with step k+1's
        copies/kernels (bench.py streaming mode)."""
        r = self.eng.ops.extract_all(
            [t[0] for t in self.tables], [t[1] for t in self.tables],
            [t[2] for t in self.tables], self.agg_slots,
            self.sd["state"], self.sd["hash"], self.sd["id"],
            self.sd["off"], self.sd["len"], self.sd["data"],
            self.sd["used"], self.sd["next"],
            self.nd["state"], self.nd["bits"], self.nd["id"],
            self.nd["next"], self.counters, self.dict_slots,
            self.partials)
        h = {"cnt": r[0], "n_str": r[1], "n_num": r[2], "used": r[3],
             "blob": r[4], "str_off": r[5], "str_len": r[6],
             "numbers": r[7],
             "aggs": [tuple(r[8 + 3 * m:11 + 3 * m])
                      for m in range(len(queries))]}
        torch = self.t
        h["ev"] = torch.cuda.Event()
        h["ev"].record()
        return h

    def decode_extracted(self, h, queries):
        """Host-side half of finalize(): wait for the extraction
        snapshot (ONLY — via its event, on a dedicated D2H stream, so
        in-flight work from a subsequent step is not drained) and
        decode into Aggregators."""
        torch = self.t
        if self._fin_stream is None:
            self._fin_stream = torch.cuda.Stream(device=self.eng.device)
        with torch.cuda.stream(self._fin_stream):
            self._fin_stream.wait_event(h["ev"])
            cnt = h["cnt"].cpu().numpy()

            # dictionaries
            n_str = int(h["n_str"].item())
            n_num = int(h["n_num"].item())
            strings = []
            if n_str:
                off = h["str_off"][:n_str].cpu().numpy().astype(np.uint32)
                ln = h["str_len"][:n_str].cpu().numpy().astype(np.uint32)
                used = int(h["used"].item())
                blob = h["blob"][:used].cpu().numpy().tobytes()
                for i in range(n_str):
                    raw = blob[off[i]:off[i] + ln[i]]
                    strings.append(_decode_json_string(raw))
            numbers = np.zeros(0)
            if n_num:
                numbers = h["numbers"][:n_num].cpu().numpy()
            hostk = []
            for m, _q in enumerate(queries):
                k_full, c_full, out_n = h["aggs"][m]
                n_out = min(int(out_n.item()), self.agg_slots)
                hostk.append((k_full[:n_out].cpu().numpy(),
                              c_full[:n_out].cpu().numpy()))

        aggs = []
        for m, q in enumerate(queries):
            k, c = hostk[m]
            k = k.astype(np.uint32)
            agg = Aggregator(q)
            mc = cnt[C_GLOBAL_N + m * CM_N:
                     C_GLOBAL_N + (m + 1) * CM_N]
            agg.ninputs = int(mc[CM_AGG_IN])
            agg.ndropped_nonnumeric = int(mc[CM_NONNUMERIC])
            nk = len(q.breakdowns)
            for i in range(k.shape[0]):
                key = planmod.decode_key(k[i, :nk], q, strings, numbers)
                v = c[i]
                v = int(v) if float(v).is_integer() else float(v)
                agg.table[key] = agg.table.get(key, 0) + v
            aggs.append(agg)

        stages = self._counter_stages(cnt, queries)
        for name, c in stages:
            if name == "Aggregator":
                c["noutputs"] = aggs[0].noutputs()
        return aggs, stages

    def _counter_stages(self, cnt, queries):
        """Reconstruct the reference pipeline counter stages from the
        device counters (mirrors scan_cpu.ScanPipeline.counter_stages)."""
        cp = self.cplan
        stages = [("json parser", {
            "ninputs": int(cnt[0]), "noutputs": int(cnt[2]),
            "invalid json": int(cnt[1])})]
        if cp.data_format == "json":
            stages.append(("SkinnerAdapterStream",
                           {"ninputs": int(cnt[2]),
                            "noutputs": int(cnt[2])}))
        # program 0 is the ds filter; OP_TRUE (8) means "no filter"
        progs, bounds = cp.programs
        if progs[bounds[0][0]][0] != 8:
            n_in = int(cnt[2])
            stages.append(("Datasource filter", {
                "ninputs": n_in,
                "noutputs": n_in - int(cnt[3]) - int(cnt[4]),
                "nfilteredout": int(cnt[3]),
                "nfailedeval": int(cnt[4])}))
        # metric 0's pipeline (what the CLI displays for scans)
        q = queries[0]
        mc = cnt[C_GLOBAL_N:C_GLOBAL_N + CM_N]
        n = int(mc[CM_FILTER_IN])
        if q.filter is not None:
            out = n - int(mc[CM_FILTERED]) - int(mc[CM_FAILEDEVAL])
            stages.append(("User filter", {
                "ninputs": n, "noutputs": out,
                "nfilteredout": int(mc[CM_FILTERED]),
                "nfailedeval": int(mc[CM_FAILEDEVAL])}))
            n = out
        metrics, _sreq = cp.metrics
        if metrics[0][3] > 0:  # has synthetic requirements
            out = n - int(mc[CM_UNDEF]) - int(mc[CM_BADDATE])
            stages.append(("Datetime parser", {
                "ninputs": n, "noutputs": out,
                "undef": int(mc[CM_UNDEF]),
                "baddate": int(mc[CM_BADDATE])}))
            n = out
        if metrics[0][5]:  # time filter
            out = n - int(mc[CM_TIME_OUT])
            stages.append(("Time filter", {
                "ninputs": n, "noutputs": out,
                "nfilteredout": int(mc[CM_TIME_OUT]),
                "nfailedeval": 0}))
            n = out
        self._agg_stage_n = n
        stages.append(("Aggregator", {
            "ninputs": int(mc[CM_AGG_IN]),
            "noutputs": 0,  # patched by caller if needed
            "nonnumeric": int(mc[CM_NONNUMERIC])}))
        return stages


def _build_xpose_layout(buf, gran=64):
    """Numpy construction of the wave-transposed layout (see
    _ScanContext.stage_xpose): returns (xbuf, wave_base, slot_len,
    n_slots, n_records).  Byte p of slot r lives at
    wave_base[r//64] + (p//gran)*(64*gran) + (r%64)*gran + p%gran."""
    stride = 64 * gran
    arr = np.frombuffer(buf, dtype=np.uint8)
    nl = np.flatnonzero(arr == 10).astype(np.int64)
    if nl.size == 0:
        raise ValueError("no records")
    starts = np.empty_like(nl)
    starts[0] = 0
    starts[1:] = nl[:-1] + 1
    lens = (nl - starts).astype(np.int64)
    n = int(lens.size)
    order = np.argsort(lens, kind="stable")
    nslots = (n + 63) & ~63
    nw = nslots // 64
    so = starts[order]
    lo = lens[order]
    # per-wave granule count: lengths are sorted, so a wave's max is
    # its last real lane; the granule-row stride is a constant 4096
    # (64 lanes x 64B), so variable wave sizes need no device change
    last = np.minimum(np.arange(nw) * 64 + 63, n - 1)
    gw = np.maximum(1, (lo[last] + gran - 1) // gran).astype(np.int64)
    wave_base = np.zeros(nw, dtype=np.int64)
    np.cumsum(gw[:-1] * stride, out=wave_base[1:])
    total = int(wave_base[-1] + gw[-1] * stride) + stride  # +slack
    xb = np.full(total, 10, dtype=np.uint8)
    # build waves in batches of equal granule count (few distinct
    # values after sorting) with one vectorized gather per batch
    for g in np.unique(gw):
        ws = np.flatnonzero(gw == g)
        K = int(g) * gran
        rsel = (ws[:, None] * 64 + np.arange(64)[None, :]).reshape(-1)
        rl = np.where(rsel < n, lo[np.minimum(rsel, n - 1)], 0)
        rs = np.where(rsel < n, so[np.minimum(rsel, n - 1)], 0)
        idx = rs[:, None] + np.arange(K)[None, :]
        np.minimum(idx, arr.size - 1, out=idx)
        m = arr[idx]
        m[np.arange(K)[None, :] >= rl[:, None]] = 10
        # (wave, lane, granule, 64) -> (wave, granule, lane, 64)
        blk = np.ascontiguousarray(
            m.reshape(len(ws), 64, int(g), gran).transpose(0, 2, 1, 3)
        ).reshape(len(ws), -1)
        for i, w in enumerate(ws):
            b = int(wave_base[w])
            xb[b:b + K * 64] = blk[i]
    slot_len = np.full(nslots, 0xFFFFFFFF, dtype=np.uint32)
    slot_len[:n] = lo.astype(np.uint32)
    return xb, wave_base, slot_len, nslots, n


def _decode_json_string(raw):
    """Decode raw in-record string bytes (may contain JSON escapes)
    to the same Python string json.loads would produce."""
    try:
        s = raw.decode("utf-8")
    except UnicodeDecodeError:
        return raw.decode("utf-8", "replace")
    if "\\" in s:
        try:
            return json.loads('"' + s + '"')
        except ValueError:
            return s
    return s


def _read_files(files, chunk_size, prefetch=4):
    """Chunked concatenated reader with a background prefetch thread
    (the reference reads files with concurrency 2 through catstreams,
    lib/datasource-file.js:252-288; here a reader thread keeps the
    GPU fed while it scans the previous chunk)."""
    import queue
    import threading

    q = queue.Queue(maxsize=prefetch)
    err = []

    def reader():
        try:
            for path in files:
                with open(path, "rb", buffering=0) as f:
                    while True:
                        chunk = f.read(chunk_size)
                        if not chunk:
                            break
                        q.put(chunk)
        except Exception as e:  # surfaced on the consumer side
            err.append(e)
        finally:
            q.put(None)

    t = threading.Thread(target=reader, daemon=True)
    t.start()
    while True:
        chunk = q.get()
        if chunk is None:
            break
        yield chunk
    t.join()
    if err:
        raise err[0]
