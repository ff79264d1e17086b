"""
Query-set -> device plan compiler.

Compiles (ds_filter, [QueryConfig...], time_field, data_format) into the
packed little buffers the fused HIP scan kernel consumes:

  * field table: one slot per distinct dotted path the scan needs
    (filter fields + breakdown sources + synthetic date sources), as
    64-bit FNV-1a signatures of the dotted path.  The device parser
    computes the same incremental signature while walking each record's
    nesting (dragnet_amd/ops/hip/scan_kernels.hip, K1) — dotted literal
    keys and nested paths collapse to the same signature by
    construction, mirroring the CPU lookup (points.lookup).
  * predicate programs: prefix-tree bytecode with subtree skip offsets
    so the device VM short-circuits exactly like the CPU evaluator.
  * synthetic descriptors: (source slot) pairs to date-parse (K3).
  * per-metric descriptors: program id, synthetic requirements, time
    bounds, breakdown columns (slot/synthetic, bucketizer).

Layouts are mirrored by struct definitions in ops/hip/common.h.
"""


import os

import numpy as np

from .. import krill

FNV_OFFSET = 0xCBF29CE484222325
FNV_PRIME = 0x100000001B3
MASK64 = (1 << 64) - 1

# op codes (common.h)
OP_AND = 0
OP_OR = 1
OP_EQ = 2
OP_NE = 3
OP_LT = 4
OP_LE = 5
OP_GT = 6
OP_GE = 7
OP_TRUE = 8

CONST_NUM = 0   # numbers and booleans (dval)
CONST_STR = 1   # bytes (+ dval when the string coerces to a number)
CONST_NULL = 2

BUCKET_NONE = 0
BUCKET_P2 = 1
BUCKET_LIN = 2

MAX_FIELDS = 24
MAX_DEPTH = 12
MAX_KEY = 8  # device key tuple width (common.h / gpu.py)


def fnv1a(data, h=FNV_OFFSET):
    for b in data:
        h = ((h ^ b) * FNV_PRIME) & MASK64
    return h


def _mix64(x):
    x &= MASK64
    x ^= x >> 33
    x = (x * 0xFF51AFD7ED558CCD) & MASK64
    x ^= x >> 33
    x = (x * 0xC4CEB9FE1A85EC53) & MASK64
    x ^= x >> 33
    return x


SIG_LENK = 0xFF51AFD7ED558CCD
SIG_LIT_MARK = 0xC2B2AE3D27D4EB4F  # fields-root literal-dotted marker


def _sig_mix(x):
    """5-op bijective mix used ONLY for path signatures (mirrors
    scan_kernels.hip sig_mix)."""
    x &= MASK64
    x ^= x >> 32
    x = (x * 0xD6E8FEB86659FD93) & MASK64
    return x ^ (x >> 32)


def comp_into(sig, b):
    """Fold one path component's bytes into the chained signature:
    zero-padded 8-byte little-endian words, one mix per word, plus a
    length-finalization mix — computable on-device with inline SWAR
    window folds during the key scan (scan_kernels.hip scan_key_sig /
    sig_comp_finish must produce identical values)."""
    for k in range(0, len(b), 8):
        w = int.from_bytes(b[k:k + 8].ljust(8, b"\0"), "little")
        sig = _sig_mix(sig ^ w)
    return _sig_mix(sig ^ ((len(b) * SIG_LENK) & MASK64))


def lit_sig(name):
    """Signature of a whole literal key (dots as plain bytes) at the
    top level — the companion-slot match target for json mode."""
    return comp_into(FNV_OFFSET, name.encode("utf-8"))


def path_sig(path):
    """Signature of a dotted path: chained component folds.  Dots
    split components, so a literal "a.b" key and nested a->b produce
    the same signature (the aggregation lookup rule)."""
    sig = FNV_OFFSET
    for comp in path.split("."):
        sig = comp_into(sig, comp.encode("utf-8"))
    return sig


class PlanError(Exception):
    pass


class FieldTable(object):
    def __init__(self, prefix=""):
        self.prefix = prefix  # "fields." for json-skinner
        self.paths = []       # (path, raw) — raw paths skip the prefix
        self.by_path = {}

    def slot(self, path, raw=False):
        key = (path, raw)
        if key in self.by_path:
            return self.by_path[key]
        i = len(self.paths)
        if i >= MAX_FIELDS:
            raise PlanError(
                "query references more than %d fields" % MAX_FIELDS)
        self.paths.append(key)
        self.by_path[key] = i
        return i

    def sigs(self):
        return np.array(
            [path_sig(p if raw else self.prefix + p)
             for p, raw in self.paths],
            dtype=np.uint64)


class ConstPool(object):
    def __init__(self):
        self.metas = []   # (kind, str_off, str_len, dvalid, off2, len2)
        self.dvals = []
        self.bytes = bytearray()

    def add(self, v):
        if v is None:
            self.metas.append((CONST_NULL, 0, 0, 0))
            self.dvals.append(0.0)
        elif isinstance(v, bool):
            self.metas.append((CONST_NUM, 0, 0, 1))
            self.dvals.append(1.0 if v else 0.0)
        elif isinstance(v, (int, float)):
            self.metas.append((CONST_NUM, 0, 0, 1))
            self.dvals.append(float(v))
        else:
            import json as _json
            b = v.encode("utf-8")
            off = len(self.bytes)
            self.bytes.extend(b)
            num = krill.to_number(v)
            valid = 0 if num != num else 1
            # records may carry the string in ESCAPED form; store the
            # canonical JSON escaping as an alternate match target
            esc = _json.dumps(v, ensure_ascii=False)[1:-1].encode("utf-8")
            if esc != b:
                off2 = len(self.bytes)
                self.bytes.extend(esc)
                alt = (off2, len(esc))
            else:
                alt = (0, 0)  # len2 == 0: no alternate form
            self.metas.append((CONST_STR, off, len(b), valid) + alt)
            self.dvals.append(num if valid else 0.0)
            return len(self.metas) - 1
        self.metas.append(self.metas.pop() + (0, 0))
        return len(self.metas) - 1


def compile_predicate(pred, fields, consts):
    """Flatten a predicate into [op, a, b, next] i32 nodes (prefix
    order; `next` = index just past the node's subtree)."""
    nodes = []

    def emit(p):
        idx = len(nodes)
        if len(p) == 0:
            nodes.append([OP_TRUE, 0, 0, idx + 1])
            return
        (key, val), = p.items()
        if key == "and" or key == "or":
            nodes.append([OP_AND if key == "and" else OP_OR,
                          len(val), 0, 0])
            for sub in val:
                emit(sub)
            nodes[idx][3] = len(nodes)
            return
        op = {"eq": OP_EQ, "ne": OP_NE, "lt": OP_LT,
              "le": OP_LE, "gt": OP_GT, "ge": OP_GE}[key]
        slot = fields.slot(val[0])
        cidx = consts.add(val[1])
        nodes.append([op, slot, cidx, idx + 1])

    emit(pred if pred is not None else {})
    return nodes


class CompiledPlan(object):
    """All packed buffers; everything numpy, converted to torch by the
    GPU engine."""

    def __init__(self, fields, programs, const_meta, const_dvals,
                 const_bytes, synthetic, metrics, breakdown_descs,
                 queries):
        self.fields = fields
        self.field_sigs = fields.sigs()
        self.programs = programs          # i32 [n_nodes, 4]
        self.const_meta = const_meta      # i32 [n_consts, 4]
        self.const_dvals = const_dvals    # f64 [n_consts]
        self.const_bytes = const_bytes    # u8
        self.synthetic = synthetic        # i32 [n_synth]  (source slot)
        self.metrics = metrics            # i32 [n_metrics, 8] + f64 cols
        self.breakdown_descs = breakdown_descs
        self.queries = queries

    def describe(self):
        return {
            "fields": list(self.fields.paths),
            "n_metrics": int(self.metrics[0].shape[0]),
        }


def compile_plan(queries, ds_filter=None, time_field=None,
                 data_format="json"):
    """Compile the full scan plan.

    Returns a CompiledPlan.  Metric descriptor row layout (i32):
      [prog_id, n_breakdowns, bd_off, n_synth_req, synth_req_off,
       has_time_filter, t_ge, t_lt]
    Breakdown descriptor row (i32 + f64 side array):
      [kind(0=field,1=synthetic), slot_or_synth, bucket_kind, pad]
      f64: lin step
    """
    prefix = "fields." if data_format == "json-skinner" else ""
    fields = FieldTable(prefix)
    consts = ConstPool()
    prog_nodes = []
    prog_bounds = []  # (start, end) per program

    def add_program(pred):
        start = len(prog_nodes)
        nodes = compile_predicate(pred, fields, consts)
        for n in nodes:
            n[3] += start  # next/end indices are global
        prog_nodes.extend(nodes)
        prog_bounds.append((start, len(prog_nodes)))
        return len(prog_bounds) - 1

    ds_prog = add_program(ds_filter)  # program 0: datasource filter

    synth_slots = []   # source slot per synthetic entry
    synth_by_slot = {}

    def synth_index(src_path):
        slot = fields.slot(src_path)
        if slot in synth_by_slot:
            return synth_by_slot[slot]
        synth_by_slot[slot] = len(synth_slots)
        synth_slots.append(slot)
        return synth_by_slot[slot]

    metric_rows = []
    bd_rows = []
    bd_steps = []
    synth_req = []

    for q in queries:
        if len(q.breakdowns) > MAX_KEY:
            # the device key tuple is uint32_t key[MAX_KEY]; a wider
            # query would write out of bounds — the CPU
            # engine serves such queries instead
            raise PlanError(
                "query has more than %d breakdowns (GPU key width); "
                "use the CPU engine" % MAX_KEY)
        prog_id = add_program(q.filter)

        # synthetic fields this metric needs, in the reference's
        # evaluation order (breakdown dates first, then dn_ts;
        # lib/stream-scan.js:62-72)
        my_synth = []
        for b in q.breakdowns:
            if "date" in b:
                my_synth.append(synth_index(b["field"]))
        has_tf = 0
        t_ge = t_lt = 0
        dn_ts_idx = -1
        if q.before_ms is not None:
            if not time_field:
                raise PlanError(
                    'datasource is missing "timefield" for "before" '
                    'and "after" constraints')
            dn_ts_idx = synth_index(time_field)
            my_synth.append(dn_ts_idx)
            has_tf = 1
            t_ge = -(-q.after_ms // 1000)   # ceil
            t_lt = -(-q.before_ms // 1000)

        sreq_off = len(synth_req)
        # dedupe, keep order
        seen = set()
        my_synth_u = [s for s in my_synth
                      if not (s in seen or seen.add(s))]
        synth_req.extend(my_synth_u)

        # names materialized by this query's synthetic stage: any
        # column with that NAME sees the converted unix seconds (the
        # CPU pipeline mutates fields[name]; stream-synthetic.js:80)
        synth_names = {b["name"]: b["field"]
                       for b in q.breakdowns if "date" in b}
        if q.time_field:
            synth_names.setdefault(q.time_field, q.time_field)

        bd_off = len(bd_rows)
        for b in q.breakdowns:
            if "date" in b:
                kind = 1
                ref = synth_index(b["field"])
            elif b["name"] in synth_names:
                kind = 1
                ref = synth_index(synth_names[b["name"]])
            else:
                kind = 0
                ref = fields.slot(b["name"])
            bk = q.bucketizers.get(b["name"])
            if bk is None:
                bkind, step = BUCKET_NONE, 0.0
            elif bk.aggr == "quantize":
                bkind, step = BUCKET_P2, 0.0
            else:
                bkind, step = BUCKET_LIN, float(bk.step)
            bd_rows.append([kind, ref, bkind, 0])
            bd_steps.append(step)

        metric_rows.append([
            prog_id, len(q.breakdowns), bd_off,
            len(my_synth_u), sreq_off, has_tf,
            int(t_ge), int(t_lt),
        ])

    # json-skinner: top-level "value" (weight) and "fields" (presence
    # check), both outside the "fields." prefix
    value_slot = -1
    fields_slot = -1
    if data_format == "json-skinner":
        value_slot = fields.slot("value", raw=True)
        fields_slot = fields.slot("fields", raw=True)

    # Companion slots: a TOP-LEVEL literal dotted key ("a.b": v) is
    # addressable by the aggregation lookup (points.lookup checks the
    # whole literal name before plucking) but invisible to krill pluck
    # and synthetic sources.  In json mode the device hashes each key
    # as ONE component (dots are plain bytes), so the companion's sig
    # below — the whole literal name as one root-level component —
    # captures exactly those keys into the companion physical slot;
    # breakdown key extraction reads it first, everything else reads
    # the primary (pluck) slot.
    nf_match = len(fields.paths)
    comp_slot = np.full(max(nf_match, 1), -1, dtype=np.int32)
    comp_sigs = []
    no_companions = os.environ.get("DRAGNET_NO_COMPANIONS") == "1"
    if data_format != "json-skinner" and not no_companions:
        for i, (pth, raw) in enumerate(fields.paths):
            if not raw and "." in pth:
                phys = nf_match + len(comp_sigs)
                if phys >= MAX_FIELDS:
                    raise PlanError(
                        "query references more than %d fields "
                        "(incl. dotted-key companions)" % MAX_FIELDS)
                comp_slot[i] = phys
                comp_sigs.append(lit_sig(pth))
    elif data_format == "json-skinner" and not no_companions:
        # skinner: a literal dotted KEY at point.fields top level
        # chains (dot-splits) to the SAME sig as the plucked nested
        # path — correct for the aggregation readout (points.lookup is
        # literal-first) but it must stay INVISIBLE to predicates and
        # synthetic sources (krill pluck).  The device marks
        # fields-root keys that contained a dot by folding
        # SIG_LIT_MARK into their sig, landing them in the companion
        # slot; breakdown readout consults the companion first,
        # everything else reads the primary (pluck) slot.
        for i, (pth, raw) in enumerate(fields.paths):
            if not raw and "." in pth:
                phys = nf_match + len(comp_sigs)
                if phys >= MAX_FIELDS:
                    raise PlanError(
                        "query references more than %d fields "
                        "(incl. dotted-key companions)" % MAX_FIELDS)
                comp_slot[i] = phys
                comp_sigs.append(_sig_mix(
                    path_sig(prefix + pth) ^ SIG_LIT_MARK))

    programs = np.array(prog_nodes, dtype=np.int32).reshape(-1, 4)
    bounds = np.array(prog_bounds, dtype=np.int32).reshape(-1, 2)
    const_meta = np.array(consts.metas, dtype=np.int32).reshape(-1, 6)
    const_dvals = np.array(consts.dvals, dtype=np.float64)
    const_bytes = np.frombuffer(
        bytes(consts.bytes) or b"\0", dtype=np.uint8).copy()
    synthetic = np.array(synth_slots or [0], dtype=np.int32)
    metrics = np.array(metric_rows, dtype=np.int32).reshape(-1, 8)
    bds = np.array(bd_rows or [[0, 0, 0, 0]],
                   dtype=np.int32).reshape(-1, 4)
    steps = np.array(bd_steps or [0.0], dtype=np.float64)
    sreq = np.array(synth_req or [0], dtype=np.int32)

    plan = CompiledPlan(
        fields, (programs, bounds), const_meta, const_dvals,
        const_bytes, synthetic, (metrics, sreq), (bds, steps), queries)
    plan.n_synth = len(synth_slots)
    plan.comp_slot = comp_slot
    plan.nf_match = nf_match
    bloom = 0
    for sg in ([int(x) for x in fields.sigs()] + comp_sigs):
        bloom |= 1 << (sg & 63)
    # int64 for the binding (torch scalar args are signed)
    plan.sig_bloom = bloom - (1 << 64) if bloom >= (1 << 63) else bloom
    if comp_sigs:
        plan.field_sigs = np.concatenate(
            [plan.field_sigs,
             np.array(comp_sigs, dtype=np.uint64)])
    plan.value_slot = value_slot
    plan.fields_slot = fields_slot
    # parent sig of point.fields children (the device detects
    # fields-root literal-dotted keys by comparing parents)
    fps = path_sig("fields") if data_format == "json-skinner" else 0
    plan.fields_parent_sig = fps - (1 << 64) if fps >= (1 << 63) else fps
    plan.ds_prog = ds_prog
    plan.data_format = data_format
    return plan


# ---- output decoding ----

TAG_ORD = 0
TAG_STR = 1
TAG_NUM = 2
TAG_SPECIAL = 3
ORD_BIAS = 1 << 29
SPECIAL_NULL = 0
SPECIAL_UNDEF = 1
SPECIAL_TRUE = 2
SPECIAL_FALSE = 3
SPECIAL_OBJECT = 4
SPECIAL_ARRAY = 5
SPECIAL_ARRJSON = 6  # | (string-dict id << 3)
EMPTY_CODE = 0xFFFFFFFF


def decode_key(codes, query, strings, numbers):
    """Decode one metric key tuple (list of u32 codes) into the CPU
    aggregator's canonical key tuple."""
    from ..points import js_num_str
    out = []
    for b, code in zip(query.breakdowns, codes):
        tag = code >> 30
        val = code & 0x3FFFFFFF
        bk = query.bucketizers.get(b["name"])
        if bk is not None:
            if tag == TAG_ORD:
                out.append(int(val) - ORD_BIAS)
            else:
                assert tag == TAG_NUM
                out.append(int(numbers[val]))
        elif tag == TAG_STR:
            out.append(strings[val])
        elif tag == TAG_NUM:
            num = numbers[val]
            if "date" in b:
                out.append(int(num))
            else:
                out.append(js_num_str(
                    int(num) if float(num).is_integer()
                    and abs(num) < 2**53 else float(num)))
        elif tag == TAG_SPECIAL:
            if (val & 7) == SPECIAL_ARRJSON:
                import json as _json
                from ..points import js_array_str
                raw = strings[val >> 3]
                try:
                    out.append(js_array_str(_json.loads(raw)))
                except ValueError:
                    out.append(raw)
            else:
                out.append({
                    SPECIAL_NULL: "null", SPECIAL_UNDEF: "undefined",
                    SPECIAL_TRUE: "true", SPECIAL_FALSE: "false",
                    SPECIAL_OBJECT: "[object Object]",
                    SPECIAL_ARRAY: "<array>"}[val])
        else:
            raise PlanError("bad code tag %d" % tag)
    return tuple(out)
