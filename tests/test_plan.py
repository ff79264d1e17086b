"""Plan compiler tests (CPU-only: structure + constant parity with
ops/hip/common.h)."""

import os
import re

from dragnet_amd.engine import plan
from dragnet_amd.query import query_load

HDR = os.path.join(os.path.dirname(__file__), "..", "dragnet_amd",
                   "ops", "hip", "common.h")


def test_constants_match_header():
    src = open(HDR).read()

    def cval(name):
        m = re.search(r"\b%s\s*=\s*([0-9a-fA-Fxu<() ]+?)\s*[,;}]" % name,
                      src)
        assert m, name
        return eval(m.group(1).replace("u", ""))

    assert cval("OP_AND") == plan.OP_AND
    assert cval("OP_OR") == plan.OP_OR
    assert cval("OP_EQ") == plan.OP_EQ
    assert cval("OP_TRUE") == plan.OP_TRUE
    assert cval("CONST_NUM") == plan.CONST_NUM
    assert cval("CONST_STR") == plan.CONST_STR
    assert cval("BUCKET_P2") == plan.BUCKET_P2
    assert cval("BUCKET_LIN") == plan.BUCKET_LIN
    assert cval("TAG_STR") == plan.TAG_STR
    assert cval("TAG_NUM") == plan.TAG_NUM
    assert cval("TAG_SPECIAL") == plan.TAG_SPECIAL
    assert cval("ORD_BIAS") == plan.ORD_BIAS
    assert cval("SPECIAL_NULL") == plan.SPECIAL_NULL
    assert cval("SPECIAL_UNDEF") == plan.SPECIAL_UNDEF
    assert cval("MAX_FIELDS") == plan.MAX_FIELDS


def test_program_layout():
    q = query_load(
        filter={"and": [{"eq": ["a", 1]},
                        {"or": [{"lt": ["b", 2]}, {"ge": ["c", "x"]}]}]},
        breakdown_specs="a")
    p = plan.compile_plan([q])
    progs, bounds = p.programs
    # program 0: empty ds filter -> OP_TRUE
    assert progs[bounds[0][0]][0] == plan.OP_TRUE
    # program 1: and(eq, or(lt, ge))
    s, e = bounds[1]
    assert progs[s][0] == plan.OP_AND and progs[s][1] == 2
    assert progs[s][3] == e  # skip offset is global
    assert progs[s + 1][0] == plan.OP_EQ
    assert progs[s + 2][0] == plan.OP_OR
    assert progs[s + 2][3] == e
    assert progs[s + 3][0] == plan.OP_LT
    assert progs[s + 4][0] == plan.OP_GE


def test_field_dedup_and_sigs():
    q1 = query_load(filter={"eq": ["req.method", "GET"]},
                    breakdown_specs="req.method,host")
    p = plan.compile_plan([q1])
    assert len(p.fields.paths) == 2
    assert p.field_sigs[0] == plan.path_sig("req.method")


def test_synth_and_time_bounds():
    q = query_load(
        breakdown_specs="ts[date,field=time,aggr=lquantize,step=60]",
        time_after="2014-05-02", time_before="2014-05-03")
    p = plan.compile_plan([q], time_field="time")
    metrics, sreq = p.metrics
    m = metrics[0]
    assert m[3] == 1          # one (deduped) synthetic requirement
    assert m[5] == 1          # has time filter
    assert m[6] == 1398988800
    assert m[7] == 1399075200
    bds, steps = p.breakdown_descs
    assert bds[0][0] == 1     # synthetic-kind breakdown
    assert bds[0][2] == plan.BUCKET_LIN
    assert steps[0] == 60.0


def test_skinner_plan():
    q = query_load(breakdown_specs="req.method")
    p = plan.compile_plan([q], data_format="json-skinner")
    paths = p.fields.paths
    assert ("req.method", False) in paths
    assert ("value", True) in paths
    assert ("fields", True) in paths
    # prefixed sig
    i = paths.index(("req.method", False))
    assert p.field_sigs[i] == plan.path_sig("fields.req.method")


def test_decode_key():
    q = query_load(breakdown_specs="m,lat[aggr=quantize]")
    strings = ["GET", "PUT"]
    numbers = [200.0]
    code_str = plan.TAG_STR << 30 | 1
    code_ord = plan.TAG_ORD << 30 | (plan.ORD_BIAS + 5)
    assert plan.decode_key([code_str, code_ord], q, strings, numbers) \
        == ("PUT", 5)
    q2 = query_load(breakdown_specs="res.statusCode")
    code_num = plan.TAG_NUM << 30 | 0
    assert plan.decode_key([code_num], q2, strings, numbers) == ("200",)
    code_undef = plan.TAG_SPECIAL << 30 | plan.SPECIAL_UNDEF
    assert plan.decode_key([code_undef], q2, strings, numbers) \
        == ("undefined",)


def test_breakdown_width_guard():
    # > MAX_KEY breakdowns would overflow the device key tuple
    # (uint32_t key[8]); compile_plan must refuse
    import pytest
    specs = ",".join("f%d" % i for i in range(9))
    q = query_load(breakdown_specs=specs)
    with pytest.raises(plan.PlanError):
        plan.compile_plan([q])
    ok = query_load(breakdown_specs=",".join(
        "f%d" % i for i in range(8)))
    plan.compile_plan([ok])  # 8 wide is fine
